"""SQS messenger driver — a real cloud pub/sub transport.

The reference registers awssqs:// (among 6 gocloud drivers) at
internal/manager/run.go:47-53; its messenger consumes subscriptions with
ack/nack semantics (internal/messenger/messenger.go:82-178). This driver
speaks the genuine AWS SQS query protocol over HTTP — SendMessage /
ReceiveMessage (long poll) / DeleteMessage (ack) / ChangeMessageVisibility
(nack => immediate redelivery) — with SigV4 request signing when AWS
credentials are present in the environment, so it works against real SQS
or localstack; tests run it against the in-process FakeSqsServer
(fakesqs.py). Payload bytes ride base64 in MessageBody (SQS bodies are
text-only).

URL forms (broker_from_url):
  awssqs://sqs.<region>.amazonaws.com/<account>/<topic>   (https)
  sqs+http://host:port[/base]/<topic>                     (plain http, dev)
"""
from __future__ import annotations

import asyncio
import base64
import datetime
import hashlib
import hmac
import os
import urllib.parse
import xml.etree.ElementTree as ET
from typing import Optional

import httpx


def _sigv4_headers(method: str, url: str, body: bytes,
                   region: str, service: str = "sqs") -> dict:
    """Minimal AWS Signature Version 4 for form-encoded POSTs."""
    access = os.environ.get("AWS_ACCESS_KEY_ID")
    secret = os.environ.get("AWS_SECRET_ACCESS_KEY")
    if not access or not secret:
        return {}
    u = urllib.parse.urlparse(url)
    now = datetime.datetime.now(datetime.timezone.utc)
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    payload_hash = hashlib.sha256(body).hexdigest()
    headers = {
        "host": u.netloc,
        "x-amz-date": amz_date,
        "content-type": "application/x-www-form-urlencoded",
    }
    signed = ";".join(sorted(headers))
    canonical = "\n".join([
        method, u.path or "/", "",
        "".join(f"{k}:{headers[k]}\n" for k in sorted(headers)),
        signed, payload_hash,
    ])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    to_sign = "\n".join([
        "AWS4-HMAC-SHA256", amz_date, scope,
        hashlib.sha256(canonical.encode()).hexdigest(),
    ])

    def _hmac(key: bytes, msg: str) -> bytes:
        return hmac.new(key, msg.encode(), hashlib.sha256).digest()

    k = _hmac(_hmac(_hmac(_hmac(
        ("AWS4" + secret).encode(), datestamp), region), service),
        "aws4_request")
    sig = hmac.new(k, to_sign.encode(), hashlib.sha256).hexdigest()
    return {
        "X-Amz-Date": amz_date,
        "Authorization": (
            f"AWS4-HMAC-SHA256 Credential={access}/{scope}, "
            f"SignedHeaders={signed}, Signature={sig}"
        ),
    }


class SqsBroker:
    """Broker driver over the SQS HTTP API (same contract as MemBroker /
    FileBroker: publish / receive / receive_with_ack)."""

    def __init__(self, base_url: str, region: str = "us-east-1",
                 wait_seconds: int = 2, visibility_timeout: int = 30):
        self.base = base_url.rstrip("/")
        self.region = region
        self.wait_seconds = wait_seconds
        self.visibility_timeout = visibility_timeout
        self._client: Optional[httpx.AsyncClient] = None

    def _http(self) -> httpx.AsyncClient:
        if self._client is None:
            self._client = httpx.AsyncClient(timeout=self.wait_seconds + 30)
        return self._client

    def queue_url(self, topic: str) -> str:
        return f"{self.base}/{topic}"

    async def _call(self, topic: str, params: dict) -> ET.Element:
        url = self.queue_url(topic)
        body = urllib.parse.urlencode(params).encode()
        headers = {
            "Content-Type": "application/x-www-form-urlencoded",
            **_sigv4_headers("POST", url, body, self.region),
        }
        r = await self._http().post(url, content=body, headers=headers)
        if r.status_code >= 400:
            raise RuntimeError(f"sqs {params.get('Action')}: "
                               f"{r.status_code} {r.text[:200]}")
        return ET.fromstring(r.text)

    # ----------------------------------------------------------- contract
    async def publish(self, topic: str, payload: bytes) -> None:
        await self._call(topic, {
            "Action": "SendMessage", "Version": "2012-11-05",
            "MessageBody": base64.b64encode(payload).decode(),
        })

    async def receive_with_ack(self, topic: str):
        while True:
            root = await self._call(topic, {
                "Action": "ReceiveMessage", "Version": "2012-11-05",
                "MaxNumberOfMessages": "1",
                "WaitTimeSeconds": str(self.wait_seconds),
                "VisibilityTimeout": str(self.visibility_timeout),
            })
            msg = root.find(".//{*}Message") or root.find(".//Message")
            if msg is None:
                continue  # empty long poll; poll again

            def _text(tag: str) -> str:
                el = msg.find(f"{{*}}{tag}")
                if el is None:
                    el = msg.find(tag)
                return el.text or ""

            payload = base64.b64decode(_text("Body"))
            handle = _text("ReceiptHandle")
            loop = asyncio.get_running_loop()

            def ack() -> None:
                loop.create_task(self._delete(topic, handle))

            def nack() -> None:
                loop.create_task(self._release(topic, handle))

            return payload, ack, nack

    async def receive(self, topic: str) -> bytes:
        payload, ack, _ = await self.receive_with_ack(topic)
        ack()
        return payload

    async def _delete(self, topic: str, handle: str) -> None:
        try:
            await self._call(topic, {
                "Action": "DeleteMessage", "Version": "2012-11-05",
                "ReceiptHandle": handle,
            })
        except Exception:  # noqa: BLE001 (ack best-effort; redelivery ok)
            pass

    async def _release(self, topic: str, handle: str) -> None:
        try:
            await self._call(topic, {
                "Action": "ChangeMessageVisibility", "Version": "2012-11-05",
                "ReceiptHandle": handle, "VisibilityTimeout": "0",
            })
        except Exception:  # noqa: BLE001
            pass

    async def close(self) -> None:
        if self._client is not None:
            await self._client.aclose()
