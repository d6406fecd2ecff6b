"""Async messaging ingress (reference: internal/messenger/messenger.go).

Message {metadata, path, body} -> parse -> scale-at-least-one -> await
endpoint -> POST to engine -> publish {metadata, status_code, body} with
ack/nack, bounded handler concurrency, consecutive-error backoff.

Drivers are selected by URL scheme like the reference's gocloud wiring
(internal/messenger registers mem/SQS/PubSub/Kafka by URL): `mem://topic`
is in-process (tests/local, reference tests use the same) and
`file:///dir/topic` is a durable cross-process queue over a shared
directory (single-node multi-process deployments; survives restarts).
Cloud transports (SQS/Kafka) are deployment concerns behind the same
Broker protocol.
"""
from __future__ import annotations

import asyncio
import json
import os
import tempfile
from typing import Optional
from urllib.parse import urlparse

import httpx

from .apiutils import APIError, parse_request
from .loadbalancer import LoadBalancer
from .modelclient import ModelClient


class MemBroker:
    """In-memory pubsub (mem:// analog)."""

    def __init__(self):
        self.topics: dict[str, asyncio.Queue] = {}

    def topic(self, name: str) -> asyncio.Queue:
        q = self.topics.get(name)
        if q is None:
            q = asyncio.Queue()
            self.topics[name] = q
        return q

    async def publish(self, topic: str, payload: bytes) -> None:
        self.topic(topic).put_nowait(payload)

    async def receive(self, topic: str) -> bytes:
        return await self.topic(topic).get()

    async def receive_with_ack(self, topic: str):
        """(payload, ack, nack): nack requeues for redelivery (the
        reference's msg.Nack() analog, messenger.go ack/nack flow)."""
        payload = await self.topic(topic).get()

        def ack() -> None:
            pass

        def nack() -> None:
            self.topic(topic).put_nowait(payload)

        return payload, ack, nack


class FileBroker:
    """Durable cross-process pubsub over a shared directory (file:// driver).

    A topic is a directory; a message is one file, written atomically
    (tmp + rename within the topic dir). A consumer CLAIMS a message by
    renaming it to *.claimed.<pid> — rename is atomic on POSIX, so exactly
    one of N competing consumer processes wins each message.
    """

    def __init__(self, root: str, poll_interval: float = 0.05):
        self.root = root
        self.poll_interval = poll_interval
        self._seq = 0
        os.makedirs(root, exist_ok=True)

    def _topic_dir(self, topic: str) -> str:
        d = os.path.join(self.root, topic.strip("/").replace("/", "_"))
        os.makedirs(d, exist_ok=True)
        return d

    async def publish(self, topic: str, payload: bytes) -> None:
        d = self._topic_dir(topic)
        self._seq += 1
        # monotonic-ish lexicographic name: (coarse clock, pid, seq)
        name = f"{int(asyncio.get_event_loop().time() * 1e6):018d}-{os.getpid()}-{self._seq:06d}.msg"
        fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
        with os.fdopen(fd, "wb") as f:
            f.write(payload)
        os.replace(tmp, os.path.join(d, name))

    async def receive_with_ack(self, topic: str):
        """(payload, ack, nack): the claim is the rename; ack deletes the
        claimed file, nack renames it back (redelivery, original order)."""
        d = self._topic_dir(topic)
        while True:
            for name in sorted(os.listdir(d)):
                if not name.endswith(".msg"):
                    continue
                src = os.path.join(d, name)
                claimed = f"{src}.claimed.{os.getpid()}"
                try:
                    os.rename(src, claimed)  # atomic claim; loser gets ENOENT
                except OSError:
                    continue
                with open(claimed, "rb") as f:
                    payload = f.read()

                def ack(p=claimed):
                    try:
                        os.unlink(p)
                    except OSError:
                        pass

                def nack(p=claimed, orig=src):
                    try:
                        os.rename(p, orig)
                    except OSError:
                        pass

                return payload, ack, nack
            await asyncio.sleep(self.poll_interval)

    async def receive(self, topic: str) -> bytes:
        payload, ack, _ = await self.receive_with_ack(topic)
        ack()
        return payload


def broker_from_url(url: str):
    """gocloud-style driver selection (reference registers its drivers at
    internal/manager/run.go:47-53): mem:// | file:///dir | awssqs://host/
    account (https, SigV4 from env creds) | sqs+http://host:port (dev)."""
    u = urlparse(url)
    if u.scheme in ("", "mem"):
        return MemBroker()
    if u.scheme == "file":
        return FileBroker(u.path or "/tmp/kubeai-msgs")
    if u.scheme in ("awssqs", "sqs", "sqs+http"):
        from .sqs import SqsBroker

        proto = "http" if u.scheme == "sqs+http" else "https"
        # last path segment is the queue/topic; the rest is the base URL
        base_path = u.path.rsplit("/", 1)[0] if u.path.strip("/") else ""
        region = "us-east-1"
        if u.hostname and u.hostname.startswith("sqs."):
            region = u.hostname.split(".")[1]
        return SqsBroker(f"{proto}://{u.netloc}{base_path}", region=region)
    raise ValueError(
        f"unknown messenger driver: {u.scheme}:// (have mem, file, awssqs, sqs+http)"
    )


def stream_transport(requests_url: str, responses_url: str):
    """Resolve one messaging stream to (broker, requests_topic,
    responses_topic). For file:// the directory is the broker root and the
    last path segment the topic, so both topics share one root dir."""
    u = urlparse(requests_url)
    if u.scheme == "file":
        root = os.path.dirname(u.path) or "/tmp/kubeai-msgs"
        return (
            FileBroker(root),
            os.path.basename(u.path),
            os.path.basename(urlparse(responses_url).path),
        )
    if u.scheme in ("awssqs", "sqs", "sqs+http"):
        # queue name = last path segment; both queues share the base URL
        broker = broker_from_url(requests_url)
        return (
            broker,
            u.path.rstrip("/").rsplit("/", 1)[-1],
            urlparse(responses_url).path.rstrip("/").rsplit("/", 1)[-1],
        )
    broker = broker_from_url(requests_url)
    return (
        broker,
        requests_url.split("://", 1)[-1],
        responses_url.split("://", 1)[-1],
    )


class Messenger:
    def __init__(
        self,
        broker,  # any Broker driver (MemBroker / FileBroker)
        requests_topic: str,
        responses_topic: str,
        model_client: ModelClient,
        lb: LoadBalancer,
        max_handlers: int = 8,
        endpoint_timeout: float = 120.0,
    ):
        self.broker = broker
        self.requests_topic = requests_topic
        self.responses_topic = responses_topic
        self.model_client = model_client
        self.lb = lb
        self.sem = asyncio.Semaphore(max_handlers)
        self.endpoint_timeout = endpoint_timeout
        self._client = httpx.AsyncClient(timeout=600.0)
        self._task: Optional[asyncio.Task] = None
        self.consecutive_errors = 0

    def start(self) -> None:
        self._task = asyncio.create_task(self._loop())

    async def stop(self) -> None:
        if self._task:
            self._task.cancel()
        await self._client.aclose()

    async def _loop(self) -> None:
        while True:
            if hasattr(self.broker, "receive_with_ack"):
                payload, ack, nack = await self.broker.receive_with_ack(
                    self.requests_topic
                )
            else:
                payload = await self.broker.receive(self.requests_topic)
                ack = nack = lambda: None
            await self.sem.acquire()
            asyncio.create_task(self._handle(payload, ack, nack))

    async def _handle(self, payload: bytes, ack=lambda: None,
                      nack=lambda: None) -> None:
        try:
            metadata, status, body = await self.handle_request(payload)
            out = json.dumps(
                {"metadata": metadata, "status_code": status, "body": body}
            ).encode()
            await self.broker.publish(self.responses_topic, out)
            self.consecutive_errors = 0
            ack()
        except Exception:  # noqa: BLE001
            # nack -> redelivery (reference: messenger.go nack + linear
            # consecutive-error backoff)
            nack()
            self.consecutive_errors += 1
            await asyncio.sleep(min(self.consecutive_errors, 3))
        finally:
            self.sem.release()

    async def handle_request(self, payload: bytes):
        msg = json.loads(payload)
        metadata = msg.get("metadata", {})
        path = msg.get("path", "/v1/chat/completions")
        body = msg.get("body", {})
        try:
            pr = parse_request(body, path, self.model_client.lookup_model)
        except APIError as e:
            return metadata, e.status, {"error": e.message}
        self.model_client.scale_at_least_one_replica(pr.model)
        addr, done = await self.lb.await_best_address(
            pr.model, adapter=pr.adapter, prefix=pr.prefix,
            timeout=self.endpoint_timeout,
        )
        try:
            resp = await self._client.post(f"http://{addr}{path}", json=pr.body)
            try:
                body_out = resp.json()
            except Exception:
                body_out = {"raw": resp.text}
            return metadata, resp.status_code, body_out
        finally:
            done()
