"""Async messaging ingress (reference: internal/messenger/messenger.go).

Message {metadata, path, body} -> parse -> scale-at-least-one -> await
endpoint -> POST to engine -> publish {metadata, status_code, body} with
ack/nack, bounded handler concurrency, consecutive-error backoff.

Drivers: mem:// (tests/local; the reference registers SQS/PubSub/Kafka/...
through gocloud — those transports are deployment concerns; the mem driver
exercises the full flow, reference tests do the same with mem://).
"""
from __future__ import annotations

import asyncio
import json
from typing import Optional

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


class Messenger:
    def __init__(
        self,
        broker: MemBroker,
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
            payload = await self.broker.receive(self.requests_topic)
            await self.sem.acquire()
            asyncio.create_task(self._handle(payload))

    async def _handle(self, payload: bytes) -> None:
        try:
            metadata, status, body = await self.handle_request(payload)
            out = json.dumps(
                {"metadata": metadata, "status_code": status, "body": body}
            ).encode()
            await self.broker.publish(self.responses_topic, out)
            self.consecutive_errors = 0
        except Exception:  # noqa: BLE001
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
