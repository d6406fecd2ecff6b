"""Model proxy — parse, scale-from-zero, route, forward with retries.

Reference parity (internal/modelproxy/handler.go):
  parse -> active-request metric +- -> ScaleAtLeastOneReplica ->
  AwaitBestAddress -> reverse proxy to http://<addr> -> retries <=3 on
  {500,502,503,504} or connection error (handler.go:127-158); body kept
  re-readable; streaming responses pass through untouched.
"""
from __future__ import annotations

import dataclasses
import json
import re
from typing import Optional

import httpx
from starlette.requests import Request
from starlette.responses import JSONResponse, Response, StreamingResponse

from .apiutils import APIError, parse_request
from .loadbalancer import LoadBalancer
from .modelclient import ModelClient

RETRYABLE = {500, 502, 503, 504}


class ProxyHandler:
    def __init__(
        self,
        model_client: ModelClient,
        lb: LoadBalancer,
        max_retries: int = 3,
        timeout: float = 600.0,
        endpoint_timeout: float = 120.0,
        priority_classes: Optional[dict] = None,
    ):
        self.model_client = model_client
        self.lb = lb
        self.max_retries = max_retries
        self.endpoint_timeout = endpoint_timeout
        # k8s PriorityClass analog: name -> importance (higher = more
        # important); mapped onto engine admission priority (lower-first)
        self.priority_classes = priority_classes or {}
        self.client = httpx.AsyncClient(timeout=timeout)

    async def close(self) -> None:
        await self.client.aclose()

    async def handle(self, request: Request, path: str) -> Response:
        # W3C TraceContext server span (reference wraps handlers in
        # otelhttp route tags; tracing.py is the exporter-less analog).
        # For streaming responses the span covers time-to-headers.
        from .tracing import TRACER, parse_traceparent

        parent = parse_traceparent(request.headers.get("traceparent"))
        span = TRACER.start_span(f"proxy {path}", parent)
        span.set("http.route", path)
        request.state.trace_ctx = span.ctx
        try:
            resp = await self._handle_inner(request, path)
            span.set("http.status_code", getattr(resp, "status_code", 0))
            return resp
        except BaseException:
            span.set("error", True)
            raise
        finally:
            span.end()

    async def _handle_inner(self, request: Request, path: str) -> Response:
        try:
            raw = await request.body()
            ctype = request.headers.get("content-type", "")
            multipart = ctype.startswith("multipart/form-data")
            if multipart:
                # audio/transcriptions etc. — extract the model form field,
                # forward the multipart body untouched (reference:
                # apiutils/request.go:109-165)
                model_field = _extract_multipart_field(raw, ctype, "model")
                body = {"model": model_field}
            else:
                try:
                    body = json.loads(raw) if raw else {}
                except json.JSONDecodeError:
                    return JSONResponse({"error": "invalid JSON body"}, status_code=400)
            selectors = _parse_selectors(request.headers.get("X-Label-Selector"))
            try:
                pr = parse_request(body, path, self.model_client.lookup_model, selectors)
            except APIError as e:
                return JSONResponse({"error": e.message}, status_code=e.status)
            if multipart:
                pr = dataclasses.replace(pr, body=None)  # raw passthrough
            elif pr.body is not None and "priority" not in pr.body:
                # Model.spec.priorityClassName -> engine admission priority
                # (engine serves LOWER values first; class value is k8s-style
                # higher-is-more-important, hence the negation)
                m = self.model_client.lookup_model(pr.model)
                pcn = m.spec.priority_class_name if m else ""
                if pcn and pcn in self.priority_classes:
                    pr.body["priority"] = -int(self.priority_classes[pcn])

            self.model_client.scale_at_least_one_replica(pr.model)
            payload = raw if pr.body is None else json.dumps(pr.body).encode()
            attempt = 0
            while True:
                attempt += 1
                try:
                    addr, done = await self.lb.await_best_address(
                        pr.model,
                        adapter=pr.adapter,
                        prefix=pr.prefix,
                        timeout=self.endpoint_timeout,
                    )
                except TimeoutError as e:
                    return JSONResponse({"error": str(e)}, status_code=503)
                try:
                    resp = await self._forward(addr, path, payload, request)
                except (httpx.ConnectError, httpx.ReadError, httpx.RemoteProtocolError):
                    done()
                    if attempt <= self.max_retries:
                        continue
                    return JSONResponse(
                        {"error": "backend connection failed"}, status_code=502
                    )
                except BaseException:
                    done()  # never leak in-flight accounting
                    raise
                if resp.status_code in RETRYABLE and attempt <= self.max_retries:
                    await resp.aclose()
                    done()
                    continue
                return self._to_response(resp, done)
        except Exception as e:  # noqa: BLE001
            import traceback

            traceback.print_exc()
            return JSONResponse({"error": "internal proxy error"}, status_code=500)

    async def _forward(self, addr: str, path: str, payload: bytes, request: Request):
        url = f"http://{addr}{path}"
        headers = {
            k: v
            for k, v in request.headers.items()
            if k.lower() in ("content-type", "accept", "authorization", "traceparent")
        }
        headers.setdefault("content-type", "application/json")
        ctx = getattr(request.state, "trace_ctx", None)
        if ctx is not None:
            # engine sees the CHILD context, not the client's original
            headers["traceparent"] = ctx.traceparent()
        req = self.client.build_request("POST", url, content=payload, headers=headers)
        return await self.client.send(req, stream=True)

    def _to_response(self, resp: httpx.Response, done) -> Response:
        async def body_iter():
            try:
                async for chunk in resp.aiter_raw():
                    yield chunk
            finally:
                await resp.aclose()
                done()

        headers = {
            k: v
            for k, v in resp.headers.items()
            if k.lower() in ("content-type", "cache-control")
        }
        return StreamingResponse(
            body_iter(), status_code=resp.status_code, headers=headers
        )


def _extract_multipart_field(raw: bytes, content_type: str, field: str) -> Optional[str]:
    """Minimal multipart/form-data field extraction (no multipart lib in
    this image). Good for small text fields like `model`."""
    pat = re.compile(
        rb'Content-Disposition:\s*form-data;\s*name="'
        + re.escape(field.encode())
        + rb'"\r?\n\r?\n([^\r\n]*)',
        re.IGNORECASE,
    )
    m = pat.search(raw)
    return m.group(1).decode(errors="replace") if m else None


def _parse_selectors(header: Optional[str]) -> Optional[dict[str, str]]:
    if not header:
        return None
    out = {}
    for part in header.split(","):
        k, _, v = part.partition("=")
        if k:
            out[k.strip()] = v.strip()
    return out
