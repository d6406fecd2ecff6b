"""OpenAI gateway mux (reference: internal/openaiserver/handler.go).

Routes under /openai/v1/* go to the proxy; /openai/v1/models lists Models
by feature label + adapters as "model_adapter" entries (models.go:13-109).
Admin paths (/v1/load_lora_adapter etc.) are deliberately NOT proxied
(security note, handler.go:33-36).
"""
from __future__ import annotations

from starlette.applications import Starlette
from starlette.requests import Request
from starlette.responses import JSONResponse, PlainTextResponse, Response
from starlette.routing import Route

from . import metrics
from .crd import FEATURE_LABEL_DOMAIN
from .modelclient import ModelClient
from .proxy import ProxyHandler, _parse_selectors

PROXIED_PATHS = (
    "/v1/chat/completions",
    "/v1/completions",
    "/v1/embeddings",
    "/v1/rerank",
    "/v1/audio/transcriptions",
)


def build_gateway_app(model_client: ModelClient, proxy: ProxyHandler) -> Starlette:
    async def models(request: Request) -> Response:
        feature = request.query_params.get("feature", "TextGeneration")
        selectors = _parse_selectors(request.headers.get("X-Label-Selector")) or {}
        selectors[f"{FEATURE_LABEL_DOMAIN}/{feature}"] = "true"
        data = []
        for m in model_client.list_all_models():
            if any(m.labels.get(k) != v for k, v in selectors.items()):
                continue
            data.append(
                {"id": m.name, "object": "model", "owned_by": m.spec.owner or "kubeai",
                 "features": m.spec.features}
            )
            for a in m.spec.adapters:
                data.append(
                    {
                        "id": f"{m.name}_{a.name}",
                        "object": "model",
                        "owned_by": m.spec.owner or "kubeai",
                        "parent": m.name,
                    }
                )
        return JSONResponse({"object": "list", "data": data})

    async def proxied(request: Request) -> Response:
        path = request.url.path
        assert path.startswith("/openai")
        return await proxy.handle(request, path[len("/openai") :])

    async def metrics_endpoint(request: Request) -> Response:
        return PlainTextResponse(metrics.render().decode())

    async def healthz(request: Request) -> Response:
        return JSONResponse({"status": "ok"})

    async def debug_traces(request: Request) -> Response:
        # exporter-less trace inspection (tracing.py ring buffer)
        from .tracing import TRACER

        limit = int(request.query_params.get("limit", "100"))
        return JSONResponse({"spans": TRACER.recent(limit)})

    routes = [
        Route("/openai/v1/models", models, methods=["GET"]),
        Route("/metrics", metrics_endpoint, methods=["GET"]),
        Route("/healthz", healthz, methods=["GET"]),
        Route("/debug/traces", debug_traces, methods=["GET"]),
    ]
    for p in PROXIED_PATHS:
        routes.append(Route("/openai" + p, proxied, methods=["POST"]))
    return Starlette(routes=routes)
