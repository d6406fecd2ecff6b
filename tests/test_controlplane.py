"""Control-plane integration tests — the envtest-analog harness.

Mirrors the reference integration suite (test/integration/): the whole
Manager runs in-process with a FakeRuntime (replicas never actually run;
tests flip readiness manually) and a fake engine backend HTTP server that
ready replicas point at (the reference's pod-address-override trick,
utils_test.go:150-159).
"""
import asyncio
import contextlib
import json
import socket

import httpx
import pytest
import uvicorn
from starlette.applications import Starlette
from starlette.responses import JSONResponse
from starlette.routing import Route

from kubeai_amd.controlplane.config import AutoscalingConfig, SystemConfig
from kubeai_amd.controlplane.crd import AdapterSpec, Model, ModelSpec
from kubeai_amd.controlplane.manager import Manager
from kubeai_amd.controlplane.messenger import Messenger
from kubeai_amd.controlplane.runtime import FakeRuntime
from kubeai_amd.controlplane.store import ReplicaState


def run(coro):
    return asyncio.run(coro)


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class FakeBackend:
    """Counts requests; can fail N times; implements the engine surface the
    control plane touches."""

    def __init__(self):
        self.requests = []
        self.last_headers = {}
        self.waiting = 0      # engine queue depth exposed on /metrics
        self.kv_usage = 0.0   # KV occupancy exposed on /metrics
        self.fail_next = 0
        self.loaded_adapters = set()
        self.port = free_port()
        self.server = None

    @property
    def address(self):
        return f"127.0.0.1:{self.port}"

    def app(self):
        async def completions(request):
            body = await request.json()
            self.requests.append((request.url.path, body))
            self.last_headers = dict(request.headers)
            if self.fail_next > 0:
                self.fail_next -= 1
                return JSONResponse({"error": "boom"}, status_code=500)
            return JSONResponse(
                {
                    "choices": [{"text": "ok", "finish_reason": "stop"}],
                    "model": body.get("model"),
                    "usage": {"prompt_tokens": 2, "completion_tokens": 1,
                              "total_tokens": 3},
                }
            )

        async def load_lora(request):
            body = await request.json()
            name = body.get("lora_name")
            if name in self.loaded_adapters:
                return JSONResponse({"error": "already loaded"}, status_code=400)
            self.loaded_adapters.add(name)
            return JSONResponse({"status": "ok"})

        async def unload_lora(request):
            body = await request.json()
            name = body.get("lora_name")
            if name not in self.loaded_adapters:
                return JSONResponse({"error": "cannot be found"}, status_code=404)
            self.loaded_adapters.discard(name)
            return JSONResponse({"status": "ok"})

        async def metrics(request):
            from starlette.responses import PlainTextResponse

            return PlainTextResponse(
                f'kubeai_engine_num_requests_waiting{{model="m"}} {self.waiting}\n'
                f'kubeai_engine_kv_cache_usage_perc{{model="m"}} {self.kv_usage}\n'
            )

        async def transcriptions(request):
            raw = await request.body()
            self.requests.append((request.url.path, raw))
            return JSONResponse({"text": "transcribed"})

        return Starlette(
            routes=[
                Route("/v1/completions", completions, methods=["POST"]),
                Route("/v1/chat/completions", completions, methods=["POST"]),
                Route("/v1/audio/transcriptions", transcriptions, methods=["POST"]),
                Route("/v1/load_lora_adapter", load_lora, methods=["POST"]),
                Route("/v1/unload_lora_adapter", unload_lora, methods=["POST"]),
                Route("/metrics", metrics, methods=["GET"]),
            ]
        )

    async def start(self):
        config = uvicorn.Config(
            self.app(), host="127.0.0.1", port=self.port, log_level="error"
        )
        self.server = uvicorn.Server(config)
        self._task = asyncio.create_task(self.server.serve())
        while not self.server.started:
            await asyncio.sleep(0.01)

    async def stop(self):
        self.server.should_exit = True
        with contextlib.suppress(Exception):
            await asyncio.wait_for(self._task, timeout=5)


@contextlib.asynccontextmanager
async def harness(models=(), autoscaler_interval=0.05, messaging=False,
                  priority_classes=None):
    cfg = SystemConfig(
        autoscaling=AutoscalingConfig(
            interval_seconds=autoscaler_interval,
            time_window_seconds=autoscaler_interval * 4,
            state_path=None,
        ),
        leader_lock_path=f"/tmp/kubeai-test-{free_port()}.lock",
        priority_classes=priority_classes or {},
    )
    store_runtime = {}
    mgr = Manager(cfg, runtime="placeholder")
    runtime = FakeRuntime(mgr.store, n_gpus=8)
    mgr.runtime = runtime
    mgr.controller.runtime = runtime
    backend = FakeBackend()
    await backend.start()
    for m in models:
        mgr.store.apply_model(m)
    await mgr.start()
    try:
        yield mgr, runtime, backend
    finally:
        await mgr.stop()
        await backend.stop()


async def wait_for(cond, timeout=5.0, interval=0.01):
    loop = asyncio.get_running_loop()
    deadline = loop.time() + timeout
    while loop.time() < deadline:
        v = cond()
        if v:
            return v
        await asyncio.sleep(interval)
    raise AssertionError("condition not met within timeout")


def text_gen_model(name="m1", **kw):
    # scale-down stays OFF by default: with the harness's 0.05 s autoscaler
    # tick, delay 0 means an idle model scales to zero within ~0.2 s, which
    # races tests that mark a replica ready by hand before sending the first
    # request (observed as rare 120 s endpoint-timeout flakes). Tests that
    # exercise scale-down pass an explicit 0.
    kw.setdefault("scale_down_delay_seconds", 300)
    spec = ModelSpec(url="hf://meta/llama-tiny", min_replicas=0, max_replicas=4, **kw)
    return Model(name=name, spec=spec)


# ---------------------------------------------------------------- controller
def test_controller_creates_replicas_to_spec():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 2
        async with harness([m]) as (mgr, runtime, backend):
            reps = await wait_for(
                lambda: len(mgr.store.list_replicas("m1")) == 2 and mgr.store.list_replicas("m1")
            )
            assert all(r.state == ReplicaState.STARTING for r in reps)
            # readiness propagates to status
            for r in reps:
                runtime.mark_ready(r.name, backend.address)
            await wait_for(
                lambda: mgr.store.get_model("m1").status.replicas_ready == 2
            )

    run(body())


def test_controller_scale_down_prefers_not_ready():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 3
        async with harness([m]) as (mgr, runtime, backend):
            reps = await wait_for(
                lambda: len(mgr.store.list_replicas("m1")) == 3
                and mgr.store.list_replicas("m1")
            )
            ready_names = [r.name for r in reps[:2]]
            for n in ready_names:
                runtime.mark_ready(n, backend.address)
            await wait_for(lambda: mgr.store.get_model("m1").status.replicas_ready == 2)
            mgr.store.scale_model("m1", 2)
            await wait_for(lambda: len(mgr.store.list_replicas("m1")) == 2)
            left = {r.name for r in mgr.store.list_replicas("m1")}
            assert left == set(ready_names)  # the not-ready one was deleted

    run(body())


def test_rollout_surge_on_spec_change():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 2
        async with harness([m]) as (mgr, runtime, backend):
            reps = await wait_for(
                lambda: len(mgr.store.list_replicas("m1")) == 2
                and mgr.store.list_replicas("m1")
            )
            old_hash = reps[0].hash
            for r in reps:
                runtime.mark_ready(r.name, backend.address)
            await wait_for(lambda: mgr.store.get_model("m1").status.replicas_ready == 2)
            # change the spec -> new hash; ready out-of-date pods roll one
            # at a time (pod_plan.go:120-142)
            m2 = mgr.store.get_model("m1")
            m2.spec.args = ["--extra-flag"]
            mgr.store.apply_model(m2)
            await wait_for(
                lambda: any(r.hash != old_hash for r in mgr.store.list_replicas("m1"))
            )
            # mark new ones ready as they come; eventually all have new hash
            for _ in range(60):
                for r in mgr.store.list_replicas("m1"):
                    if not r.ready:
                        runtime.mark_ready(r.name, backend.address)
                if all(r.hash != old_hash for r in mgr.store.list_replicas("m1")) and len(
                    mgr.store.list_replicas("m1")
                ) == 2:
                    break
                await asyncio.sleep(0.02)
            reps = mgr.store.list_replicas("m1")
            assert len(reps) == 2 and all(r.hash != old_hash for r in reps)

    run(body())


def test_replica_recovery():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            # replica disappears (process crash analog)
            await runtime.delete(rep.name)
            await wait_for(
                lambda: len(mgr.store.list_replicas("m1")) == 1
                and mgr.store.list_replicas("m1")[0].name != rep.name
            )

    run(body())


def test_model_deletion_removes_replicas():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 2
        async with harness([m]) as (mgr, runtime, backend):
            await wait_for(lambda: len(mgr.store.list_replicas("m1")) == 2)
            mgr.store.delete_model("m1")
            await wait_for(lambda: not mgr.store.list_replicas("m1"))
            await wait_for(lambda: mgr.store.get_model("m1") is None)

    run(body())


def test_adapter_reconcile():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        m.spec.adapters = [AdapterSpec(name="ad1", url="hf://a/b")]
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            runtime.mark_ready(rep.name, backend.address)
            await wait_for(lambda: "ad1" in backend.loaded_adapters)
            await wait_for(
                lambda: "ad1" in mgr.store.get_replica(rep.name).adapters
            )
            # remove the adapter from the spec -> unload
            m2 = mgr.store.get_model("m1")
            m2.spec.adapters = []
            mgr.store.apply_model(m2)
            await wait_for(lambda: "ad1" not in backend.loaded_adapters)

    run(body())


# ---------------------------------------------------------------- gateway
def test_proxy_full_flow_scale_from_zero():
    async def body():
        m = text_gen_model()  # replicas None -> min 0
        async with harness([m]) as (mgr, runtime, backend):
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw"
            ) as client:
                async def send():
                    return await client.post(
                        "/openai/v1/completions",
                        json={"model": "m1", "prompt": "hello", "max_tokens": 4},
                    )

                task = asyncio.create_task(send())
                # scale-from-zero: request must trigger replica creation
                rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
                assert mgr.store.get_model("m1").spec.replicas == 1
                runtime.mark_ready(rep.name, backend.address)
                resp = await asyncio.wait_for(task, timeout=10)
                assert resp.status_code == 200
                assert resp.json()["choices"][0]["text"] == "ok"
                assert backend.requests[0][0] == "/v1/completions"

    run(body())


def test_proxy_retries_on_500():
    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            runtime.mark_ready(rep.name, backend.address)
            backend.fail_next = 2
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(transport=transport, base_url="http://gw") as client:
                resp = await client.post(
                    "/openai/v1/completions",
                    json={"model": "m1", "prompt": "x", "max_tokens": 1},
                )
            assert resp.status_code == 200  # retried past the two 500s
            assert len(backend.requests) == 3

    run(body())


def test_proxy_unknown_model_404():
    async def body():
        async with harness([]) as (mgr, runtime, backend):
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(transport=transport, base_url="http://gw") as client:
                resp = await client.post(
                    "/openai/v1/completions", json={"model": "ghost", "prompt": "x"}
                )
            assert resp.status_code == 404

    run(body())


def test_models_listing_by_feature():
    async def body():
        m1 = text_gen_model("gen")
        m2 = text_gen_model("emb")
        m2.spec.features = ["TextEmbedding"]
        m1.spec.adapters = [AdapterSpec(name="a1", url="hf://x/y")]
        async with harness([m1, m2]) as (mgr, runtime, backend):
            await wait_for(
                lambda: mgr.store.get_model("gen").labels
                and mgr.store.get_model("emb").labels
            )
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(transport=transport, base_url="http://gw") as client:
                r = await client.get("/openai/v1/models")
                ids = [d["id"] for d in r.json()["data"]]
                assert "gen" in ids and "gen_a1" in ids and "emb" not in ids
                r = await client.get("/openai/v1/models?feature=TextEmbedding")
                ids = [d["id"] for d in r.json()["data"]]
                assert ids == ["emb"]

    run(body())


def test_multi_model_routing():
    """Two Models, separate replica sets; the gateway routes by model name
    and in-flight accounting stays per-model."""

    async def body():
        m1, m2 = text_gen_model("alpha"), text_gen_model("beta")
        m1.spec.replicas = 1
        m2.spec.replicas = 1
        async with harness([m1, m2]) as (mgr, runtime, backend):
            backend2 = FakeBackend()
            await backend2.start()
            try:
                r1 = (await wait_for(lambda: mgr.store.list_replicas("alpha")))[0]
                r2 = (await wait_for(lambda: mgr.store.list_replicas("beta")))[0]
                runtime.mark_ready(r1.name, backend.address)
                runtime.mark_ready(r2.name, backend2.address)
                transport = httpx.ASGITransport(app=mgr.app)
                async with httpx.AsyncClient(
                    transport=transport, base_url="http://gw"
                ) as client:
                    for name in ("alpha", "beta", "alpha"):
                        resp = await client.post(
                            "/openai/v1/completions",
                            json={"model": name, "prompt": "x", "max_tokens": 1},
                        )
                        assert resp.status_code == 200
                assert len(backend.requests) == 2   # alpha twice
                assert len(backend2.requests) == 1  # beta once
            finally:
                await backend2.stop()

    run(body())


def test_proxy_multipart_audio_routing():
    """SpeechToText path: multipart body, model extracted from the form
    field, raw body forwarded (reference: apiutils/request.go:109-165)."""

    async def body():
        m = text_gen_model("whisper")
        m.spec.features = ["SpeechToText"]
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("whisper")))[0]
            runtime.mark_ready(rep.name, backend.address)
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(transport=transport, base_url="http://gw") as client:
                resp = await client.post(
                    "/openai/v1/audio/transcriptions",
                    files={"file": ("a.wav", b"RIFFxxxx", "audio/wav")},
                    data={"model": "whisper"},
                )
            assert resp.status_code == 200
            assert resp.json()["text"] == "transcribed"
            path, raw = backend.requests[-1]
            assert path == "/v1/audio/transcriptions"
            assert b"RIFFxxxx" in raw  # multipart forwarded untouched

    run(body())


# ---------------------------------------------------------------- autoscaler
def test_autoscaler_scales_up_and_down_to_zero():
    async def body():
        m = text_gen_model(scale_down_delay_seconds=0,)
        m.spec.target_requests = 2
        async with harness([m]) as (mgr, runtime, backend):
            from kubeai_amd.controlplane import metrics

            # simulate 8 active requests -> avg fills towards 8 -> ceil(8/2)=4
            metrics.INFERENCE_REQUESTS_ACTIVE.labels("m1").inc(8)
            await wait_for(
                lambda: (mgr.store.get_model("m1").spec.replicas or 0) >= 1,
                timeout=5,
            )
            # window avg over 4 slots of 8 = 8 -> 4 replicas
            await wait_for(
                lambda: (mgr.store.get_model("m1").spec.replicas or 0) == 4,
                timeout=5,
            )
            # load disappears -> average decays -> scale to zero
            metrics.INFERENCE_REQUESTS_ACTIVE.labels("m1").dec(8)
            await wait_for(
                lambda: (mgr.store.get_model("m1").spec.replicas or 0) == 0,
                timeout=10,
            )

    run(body())


# ---------------------------------------------------------------- messenger
def test_messenger_flow():
    async def body():
        m = text_gen_model()
        async with harness([m]) as (mgr, runtime, backend):
            msgr = Messenger(
                mgr.broker, "req", "resp", mgr.model_client, mgr.lb, max_handlers=2
            )
            msgr.start()
            try:
                await mgr.broker.publish(
                    "req",
                    json.dumps(
                        {
                            "metadata": {"id": "42"},
                            "path": "/v1/completions",
                            "body": {"model": "m1", "prompt": "via bus"},
                        }
                    ).encode(),
                )
                rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
                runtime.mark_ready(rep.name, backend.address)
                out = json.loads(
                    await asyncio.wait_for(mgr.broker.receive("resp"), timeout=10)
                )
                assert out["metadata"] == {"id": "42"}
                assert out["status_code"] == 200
                assert out["body"]["choices"][0]["text"] == "ok"
            finally:
                await msgr.stop()

    run(body())


def test_proxy_inflight_never_leaks_on_unexpected_error():
    """Any exception between endpoint acquisition and response must release
    the in-flight slot (the autoscaling signal would otherwise drift up)."""

    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            runtime.mark_ready(rep.name, backend.address)

            async def boom(*a, **kw):
                raise RuntimeError("injected")

            mgr.proxy._forward = boom
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(transport=transport, base_url="http://gw") as client:
                resp = await client.post(
                    "/openai/v1/completions",
                    json={"model": "m1", "prompt": "x", "max_tokens": 1},
                )
            assert resp.status_code == 500
            await wait_for(lambda: mgr.lb.group("m1").total_in_flight == 0)
            for ep in mgr.lb.group("m1").endpoints.values():
                assert ep.in_flight == 0

    run(body())


def test_priority_class_maps_to_request_priority():
    """Model.spec.priorityClassName + config priorityClasses -> the proxy
    injects the engine admission `priority` field (higher class value =
    more important = lower engine priority); explicit request priority
    wins."""
    async def body():
        m = text_gen_model("prio-m")
        m.spec.priority_class_name = "critical"
        m.spec.replicas = 1
        async with harness([m], priority_classes={"critical": 1000}) as (
            mgr, runtime, backend,
        ):
            rep = (await wait_for(lambda: mgr.store.list_replicas("prio-m")))[0]
            runtime.mark_ready(rep.name, backend.address)
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw"
            ) as client:
                r = await client.post(
                    "/openai/v1/completions",
                    json={"model": "prio-m", "prompt": "x", "max_tokens": 1},
                )
                assert r.status_code == 200
                assert backend.requests[-1][1]["priority"] == -1000
                r = await client.post(
                    "/openai/v1/completions",
                    json={"model": "prio-m", "prompt": "x", "max_tokens": 1,
                          "priority": 7},
                )
                assert r.status_code == 200
                assert backend.requests[-1][1]["priority"] == 7

    run(body())


def test_cache_lifecycle_and_finalizer(tmp_path):
    """cacheProfile: reconcile creates the shared-dir cache + .loaded marker,
    adds the eviction finalizer, status reflects loaded; deletion evicts the
    dir and removes the model (reference: cache.go + finalizer flow)."""
    import os

    from kubeai_amd.controlplane.crd import CACHE_EVICTION_FINALIZER

    async def body():
        m = text_gen_model("cached-m", cache_profile="shared-fs")
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            mgr.controller.cache.base_dir = str(tmp_path)
            await wait_for(lambda: mgr.store.list_replicas("cached-m"))
            model = mgr.store.get_model("cached-m")
            assert CACHE_EVICTION_FINALIZER in model.finalizers
            assert model.status.cache_loaded
            d = mgr.controller.cache.model_dir(model)
            assert os.path.exists(os.path.join(d, ".loaded"))
            # delete -> evict + remove
            mgr.store.delete_model("cached-m")
            await wait_for(lambda: mgr.store.get_model("cached-m") is None)
            assert not os.path.exists(d)

    run(body())


def test_proxy_preserves_unknown_fields():
    """vLLM-extension / vendor fields in the request body survive the proxy
    re-marshal (reference: jsontext unknown-field passthrough)."""
    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            runtime.mark_ready(rep.name, backend.address)
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw"
            ) as client:
                r = await client.post(
                    "/openai/v1/completions",
                    json={"model": "m1", "prompt": "x", "max_tokens": 1,
                          "guided_json": {"type": "object"},
                          "vendor_ext": [1, 2, 3]},
                )
                assert r.status_code == 200
                seen = backend.requests[-1][1]
                assert seen["guided_json"] == {"type": "object"}
                assert seen["vendor_ext"] == [1, 2, 3]

    run(body())


def test_failed_replica_recreated():
    """A replica whose engine process died (state=Failed via the health
    monitor) must be deleted and replaced by the reconciler."""
    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
            mgr.store.update_replica(rep.name, state=ReplicaState.FAILED)
            await wait_for(
                lambda: (lambda reps: len(reps) == 1
                         and reps[0].name != rep.name
                         and reps[0].state != ReplicaState.FAILED)(
                    mgr.store.list_replicas("m1"))
            )

    run(body())


def test_messenger_flow_file_broker(tmp_path):
    """Same messenger flow over the durable file:// driver: the request and
    response ride the shared directory instead of the in-memory queue."""
    from kubeai_amd.controlplane.messenger import broker_from_url

    async def body():
        m = text_gen_model()
        m.spec.replicas = 1
        async with harness([m]) as (mgr, runtime, backend):
            broker = broker_from_url(f"file://{tmp_path}/bus")
            msgr = Messenger(
                broker, "req", "resp", mgr.model_client, mgr.lb, max_handlers=2
            )
            msgr.start()
            try:
                rep = (await wait_for(lambda: mgr.store.list_replicas("m1")))[0]
                runtime.mark_ready(rep.name, backend.address)
                await broker.publish(
                    "req",
                    json.dumps(
                        {
                            "metadata": {"id": "f1"},
                            "path": "/v1/completions",
                            "body": {"model": "m1", "prompt": "via file bus"},
                        }
                    ).encode(),
                )
                out = json.loads(
                    await asyncio.wait_for(broker.receive("resp"), timeout=10)
                )
                assert out["metadata"] == {"id": "f1"}
                assert out["status_code"] == 200
            finally:
                await msgr.stop()

    run(body())


def test_autoscaler_kv_pressure_scales_up():
    """A replica at >=95% KV occupancy counts as one extra replica of load
    even with an empty queue (north-star: queue depth + KV occupancy)."""
    async def body():
        m = text_gen_model("kvm")
        m.spec.replicas = 1
        m.spec.max_replicas = 4
        m.spec.target_requests = 10
        async with harness([m], autoscaler_interval=0.05) as (
            mgr, runtime, backend,
        ):
            rep = (await wait_for(lambda: mgr.store.list_replicas("kvm")))[0]
            runtime.mark_ready(rep.name, backend.address)
            backend.kv_usage = 0.99  # full KV pool, nothing waiting
            # moving average fills toward target_requests -> ceil -> 2
            await wait_for(
                lambda: mgr.store.get_model("kvm").spec.replicas >= 2,
                timeout=10,
            )

    run(body())


def test_trace_propagation_and_debug_traces():
    """Gateway records a server span per request, forwards a CHILD
    traceparent to the engine, and serves /debug/traces (reference
    installs W3C propagators with no exporter — otel.go:76-81)."""
    import asyncio

    from kubeai_amd.controlplane.tracing import parse_traceparent

    async def main():
        async with harness(
            models=[
                Model(name="tr-model",
                      spec=ModelSpec(url="hf://org/m", min_replicas=1))
            ]
        ) as (mgr, runtime, backend):
            await wait_for(lambda: mgr.store.list_replicas(model="tr-model"))
            rep = mgr.store.list_replicas(model="tr-model")[0]
            runtime.mark_ready(rep.name, backend.address)
            from httpx import ASGITransport, AsyncClient

            async with AsyncClient(
                transport=ASGITransport(app=mgr.app), base_url="http://t"
            ) as client:
                tp = "00-" + "ab" * 16 + "-" + "cd" * 8 + "-01"
                r = await client.post(
                    "/openai/v1/completions",
                    json={"model": "tr-model", "prompt": "x", "max_tokens": 2},
                    headers={"traceparent": tp},
                )
                assert r.status_code == 200
                # the engine received a CHILD of the client's trace
                fwd = backend.last_headers.get("traceparent")
                ctx = parse_traceparent(fwd)
                assert ctx is not None
                assert ctx.trace_id == "ab" * 16
                assert ctx.span_id != "cd" * 8
                # span recorded with route + status
                tr = await client.get("/debug/traces")
                spans = tr.json()["spans"]
                mine = [s for s in spans if s["traceId"] == "ab" * 16]
                assert mine and mine[0]["parentSpanId"] == "cd" * 8
                assert mine[0]["attributes"]["http.status_code"] == 200

    asyncio.run(main())
