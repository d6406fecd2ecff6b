"""K8s-backed control plane against the in-process fake API server — the
reference's envtest tier (test/integration/main_test.go:83-157): Models
persist as CRs, replicas are Pods whose readiness tests flip by PATCHing
status (utils_test.go:118-132), leadership is a coordination.k8s.io Lease.
"""
import asyncio
import time

import pytest

from kubeai_amd.controlplane.controller import ModelController
from kubeai_amd.controlplane.crd import (POD_HASH_LABEL, POD_MODEL_LABEL,
                                         Model, ModelSpec)
from kubeai_amd.controlplane.fakekube import FakeKubeApiServer
from kubeai_amd.controlplane.kubeclient import KubeClient
from kubeai_amd.controlplane.kubestore import (MODELS, PODS, KubeRuntime,
                                               KubeStore, LeaseElection)


@pytest.fixture()
def api():
    srv = FakeKubeApiServer().start()
    yield srv
    srv.stop()


def kc_for(api) -> KubeClient:
    return KubeClient(api_url=api.url, namespace="kubeai")


async def wait_for(cond, timeout=10.0, msg="condition"):
    t0 = time.monotonic()
    while time.monotonic() - t0 < timeout:
        v = cond()
        if v:
            return v
        await asyncio.sleep(0.02)
    raise AssertionError(f"timed out waiting for {msg}")


def mark_pod_ready(kc: KubeClient, name: str, ip: str = "10.0.0.5") -> None:
    # the envtest pattern: tests write pod status; no kubelet exists
    kc.patch_merge(
        kc.path(*PODS, name=name, subresource="status"),
        {"status": {"phase": "Running", "podIP": ip,
                    "conditions": [{"type": "Ready", "status": "True"}]}},
    )


# ----------------------------------------------------------------------
def test_model_cr_persistence(api):
    """Models survive a control-plane restart (CRs live in the API
    server, unlike the in-memory store)."""

    async def main():
        store = KubeStore(kc_for(api))
        await store.start()
        store.apply_model(
            Model(name="m1", spec=ModelSpec(url="hf://org/m1", min_replicas=1))
        )
        store.stop()
        # "restart": a brand-new store against the same API server
        store2 = KubeStore(kc_for(api))
        await store2.start()
        assert "m1" in store2.models
        assert store2.models["m1"].spec.url == "hf://org/m1"
        assert store2.models["m1"].uid
        store2.stop()

    asyncio.run(main())


def test_controller_reconciles_pods(api):
    """Model CR -> engine Pods with the reference contract (labels,
    port annotation, /health probes); readiness flows back to status."""

    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        runtime = KubeRuntime(store)
        ctrl = ModelController(store, runtime)
        ctrl.start()
        try:
            store.apply_model(
                Model(name="m1", spec=ModelSpec(
                    url="hf://org/m1", min_replicas=2,
                    resource_profile="amd-gpu-mi355x:1",
                ))
            )
            pods = await wait_for(
                lambda: (kc.list(kc.path(*PODS),
                                 {POD_MODEL_LABEL: "m1"}) or None)
                and len(kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m1"})) == 2
                and kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m1"}),
                msg="2 pods",
            )
            pod = pods[0]
            # contract checks (reference engine_vllm.go:12-138)
            assert pod["metadata"]["labels"][POD_HASH_LABEL]
            assert pod["metadata"]["annotations"]["model-pod-port"] == "8000"
            c = pod["spec"]["containers"][0]
            assert c["readinessProbe"]["httpGet"]["path"] == "/health"
            assert c["startupProbe"]["failureThreshold"] == 5400
            assert c["resources"]["requests"]["amd.com/gpu"] == "1"
            # feature labels persisted onto the CR
            cr = kc.get(kc.path(*MODELS, name="m1"))
            assert cr["metadata"]["labels"][
                "features.kubeai.org/TextGeneration"] == "true"

            for p in pods:
                mark_pod_ready(kc, p["metadata"]["name"])
            await wait_for(
                lambda: store.models["m1"].status.replicas_ready == 2,
                msg="status ready=2",
            )
            # status also persisted on the CR
            cr = kc.get(kc.path(*MODELS, name="m1"))
            assert cr["status"]["replicas"]["ready"] == 2
            # replica address from podIP + port annotation
            reps = store.list_replicas(model="m1")
            assert all(r.address == "10.0.0.5:8000" for r in reps)
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_scale_subresource_and_scale_down(api):
    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        ctrl = ModelController(store, KubeRuntime(store))
        ctrl.start()
        try:
            store.apply_model(
                Model(name="m2", spec=ModelSpec(url="hf://org/m2",
                                                min_replicas=0, replicas=2))
            )
            await wait_for(
                lambda: len(kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m2"})) == 2,
                msg="2 pods",
            )
            # scale via the subresource (the autoscaler's write path)
            store.scale_model("m2", 1)
            await wait_for(
                lambda: len(kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m2"})) == 1,
                msg="scale down to 1",
            )
            sc = kc.get(kc.path(*MODELS, name="m2", subresource="scale"))
            assert sc["spec"]["replicas"] == 1
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_out_of_band_create_via_watch(api):
    """A CR created by `kubectl` (raw API client) reaches the controller
    through the watch — no in-process apply needed."""

    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        ctrl = ModelController(store, KubeRuntime(store))
        ctrl.start()
        try:
            kc.create(kc.path(*MODELS), {
                "apiVersion": "kubeai.org/v1", "kind": "Model",
                "metadata": {"name": "kubectl-model"},
                "spec": {"url": "hf://org/x", "minReplicas": 1},
            })
            await wait_for(
                lambda: len(kc.list(kc.path(*PODS),
                                    {POD_MODEL_LABEL: "kubectl-model"})) == 1,
                msg="pod for kubectl-created model",
            )
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_model_delete_removes_pods_and_cr(api):
    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        ctrl = ModelController(store, KubeRuntime(store))
        ctrl.start()
        try:
            store.apply_model(
                Model(name="m3", spec=ModelSpec(url="hf://org/m3",
                                                min_replicas=1))
            )
            await wait_for(
                lambda: len(kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m3"})) == 1,
                msg="pod",
            )
            store.delete_model("m3")
            await wait_for(
                lambda: not kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m3"}),
                msg="pods gone",
            )
            await wait_for(
                lambda: not kc.list(kc.path(*MODELS)),
                msg="CR gone",
            )
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_cache_finalizer_deletion_flow(api, tmp_path):
    """cacheProfile models: eviction finalizer added; DELETE leaves the CR
    with deletionTimestamp until the controller evicts and strips it
    (reference cache.go:136-217)."""

    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        from kubeai_amd.controlplane.controller import CacheManager

        ctrl = ModelController(
            store, KubeRuntime(store),
            cache=CacheManager(base_dir=str(tmp_path / "cache")),
        )
        ctrl.start()
        try:
            store.apply_model(
                Model(name="m4", spec=ModelSpec(
                    url="hf://org/m4", min_replicas=0,
                    cache_profile="shared-fs",
                ))
            )
            await wait_for(
                lambda: "kubeai.org/cache-eviction"
                in (kc.get(kc.path(*MODELS, name="m4"))["metadata"].get(
                    "finalizers") or []),
                msg="finalizer added",
            )
            store.delete_model("m4")
            await wait_for(
                lambda: not store.models.get("m4")
                or store.models["m4"].deleted is True,
                msg="deletion seen",
            )
            # finalizer cleared -> CR actually removed
            await wait_for(
                lambda: not kc.list(kc.path(*MODELS)),
                msg="CR fully removed after finalizer",
            )
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_lease_election(api):
    async def main():
        kc = kc_for(api)
        e1 = LeaseElection(kc, identity="a", lease_duration=0.6)
        e2 = LeaseElection(kc_for(api), identity="b", lease_duration=0.6)
        assert e1.tick() is True          # a acquires
        assert e2.tick() is False         # b blocked by live lease
        assert e1.tick() is True          # a renews
        await asyncio.sleep(0.8)          # a's lease expires (no renew)
        assert e2.tick() is True          # b takes over
        assert e1.tick() is False         # a sees b's live lease
        await e2.stop()                   # b releases the holder
        assert e1.tick() is True          # a reacquires after release

    asyncio.run(main())


def test_pod_failed_phase_recreated(api):
    """A pod that dies (phase=Failed) is recreated by the reconciler —
    the reference's pod-recovery behavior (model_pod_recovery_test.go)."""

    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        ctrl = ModelController(store, KubeRuntime(store))
        ctrl.start()
        try:
            store.apply_model(
                Model(name="m5", spec=ModelSpec(url="hf://org/m5",
                                                min_replicas=1))
            )
            pods = await wait_for(
                lambda: kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m5"}) or None,
                msg="pod",
            )
            first = pods[0]["metadata"]["name"]
            kc.patch_merge(
                kc.path(*PODS, name=first, subresource="status"),
                {"status": {"phase": "Failed"}},
            )
            await wait_for(
                lambda: (lambda ps: len(ps) == 1
                         and ps[0]["metadata"]["name"] != first)(
                    kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "m5"})),
                msg="failed pod replaced",
            )
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_full_manager_kube_mode_inference(api):
    """The whole Manager on the K8s substrate: Model CR -> Pod -> mark
    ready (with the model-pod-ip override pointing at a fake engine) ->
    chat completion through the gateway proxies to it. The reference's
    envtest inference-flow trick (utils_test.go:150-159, main_test.go:258).
    """
    from httpx import ASGITransport, AsyncClient

    from tests.test_controlplane import FakeBackend
    from kubeai_amd.controlplane.config import SystemConfig
    from kubeai_amd.controlplane.manager import Manager

    async def main():
        backend = FakeBackend()
        await backend.start()
        cfg = SystemConfig(
            kubernetes={"apiUrl": api.url, "namespace": "kubeai"}
        )
        mgr = Manager(cfg)
        await mgr.start()
        kc = mgr.store.kc
        try:
            ip, port = backend.address.split(":")
            mgr.store.apply_model(
                Model(
                    name="chat-model",
                    spec=ModelSpec(url="hf://org/chat", min_replicas=1),
                    annotations={"model-pod-ip": ip, "model-pod-port": port},
                )
            )
            pods = await wait_for(
                lambda: kc.list(kc.path(*PODS), {POD_MODEL_LABEL: "chat-model"})
                or None,
                msg="pod created",
            )
            mark_pod_ready(kc, pods[0]["metadata"]["name"])
            await wait_for(
                lambda: any(
                    r.ready for r in mgr.store.list_replicas(model="chat-model")
                ),
                msg="replica ready",
            )
            async with AsyncClient(
                transport=ASGITransport(app=mgr.app), base_url="http://kubeai"
            ) as client:
                r = await client.post(
                    "/openai/v1/completions",
                    json={"model": "chat-model", "prompt": "hi", "max_tokens": 4},
                )
                assert r.status_code == 200, r.text
                assert r.json()["choices"][0]["text"] == "ok"
            assert backend.requests, "request never reached the engine"
        finally:
            await mgr.stop()
            await backend.stop()

    asyncio.run(main())


def test_source_pod_additions():
    """Per-scheme credential/volume wiring on engine pods (reference
    model_source.go:82-227)."""
    from kubeai_amd.controlplane.kubeclient import (pod_manifest_for,
                                                    source_pod_additions)

    env, vols, mounts = source_pod_additions("s3://bucket/model")
    names = {e["name"] for e in env}
    assert names == {"AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY"}
    assert env[0]["valueFrom"]["secretKeyRef"]["name"] == "aws"

    env, vols, mounts = source_pod_additions("gs://bucket/model")
    assert env[0]["name"] == "GOOGLE_APPLICATION_CREDENTIALS"
    assert vols and mounts

    env, vols, mounts = source_pod_additions("oss://bucket/model")
    assert {e["name"] for e in env} == {"OSS_ACCESS_KEY_ID",
                                        "OSS_ACCESS_KEY_SECRET"}

    env, vols, mounts = source_pod_additions("hf://org/model")
    assert env[0]["name"] == "HF_TOKEN"

    env, vols, mounts = source_pod_additions("pvc://my-claim/sub/dir")
    assert vols[0]["persistentVolumeClaim"]["claimName"] == "my-claim"
    assert mounts[0]["mountPath"] == "/model"
    assert mounts[0]["subPath"] == "sub/dir"

    m = Model(name="s", spec=ModelSpec(url="pvc://claim/p"))
    pod = pod_manifest_for(m, "p1", "h", 1, "img", "ns")
    assert pod["spec"]["volumes"][0]["persistentVolumeClaim"][
        "claimName"] == "claim"
    assert pod["spec"]["containers"][0]["volumeMounts"][0][
        "mountPath"] == "/model"


def test_engine_pod_builders():
    """Per-engine container contracts (reference engine_vllm.go /
    engine_ollama.go / engine_fasterwhisper.go / engine_infinity.go)."""
    from kubeai_amd.controlplane.kubeclient import pod_manifest_for

    def pod(engine, url="hf://org/m", args=()):
        m = Model(name="m", spec=ModelSpec(url=url, engine=engine,
                                           args=list(args)))
        return pod_manifest_for(m, "p", "h", 1, "img", "ns")

    c = pod("VLLM")["spec"]["containers"][0]
    assert c["command"][-1] == "vllm.entrypoints.openai.api_server"
    assert "--served-model-name" in c["args"]

    c = pod("OLlama", url="ollama://qwen2:0.5b")["spec"]["containers"][0]
    env = {e["name"]: e.get("value") for e in c["env"]}
    assert env["OLLAMA_HOST"] == "0.0.0.0:8000"
    assert env["OLLAMA_KEEP_ALIVE"] == "999999h"
    probe_cmd = " ".join(c["startupProbe"]["exec"]["command"])
    assert "ollama pull qwen2:0.5b" in probe_cmd
    assert "ollama cp qwen2:0.5b m" in probe_cmd

    c = pod("FasterWhisper")["spec"]["containers"][0]
    env = {e["name"]: e.get("value") for e in c["env"]}
    assert env["WHISPER__MODEL"] == "org/m"
    assert env["ENABLE_UI"] == "false"

    c = pod("Infinity")["spec"]["containers"][0]
    env = {e["name"]: e.get("value") for e in c["env"]}
    assert env["INFINITY_MODEL_ID"] == "org/m"
    assert env["INFINITY_SERVED_MODEL_NAME"] == "m"

    # KubeAIEngine + pvc: model arg is the mount path
    c = pod("KubeAIEngine", url="pvc://claim/path")["spec"]["containers"][0]
    assert c["args"][c["args"].index("--model") + 1] == "/model"


def test_model_server_pods_and_json_patches():
    """modelServerPods admin settings + RFC-6902 jsonPatches applied to
    every engine pod (reference system.go:243-260, pod_plan.go:42-44)."""
    from kubeai_amd.controlplane.kubeclient import pod_manifest_for

    m = Model(name="m", spec=ModelSpec(url="hf://org/m"))
    pc = {
        "serviceAccountName": "kubeai-engine",
        "podSecurityContext": {"runAsUser": 1000},
        "securityContext": {"allowPrivilegeEscalation": False},
        "imagePullSecrets": [{"name": "regcred"}],
        "jsonPatches": [
            {"op": "add", "path": "/metadata/labels/team", "value": "ml"},
            {"op": "add", "path": "/spec/containers/0/env/-",
             "value": {"name": "EXTRA", "value": "1"}},
            {"op": "replace", "path": "/spec/restartPolicy",
             "value": "Always"},
        ],
    }
    pod = pod_manifest_for(m, "p", "h", 1, "img", "ns", pod_config=pc)
    assert pod["spec"]["serviceAccountName"] == "kubeai-engine"
    assert pod["spec"]["securityContext"]["runAsUser"] == 1000
    assert pod["spec"]["containers"][0]["securityContext"][
        "allowPrivilegeEscalation"] is False
    assert pod["spec"]["imagePullSecrets"] == [{"name": "regcred"}]
    assert pod["metadata"]["labels"]["team"] == "ml"
    assert pod["spec"]["containers"][0]["env"][-1]["name"] == "EXTRA"
    assert pod["spec"]["restartPolicy"] == "Always"


def test_jsonpatch_ops():
    from kubeai_amd.controlplane.jsonpatch import PatchError, apply_patch

    doc = {"a": {"b": [1, 2, 3]}, "x": 5}
    out = apply_patch(doc, [
        {"op": "test", "path": "/x", "value": 5},
        {"op": "add", "path": "/a/b/1", "value": 99},
        {"op": "remove", "path": "/x"},
        {"op": "copy", "from": "/a/b/0", "path": "/c"},
        {"op": "move", "from": "/a/b/3", "path": "/moved"},
        {"op": "replace", "path": "/c", "value": "z"},
    ])
    # add at /a/b/1 -> [1,99,2,3]; move /a/b/3 (value 3) out -> [1,99,2]
    assert out == {"a": {"b": [1, 99, 2]}, "c": "z", "moved": 3}
    assert doc["x"] == 5  # original untouched
    import pytest as _p

    with _p.raises(PatchError):
        apply_patch(doc, [{"op": "test", "path": "/x", "value": 6}])
    with _p.raises(PatchError):
        apply_patch(doc, [{"op": "replace", "path": "/nope", "value": 1}])


def test_kube_cache_pvc_job_flow(api):
    """K8s-mode cache machinery: profile PVC + loader Job + PVC annotation
    + eviction Job driven by the finalizer (reference cache.go end to
    end)."""

    async def main():
        kc = kc_for(api)
        store = KubeStore(kc)
        await store.start()
        from kubeai_amd.controlplane.kubeclient import (
            KubeCacheManager, PVC_MODEL_ANN_PREFIX, job_completed,
        )

        cache = KubeCacheManager(
            kc,
            cache_profiles={"shared-fs": {
                "sharedFilesystem": {"storageClassName": "nfs-fast"},
            }},
            loader_image="loader:test",
        )
        ctrl = ModelController(store, KubeRuntime(store), cache=cache)
        ctrl.start()
        ns = kc.namespace
        pvc_path = f"/api/v1/namespaces/{ns}/persistentvolumeclaims/shared-model-cache-shared-fs"
        load_path = f"/apis/batch/v1/namespaces/{ns}/jobs/load-cache-m6"
        evict_path = f"/apis/batch/v1/namespaces/{ns}/jobs/evict-cache-m6"
        try:
            store.apply_model(
                Model(name="m6", spec=ModelSpec(
                    url="hf://org/m6", min_replicas=1,
                    cache_profile="shared-fs",
                ))
            )
            # PVC + loader Job appear; no pods until the cache loads
            await wait_for(lambda: kc.get_opt(pvc_path) is not None,
                           msg="cache PVC created")
            await wait_for(lambda: kc.get_opt(load_path) is not None,
                           msg="loader job created")
            pvc = kc.get(pvc_path)
            assert pvc["spec"]["storageClassName"] == "nfs-fast"
            assert pvc["spec"]["accessModes"] == ["ReadWriteMany"]
            job = kc.get(load_path)
            args = job["spec"]["template"]["spec"]["containers"][0]["args"]
            assert args[0] == "hf://org/m6" and args[1].startswith("/models/m6-")
            assert not kc.list(f"/api/v1/namespaces/{ns}/pods")
            assert not job_completed(job)

            # complete the loader Job (a kubelet would); nudge a reconcile
            kc.patch_merge(load_path, {"status": {"succeeded": 1}})
            kc.patch_merge(kc.path(*MODELS, name="m6"),
                           {"metadata": {"annotations": {"poke": "1"}}})
            model_uid = store.models["m6"].uid
            await wait_for(
                lambda: PVC_MODEL_ANN_PREFIX + "m6"
                in ((kc.get(pvc_path)["metadata"].get("annotations")) or {}),
                msg="PVC model annotation recorded",
            )
            await wait_for(lambda: kc.get_opt(load_path) is None,
                           msg="finished loader job cleaned up")
            await wait_for(
                lambda: len(kc.list(f"/api/v1/namespaces/{ns}/pods")) == 1,
                msg="server pod scheduled after cache load",
            )
            pod = kc.list(f"/api/v1/namespaces/{ns}/pods")[0]
            vols = {v["name"] for v in pod["spec"]["volumes"]}
            assert "model-cache" in vols
            cargs = pod["spec"]["containers"][0]["args"]
            assert cargs[cargs.index("--model") + 1] == f"/models/m6-{model_uid}"

            # deletion: eviction Job runs to completion before the
            # finalizer clears
            store.delete_model("m6")
            await wait_for(lambda: kc.get_opt(evict_path) is not None,
                           msg="eviction job created")
            assert kc.get_opt(kc.path(*MODELS, name="m6")) is not None, \
                "finalizer must hold the CR while eviction runs"
            kc.patch_merge(evict_path, {"status": {"conditions": [
                {"type": "Complete", "status": "True"}]}})
            kc.patch_merge(kc.path(*MODELS, name="m6"),
                           {"metadata": {"annotations": {"poke": "2"}}})
            await wait_for(lambda: not kc.list(kc.path(*MODELS)),
                           msg="CR removed after eviction")
            anns = (kc.get(pvc_path)["metadata"].get("annotations")) or {}
            assert PVC_MODEL_ANN_PREFIX + "m6" not in anns
        finally:
            await ctrl.stop()
            store.stop()

    asyncio.run(main())


def test_kube_cache_unknown_profile_errors(api):
    async def main():
        kc = kc_for(api)
        from kubeai_amd.controlplane.kubeclient import KubeCacheManager

        cache = KubeCacheManager(kc, cache_profiles={})

        class M:
            name = "x"
            uid = "u1"

            class spec:
                cache_profile = "nope"

        try:
            await cache.ensure(M())
            raise AssertionError("expected ValueError")
        except ValueError as e:
            assert "cacheProfile" in str(e)

    asyncio.run(main())


def test_configmap_autoscaler_state_store(api):
    from kubeai_amd.controlplane.kubestore import ConfigMapStateStore

    kc = kc_for(api)
    st = ConfigMapStateStore(kc, name="as-state-test")
    assert st.load() == {}
    st.save({"m1": [1.0, 2.0, 3.0]})
    assert st.load() == {"m1": [1.0, 2.0, 3.0]}
    st.save({"m1": [4.0]})  # update path (CM exists)
    assert st.load() == {"m1": [4.0]}
