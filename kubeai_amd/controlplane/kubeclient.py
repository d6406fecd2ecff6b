"""Minimal Kubernetes API client + Model/Pod manifest conversion.

No `kubernetes` package exists in this image, so this speaks the REST API
directly (the reference uses controller-runtime/client-go for the same
verbs: typed GET/LIST/WATCH/PATCH — internal/manager/run.go:142-174,
internal/k8sutils/apply.go:10-17). Scope is exactly what the control
plane needs:

  - CRUD + merge-patch on namespaced resources (Models, Pods, Leases,
    ConfigMaps) and the Model `scale` subresource
  - LIST with equality label selectors
  - WATCH as a line-delimited JSON stream (informer feed)

Auth: in-cluster service-account token/CA when present, else plain
`api_url` (tests run against the in-process fake API server,
kubeai_amd/controlplane/fakekube.py — the envtest analog).
"""
from __future__ import annotations

import json
import time
import os
from typing import Iterator, Optional

import requests

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class ApiError(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"kube api {status}: {body[:300]}")
        self.status = status
        self.body = body


class KubeClient:
    def __init__(
        self,
        api_url: Optional[str] = None,
        namespace: Optional[str] = None,
        token: Optional[str] = None,
        verify=None,
        timeout: float = 10.0,
    ):
        if api_url is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            if not host:
                raise ValueError("no api_url and not running in-cluster")
            api_url = f"https://{host}:{port}"
            if token is None and os.path.exists(os.path.join(SA_DIR, "token")):
                with open(os.path.join(SA_DIR, "token")) as f:
                    token = f.read().strip()
            if verify is None and os.path.exists(os.path.join(SA_DIR, "ca.crt")):
                verify = os.path.join(SA_DIR, "ca.crt")
        if namespace is None:
            ns_file = os.path.join(SA_DIR, "namespace")
            if os.path.exists(ns_file):
                with open(ns_file) as f:
                    namespace = f.read().strip()
            else:
                namespace = "default"
        self.base = api_url.rstrip("/")
        self.namespace = namespace
        self.timeout = timeout
        self._s = requests.Session()
        if token:
            self._s.headers["Authorization"] = f"Bearer {token}"
        if verify is not None:
            self._s.verify = verify

    # ------------------------------------------------------------- paths
    def path(self, group: str, version: str, plural: str,
             name: Optional[str] = None, subresource: Optional[str] = None,
             namespace: Optional[str] = None) -> str:
        ns = namespace or self.namespace
        root = "/api" if group == "" else f"/apis/{group}"
        p = f"{root}/{version}/namespaces/{ns}/{plural}"
        if name:
            p += f"/{name}"
        if subresource:
            p += f"/{subresource}"
        return p

    # ------------------------------------------------------------- verbs
    def _req(self, method: str, path: str, *, params=None, body=None,
             content_type: str = "application/json"):
        r = self._s.request(
            method,
            self.base + path,
            params=params,
            data=json.dumps(body) if body is not None else None,
            headers={"Content-Type": content_type},
            timeout=self.timeout,
        )
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text)
        return r.json() if r.text else None

    def get(self, path: str) -> dict:
        return self._req("GET", path)

    def list(self, path: str, label_selector: Optional[dict] = None) -> list[dict]:
        params = {}
        if label_selector:
            params["labelSelector"] = ",".join(
                f"{k}={v}" for k, v in sorted(label_selector.items())
            )
        out = self._req("GET", path, params=params)
        return out.get("items", [])

    def create(self, path: str, manifest: dict) -> dict:
        return self._req("POST", path, body=manifest)

    def replace(self, path: str, manifest: dict) -> dict:
        return self._req("PUT", path, body=manifest)

    def patch_merge(self, path: str, patch: dict) -> dict:
        return self._req(
            "PATCH", path, body=patch,
            content_type="application/merge-patch+json",
        )

    def delete(self, path: str) -> Optional[dict]:
        return self._req("DELETE", path)

    def get_opt(self, path: str) -> Optional[dict]:
        """GET that maps 404 to None."""
        try:
            return self._req("GET", path)
        except ApiError as e:
            if e.status == 404:
                return None
            raise

    def delete_opt(self, path: str) -> None:
        """DELETE that swallows 404."""
        try:
            self._req("DELETE", path)
        except ApiError as e:
            if e.status != 404:
                raise

    def watch(self, path: str, resource_version: str = "0",
              timeout: Optional[float] = None) -> Iterator[dict]:
        """Stream watch events {type: ADDED|MODIFIED|DELETED, object: …}.
        Raises on disconnect — callers re-list + re-watch (informer loop)."""
        r = self._s.get(
            self.base + path,
            params={"watch": "true", "resourceVersion": resource_version},
            stream=True,
            timeout=timeout or 3600,
        )
        if r.status_code >= 400:
            raise ApiError(r.status_code, r.text)
        for line in r.iter_lines():
            if line:
                yield json.loads(line)


# ======================================================================
# Model CR <-> dataclass (field names match deploy/crds/kubeai.org_models
# .yaml, which mirrors the reference api/k8s/v1/model_types.go)

def model_to_manifest(m) -> dict:
    from .crd import Model  # noqa: F401

    s = m.spec
    spec: dict = {
        "url": s.url,
        "features": list(s.features),
        "engine": s.engine,
    }
    if s.adapters:
        spec["adapters"] = [{"name": a.name, "url": a.url} for a in s.adapters]
    if s.resource_profile:
        spec["resourceProfile"] = s.resource_profile
    if s.cache_profile:
        spec["cacheProfile"] = s.cache_profile
    if s.image:
        spec["image"] = s.image
    if s.args:
        spec["args"] = list(s.args)
    if s.env:
        spec["env"] = dict(s.env)
    if s.replicas is not None:
        spec["replicas"] = s.replicas
    spec["minReplicas"] = s.min_replicas
    if s.max_replicas is not None:
        spec["maxReplicas"] = s.max_replicas
    if s.autoscaling_disabled:
        spec["autoscalingDisabled"] = True
    spec["targetRequests"] = s.target_requests
    spec["scaleDownDelaySeconds"] = s.scale_down_delay_seconds
    lb = s.load_balancing
    spec["loadBalancing"] = {
        "strategy": lb.strategy,
        "prefixHash": {
            "meanLoadFactor": lb.prefix_hash.mean_load_percentage,
            "replication": lb.prefix_hash.replication,
            "prefixCharLength": lb.prefix_hash.prefix_char_length,
        },
    }
    if s.files:
        spec["files"] = [{"path": f.path, "content": f.content} for f in s.files]
    if s.priority_class_name:
        spec["priorityClassName"] = s.priority_class_name
    if s.owner:
        spec["owner"] = s.owner
    meta: dict = {"name": m.name}
    if m.labels:
        meta["labels"] = dict(m.labels)
    if m.annotations:
        meta["annotations"] = dict(m.annotations)
    if m.finalizers:
        meta["finalizers"] = list(m.finalizers)
    if m.uid:
        meta["uid"] = m.uid
    return {
        "apiVersion": "kubeai.org/v1",
        "kind": "Model",
        "metadata": meta,
        "spec": spec,
        "status": {
            "replicas": {
                "all": m.status.replicas_all,
                "ready": m.status.replicas_ready,
            },
            "cache": {"loaded": m.status.cache_loaded},
        },
    }


def model_from_manifest(obj: dict):
    from .crd import (AdapterSpec, FileSpec, LoadBalancingSpec, Model,
                      ModelSpec, ModelStatus, PrefixHashSpec)

    meta = obj.get("metadata", {})
    s = obj.get("spec", {})
    lb = s.get("loadBalancing", {}) or {}
    ph = lb.get("prefixHash", {}) or {}
    spec = ModelSpec(
        url=s.get("url", ""),
        features=list(s.get("features") or ["TextGeneration"]),
        engine=s.get("engine", "KubeAIEngine"),
        adapters=[
            AdapterSpec(name=a.get("name", ""), url=a.get("url", ""))
            for a in (s.get("adapters") or [])
        ],
        resource_profile=s.get("resourceProfile", ""),
        cache_profile=s.get("cacheProfile", ""),
        image=s.get("image", ""),
        args=list(s.get("args") or []),
        env=dict(s.get("env") or {}),
        replicas=s.get("replicas"),
        min_replicas=int(s.get("minReplicas") or 0),
        max_replicas=s.get("maxReplicas"),
        autoscaling_disabled=bool(s.get("autoscalingDisabled") or False),
        target_requests=int(s.get("targetRequests") or 100),
        scale_down_delay_seconds=int(s.get("scaleDownDelaySeconds") or 30),
        load_balancing=LoadBalancingSpec(
            strategy=lb.get("strategy", "LeastLoad"),
            prefix_hash=PrefixHashSpec(
                mean_load_percentage=int(ph.get("meanLoadFactor") or 125),
                replication=int(ph.get("replication") or 256),
                prefix_char_length=int(ph.get("prefixCharLength") or 100),
            ),
        ),
        files=[
            FileSpec(path=f.get("path", ""), content=f.get("content", ""))
            for f in (s.get("files") or [])
        ],
        priority_class_name=s.get("priorityClassName", ""),
        owner=s.get("owner", ""),
    )
    st = obj.get("status", {}) or {}
    reps = st.get("replicas", {}) or {}
    status = ModelStatus(
        replicas_all=int(reps.get("all") or 0),
        replicas_ready=int(reps.get("ready") or 0),
        cache_loaded=bool((st.get("cache") or {}).get("loaded") or False),
    )
    m = Model(
        name=meta.get("name", ""),
        spec=spec,
        status=status,
        labels=dict(meta.get("labels") or {}),
        annotations=dict(meta.get("annotations") or {}),
        finalizers=list(meta.get("finalizers") or []),
        generation=int(meta.get("generation") or 0),
        deleted=bool(meta.get("deletionTimestamp")),
        uid=meta.get("uid", ""),
    )
    return m


# ======================================================================
# Engine pod manifest — the reference contract (engine_vllm.go:12-167):
# container port 8000 + port annotation, /health startup/readiness/
# liveness probes, `model`/`pod-hash` labels, resource-profile GPU counts.

ENGINE_PORT = 8000

# per-scheme credential secret names (reference config SecretNames,
# internal/config/system.go:148-153)
SECRET_NAMES = {"aws": "aws", "gcp": "gcp", "alibaba": "alibaba",
                "huggingface": "huggingface"}


def _secret_env(name: str, secret: str, key: str) -> dict:
    return {
        "name": name,
        "valueFrom": {"secretKeyRef": {"name": secret, "key": key,
                                       "optional": True}},
    }


def source_pod_additions(url: str) -> tuple[list, list, list]:
    """(env, volumes, volume_mounts) for a Model source URL — the
    credential/volume wiring the reference applies per scheme
    (internal/modelcontroller/model_source.go:82-227)."""
    env: list = []
    vols: list = []
    mounts: list = []
    if url.startswith("hf://"):
        env.append(_secret_env("HF_TOKEN", SECRET_NAMES["huggingface"],
                               "token"))
    elif url.startswith("s3://"):
        env.append(_secret_env("AWS_ACCESS_KEY_ID", SECRET_NAMES["aws"],
                               "accessKeyID"))
        env.append(_secret_env("AWS_SECRET_ACCESS_KEY", SECRET_NAMES["aws"],
                               "secretAccessKey"))
    elif url.startswith("gs://"):
        env.append({"name": "GOOGLE_APPLICATION_CREDENTIALS",
                    "value": "/secrets/gcp/credentials.json"})
        vols.append({"name": "gcp-credentials",
                     "secret": {"secretName": SECRET_NAMES["gcp"],
                                "optional": True}})
        mounts.append({"name": "gcp-credentials",
                       "mountPath": "/secrets/gcp", "readOnly": True})
    elif url.startswith("oss://"):
        env.append(_secret_env("OSS_ACCESS_KEY_ID", SECRET_NAMES["alibaba"],
                               "accessKeyID"))
        env.append(_secret_env("OSS_ACCESS_KEY_SECRET",
                               SECRET_NAMES["alibaba"], "accessKeySecret"))
    elif url.startswith("pvc://"):
        rest = url[len("pvc://"):]
        claim, _, sub = rest.partition("/")
        vol: dict = {"name": "model-pvc",
                     "persistentVolumeClaim": {"claimName": claim}}
        mount: dict = {"name": "model-pvc", "mountPath": "/model",
                       "readOnly": True}
        if sub:
            mount["subPath"] = sub
        vols.append(vol)
        mounts.append(mount)
    return env, vols, mounts


def _engine_container(model, image: str) -> dict:
    """Per-engine command/env contract (reference engine builders:
    engine_vllm.go:86, engine_ollama.go:22-35, engine_fasterwhisper.go:29,
    engine_infinity.go:31-54). The in-house KubeAIEngine speaks the same
    vLLM-compatible CLI surface."""
    spec = model.spec
    eng = spec.engine
    ref = spec.url.split("://", 1)[-1]
    if eng in ("KubeAIEngine", "VLLM"):
        cmd = (
            ["python", "-m", "kubeai_amd.engine.server"]
            if eng == "KubeAIEngine"
            else ["python3", "-m", "vllm.entrypoints.openai.api_server"]
        )
        model_arg = "/model" if spec.url.startswith("pvc://") else spec.url
        if eng == "VLLM":
            model_arg = "/model" if spec.url.startswith("pvc://") else ref
        if spec.cache_profile:
            # cached models load from the shared PVC (cache.go
            # patchServerCacheVolumes + engine_vllm.go modelCacheDir arg)
            model_arg = f"/models/{model.name}-{model.uid}"
        return {
            "command": cmd,
            "args": ["--model", model_arg,
                     "--served-model-name", model.name,
                     "--port", str(ENGINE_PORT)] + list(spec.args),
            "env": [],
        }
    if eng == "OLlama":
        return {
            "command": [],  # image entrypoint (`ollama serve`)
            "args": list(spec.args),
            "env": [
                {"name": "OLLAMA_HOST", "value": f"0.0.0.0:{ENGINE_PORT}"},
                {"name": "OLLAMA_KEEP_ALIVE", "value": "999999h"},
            ],
            # startup probe pulls + aliases the model (engine_ollama.go:173-213)
            "startupProbeExec": [
                "/bin/sh", "-c",
                f"ollama pull {ref} && ollama cp {ref} {model.name} && "
                f"ollama run {model.name} hi",
            ],
        }
    if eng == "FasterWhisper":
        return {
            "command": [],
            "args": list(spec.args),
            "env": [
                {"name": "WHISPER__MODEL", "value": ref},
                {"name": "WHISPER__PORT", "value": str(ENGINE_PORT)},
                {"name": "ENABLE_UI", "value": "false"},
            ],
        }
    if eng == "Infinity":
        return {
            "command": [],
            "args": list(spec.args),
            "env": [
                {"name": "INFINITY_MODEL_ID", "value": ref},
                {"name": "INFINITY_SERVED_MODEL_NAME", "value": model.name},
                {"name": "INFINITY_PORT", "value": str(ENGINE_PORT)},
            ],
        }
    raise ValueError(f"no pod builder for engine {eng!r}")


def pod_manifest_for(model, name: str, spec_hash: str, n_gpus: int,
                     image: str, namespace: str,
                     gpu_resource: str = "amd.com/gpu",
                     pod_config: Optional[dict] = None) -> dict:
    """pod_config = the reference's `modelServerPods` admin settings
    (config/system.go:243-260): serviceAccountName, securityContext,
    podSecurityContext, imagePullSecrets, jsonPatches (RFC 6902, applied
    last — pod_plan.go:42-44)."""
    from .crd import POD_HASH_LABEL, POD_MODEL_LABEL

    eng = _engine_container(model, image)
    args = eng["args"]
    env = [{"name": k, "value": v} for k, v in sorted(model.spec.env.items())]
    env += eng["env"]
    src_env, src_vols, src_mounts = source_pod_additions(model.spec.url)
    if model.spec.cache_profile:
        # mount the profile cache PVC read-only at the model dir
        cdir = f"/models/{model.name}-{model.uid}"
        src_vols = src_vols + [{
            "name": "model-cache",
            "persistentVolumeClaim": {
                "claimName": f"shared-model-cache-{model.spec.cache_profile}",
                "readOnly": True,
            },
        }]
        src_mounts = src_mounts + [{
            "name": "model-cache", "mountPath": cdir,
            "subPath": cdir.lstrip("/"), "readOnly": True,
        }]
    env += src_env
    resources = {}
    if n_gpus > 0:
        resources = {
            "requests": {gpu_resource: str(n_gpus)},
            "limits": {gpu_resource: str(n_gpus)},
        }
    probe = {
        "httpGet": {"path": "/health", "port": ENGINE_PORT},
        "periodSeconds": 2,
    }
    pc = pod_config or {}
    manifest = {
        "apiVersion": "v1",
        "kind": "Pod",
        "metadata": {
            "name": name,
            "namespace": namespace,
            "labels": {
                POD_MODEL_LABEL: model.name,
                POD_HASH_LABEL: spec_hash,
            },
            "annotations": {"model-pod-port": str(ENGINE_PORT)},
        },
        "spec": {
            "containers": [
                {
                    "name": "server",
                    "image": image,
                    **({"command": eng["command"]} if eng["command"] else {}),
                    "args": args,
                    "env": env,
                    "resources": resources,
                    "ports": [{"containerPort": ENGINE_PORT}],
                    # reference probe contract engine_vllm.go:101-138;
                    # Ollama's startup probe pulls the model instead
                    # (engine_ollama.go:173-213)
                    "startupProbe": (
                        {"exec": {"command": eng["startupProbeExec"]},
                         "periodSeconds": 10, "failureThreshold": 1080}
                        if eng.get("startupProbeExec")
                        else {**probe, "failureThreshold": 5400}
                    ),
                    "readinessProbe": {**probe, "failureThreshold": 3},
                    "livenessProbe": {
                        **probe, "periodSeconds": 10, "failureThreshold": 3,
                    },
                    **({"volumeMounts": src_mounts} if src_mounts else {}),
                    **(
                        {"securityContext": pc["securityContext"]}
                        if pc.get("securityContext")
                        else {}
                    ),
                }
            ],
            **({"volumes": src_vols} if src_vols else {}),
            **(
                {"serviceAccountName": pc["serviceAccountName"]}
                if pc.get("serviceAccountName")
                else {}
            ),
            **(
                {"securityContext": pc["podSecurityContext"]}
                if pc.get("podSecurityContext")
                else {}
            ),
            **(
                {"imagePullSecrets": pc["imagePullSecrets"]}
                if pc.get("imagePullSecrets")
                else {}
            ),
            "restartPolicy": "Never",
        },
    }
    patches = pc.get("jsonPatches") or []
    if patches:
        from .jsonpatch import apply_patch

        manifest = apply_patch(manifest, patches)
    return manifest


# ---------------------------------------------------------------- cache
# Reference: internal/modelcontroller/cache.go — a shared-filesystem PVC
# per cache profile ("shared-model-cache-<profile>", RWX), a loader Job
# ("load-cache-<model>") that downloads into /models/<name>-<uid>, a
# per-model annotation on the PVC ("models.kubeai.org/<model>" ->
# {"uid","timestamp"}) marking loaded content, and an eviction Job
# ("evict-cache-<model>") driven by the cache-eviction finalizer.

PVC_MODEL_ANN_PREFIX = "models.kubeai.org/"


def model_cache_dir(model) -> str:
    return f"/models/{model.name}-{model.uid}"


def cache_pvc_name(model) -> str:
    return f"shared-model-cache-{model.spec.cache_profile}"


def cache_pvc_manifest(model, profile: dict, namespace: str) -> dict:
    fs = (profile or {}).get("sharedFilesystem") or {}
    spec = {
        "accessModes": ["ReadWriteMany"],
        "storageClassName": fs.get("storageClassName", ""),
        "resources": {"requests": {"storage": fs.get("size", "10Gi")}},
    }
    if fs.get("persistentVolumeName"):
        spec["volumeName"] = fs["persistentVolumeName"]
    return {
        "apiVersion": "v1",
        "kind": "PersistentVolumeClaim",
        "metadata": {"name": cache_pvc_name(model), "namespace": namespace},
        "spec": spec,
    }


def _cache_job_manifest(model, name: str, container: dict,
                        namespace: str) -> dict:
    container.setdefault("volumeMounts", []).append(
        {"name": "model", "mountPath": "/models", "subPath": "models"}
    )
    return {
        "apiVersion": "batch/v1",
        "kind": "Job",
        "metadata": {
            "name": name,
            "namespace": namespace,
            "labels": {"app.kubernetes.io/name": "kubeai",
                       "kubeai.org/cache-job-for": model.name},
        },
        "spec": {
            "ttlSecondsAfterFinished": 60,
            "parallelism": 1,
            "completions": 1,
            "template": {
                "spec": {
                    "restartPolicy": "OnFailure",
                    "containers": [container],
                    "volumes": [{
                        "name": "model",
                        "persistentVolumeClaim": {
                            "claimName": cache_pvc_name(model)
                        },
                    }],
                },
            },
        },
    }


def load_cache_job_manifest(model, loader_image: str, namespace: str) -> dict:
    env = [{"name": k, "value": v} for k, v in sorted(model.spec.env.items())]
    src_env, src_vols, src_mounts = source_pod_additions(model.spec.url)
    if model.spec.cache_profile:
        # mount the profile cache PVC read-only at the model dir
        cdir = f"/models/{model.name}-{model.uid}"
        src_vols = src_vols + [{
            "name": "model-cache",
            "persistentVolumeClaim": {
                "claimName": f"shared-model-cache-{model.spec.cache_profile}",
                "readOnly": True,
            },
        }]
        src_mounts = src_mounts + [{
            "name": "model-cache", "mountPath": cdir,
            "subPath": cdir.lstrip("/"), "readOnly": True,
        }]
    cdir = model_cache_dir(model)
    container = {
        "name": "loader",
        "image": loader_image,
        "env": env + src_env,
        "args": [model.spec.url, cdir],
        "volumeMounts": [
            {"name": "model", "mountPath": cdir, "subPath": cdir.lstrip("/")}
        ] + src_mounts,
    }
    m = _cache_job_manifest(model, f"load-cache-{model.name}", container,
                            namespace)
    # source secrets (hf/s3/gs/oss) piggyback on the loader pod
    m["spec"]["template"]["spec"]["volumes"].extend(src_vols)
    return m


def evict_cache_job_manifest(model, loader_image: str, namespace: str) -> dict:
    container = {
        "name": "evictor",
        "image": loader_image,
        "command": ["bash", "-c", f"rm -rf {model_cache_dir(model)}"],
        "volumeMounts": [],
    }
    return _cache_job_manifest(model, f"evict-cache-{model.name}", container,
                               namespace)


def job_completed(job: dict) -> bool:
    st = (job or {}).get("status") or {}
    if int(st.get("succeeded") or 0) >= 1:
        return True
    return any(
        c.get("type") == "Complete" and c.get("status") == "True"
        for c in st.get("conditions") or []
    )


class KubeCacheManager:
    """PVC + Job cache machinery for Kubernetes mode (cache.go analog).

    ensure(): creates the profile PVC on first use, runs the loader Job,
    records loaded content as a PVC annotation, deletes the finished Job.
    evict(): returns True when eviction has fully completed (the
    controller keeps the finalizer until then — same multi-reconcile flow
    as the reference's reconcileCache deletion branch).
    """

    def __init__(self, kc: KubeClient, cache_profiles: dict | None = None,
                 loader_image: str = "substratusai/huggingface-model-loader:main"):
        self.kc = kc
        self.profiles = cache_profiles or {}
        self.loader_image = loader_image

    def _paths(self, model):
        ns = self.kc.namespace
        return (
            f"/api/v1/namespaces/{ns}/persistentvolumeclaims/{cache_pvc_name(model)}",
            f"/apis/batch/v1/namespaces/{ns}/jobs/load-cache-{model.name}",
            f"/apis/batch/v1/namespaces/{ns}/jobs/evict-cache-{model.name}",
        )

    async def ensure(self, model) -> bool:
        prof = self.profiles.get(model.spec.cache_profile)
        if prof is None:
            raise ValueError(
                f"model {model.name}: unknown cacheProfile "
                f"{model.spec.cache_profile!r}"
            )
        pvc_path, load_path, _ = self._paths(model)
        pvc = self.kc.get_opt(pvc_path)
        if pvc is None:
            pvc = self.kc.create(
                pvc_path.rsplit("/", 1)[0],
                cache_pvc_manifest(model, prof, self.kc.namespace),
            )
        ann = ((pvc.get("metadata") or {}).get("annotations") or {}).get(
            PVC_MODEL_ANN_PREFIX + model.name
        )
        if ann:
            try:
                if json.loads(ann).get("uid") == str(model.uid):
                    self.kc.delete_opt(load_path)  # tidy a finished Job
                    return True
            except ValueError:
                pass
        job = self.kc.get_opt(load_path)
        if job is None:
            self.kc.create(
                load_path.rsplit("/", 1)[0],
                load_cache_job_manifest(model, self.loader_image,
                                        self.kc.namespace),
            )
            return False
        if not job_completed(job):
            return False
        self.kc.patch_merge(pvc_path, {
            "metadata": {"annotations": {
                PVC_MODEL_ANN_PREFIX + model.name: json.dumps(
                    {"uid": str(model.uid), "timestamp": int(time.time())}
                ),
            }},
        })
        self.kc.delete_opt(load_path)
        return True

    async def evict(self, model) -> bool:
        pvc_path, load_path, evict_path = self._paths(model)
        pvc = self.kc.get_opt(pvc_path)
        if pvc is None or (pvc.get("metadata") or {}).get("deletionTimestamp"):
            self.kc.delete_opt(load_path)
            self.kc.delete_opt(evict_path)
            return True
        anns = (pvc.get("metadata") or {}).get("annotations") or {}
        if PVC_MODEL_ANN_PREFIX + model.name in anns:
            # merge-patch null deletes the key
            self.kc.patch_merge(pvc_path, {
                "metadata": {"annotations": {
                    PVC_MODEL_ANN_PREFIX + model.name: None,
                }},
            })
        job = self.kc.get_opt(evict_path)
        if job is None:
            self.kc.create(
                evict_path.rsplit("/", 1)[0],
                evict_cache_job_manifest(model, self.loader_image,
                                         self.kc.namespace),
            )
            return False
        if not job_completed(job):
            return False
        self.kc.delete_opt(evict_path)
        self.kc.delete_opt(load_path)
        return True
