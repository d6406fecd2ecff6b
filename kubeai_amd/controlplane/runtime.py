"""Replica runtimes — the kubelet analog.

LocalProcessRuntime launches one engine server process per replica
(one GPU per single-GPU replica via HIP_VISIBLE_DEVICES; TP replicas get
several), polls /health for readiness (the reference's startup/readiness
probe contract, engine_vllm.go:101-138), and reports state into the Store.

FakeRuntime mirrors the reference's envtest setup (pods never actually run;
tests flip readiness manually — SURVEY.md §4 "Manual status control").
"""
from __future__ import annotations

import asyncio
import os
import socket
import subprocess
import sys
from typing import Optional

import httpx

from .crd import Model
from .store import Replica, ReplicaState, Store


class GPUAllocator:
    def __init__(self, n_gpus: int):
        self.free: set[int] = set(range(n_gpus))

    def acquire(self, n: int) -> Optional[list[int]]:
        if n == 0:
            return []
        if len(self.free) < n:
            return None
        got = sorted(self.free)[:n]
        for g in got:
            self.free.discard(g)
        return got

    def release(self, ids: list[int]) -> None:
        self.free.update(ids)


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class FakeRuntime:
    """Replicas are records only; tests control readiness."""

    def __init__(self, store: Store, n_gpus: int = 8):
        self.store = store
        self.gpus = GPUAllocator(n_gpus)

    async def create(self, model: Model, name: str, spec_hash: str, n_gpus: int) -> None:
        ids = self.gpus.acquire(n_gpus)
        rep = Replica(name=name, model=model.name, hash=spec_hash)
        if ids is None:
            rep.scheduled = False
            rep.state = ReplicaState.PENDING
        else:
            rep.gpu_ids = ids
            rep.state = ReplicaState.STARTING
        self.store.add_replica(rep)

    async def delete(self, name: str) -> None:
        rep = self.store.get_replica(name)
        if rep:
            self.gpus.release(rep.gpu_ids)
            self.store.remove_replica(name)

    # test helpers (envtest-style manual status control)
    def mark_ready(self, name: str, address: str = "127.0.0.1:0") -> None:
        self.store.update_replica(
            name, state=ReplicaState.READY, address=address
        )


class LocalProcessRuntime:
    """One engine-server subprocess per replica on this node."""

    def __init__(
        self,
        store: Store,
        n_gpus: Optional[int] = None,
        engine_args_default: Optional[list[str]] = None,
        health_interval: float = 0.5,
    ):
        if n_gpus is None:
            try:
                import torch

                n_gpus = torch.cuda.device_count()
            except Exception:
                n_gpus = 0
        self.store = store
        self.gpus = GPUAllocator(n_gpus)
        self.procs: dict[str, subprocess.Popen] = {}
        self.health_interval = health_interval
        self._monitors: dict[str, asyncio.Task] = {}

    SUPPORTED_ENGINES = ("KubeAIEngine",)

    async def create(self, model: Model, name: str, spec_hash: str, n_gpus: int) -> None:
        rep = Replica(name=name, model=model.name, hash=spec_hash)
        if model.spec.engine not in self.SUPPORTED_ENGINES:
            # third-party engine images (vLLM/Ollama/FasterWhisper/Infinity)
            # are cluster-deployment concerns (chart modelServers); this
            # node runtime runs the in-house engine only
            rep.scheduled = False
            self.store.add_replica(rep)
            return
        ids = self.gpus.acquire(n_gpus)
        if ids is None:
            rep.scheduled = False  # unschedulable: waits for free GPUs
            self.store.add_replica(rep)
            return
        port = _free_port()
        rep.gpu_ids = ids
        rep.state = ReplicaState.STARTING
        env = dict(os.environ)
        env.update(model.spec.env)  # reference: model_types.go:86-90
        if ids:
            env["HIP_VISIBLE_DEVICES"] = ",".join(map(str, ids))
            env["CUDA_VISIBLE_DEVICES"] = env["HIP_VISIBLE_DEVICES"]
        # spec.files -> per-replica dir (reference: files.go ConfigMap mounts)
        if model.spec.files:
            files_dir = os.path.join("/tmp/kubeai-replica-files", name)
            os.makedirs(files_dir, exist_ok=True)
            for f in model.spec.files:
                dst = os.path.join(files_dir, f.path.lstrip("/").replace("/", "_"))
                with open(dst, "w") as fh:
                    fh.write(f.content)
            env["KUBEAI_MODEL_FILES_DIR"] = files_dir
        model_ref = _model_source_path(model)
        cmd = [
            sys.executable,
            "-m",
            "kubeai_amd.engine.server",
            "--model",
            model_ref,
            "--served-model-name",
            model.name,
            "--host",
            "127.0.0.1",
            "--port",
            str(port),
        ] + list(model.spec.args)
        if "--task" not in model.spec.args:
            feats = set(model.spec.features)
            if "SpeechToText" in feats:
                cmd += ["--task", "transcribe"]
            elif feats and feats <= {"TextEmbedding", "Reranking"}:
                # pure encoder model (no TextGeneration): BERT-arch engine
                cmd += ["--task", "embed"]
        if not ids:
            cmd += ["--device", "cpu"]
        proc = subprocess.Popen(cmd, env=env)
        self.procs[name] = proc
        rep.address = f"127.0.0.1:{port}"
        self.store.add_replica(rep)
        self._monitors[name] = asyncio.create_task(self._monitor(name, port, proc))

    async def _monitor(self, name: str, port: int, proc: subprocess.Popen) -> None:
        url = f"http://127.0.0.1:{port}/health"
        async with httpx.AsyncClient(timeout=2.0) as client:
            while True:
                rep = self.store.get_replica(name)
                if rep is None:
                    return
                if proc.poll() is not None:
                    self.store.update_replica(name, state=ReplicaState.FAILED)
                    return
                try:
                    r = await client.get(url)
                    ok = r.status_code == 200
                except Exception:
                    ok = False
                if ok and rep.state != ReplicaState.READY:
                    self.store.update_replica(name, state=ReplicaState.READY)
                elif not ok and rep.state == ReplicaState.READY:
                    self.store.update_replica(name, state=ReplicaState.STARTING)
                await asyncio.sleep(self.health_interval)

    async def delete(self, name: str) -> None:
        rep = self.store.get_replica(name)
        mon = self._monitors.pop(name, None)
        if mon:
            mon.cancel()
        proc = self.procs.pop(name, None)
        if proc is not None and proc.poll() is None:
            proc.terminate()
            try:
                await asyncio.get_running_loop().run_in_executor(
                    None, proc.wait, 10
                )
            except Exception:
                proc.kill()
        if rep:
            self.gpus.release(rep.gpu_ids)
            self.store.remove_replica(name)

    async def shutdown(self) -> None:
        for name in list(self.procs):
            await self.delete(name)


def _model_source_path(model: Model) -> str:
    """Map Model.spec.url to what the engine server loads.

    Schemes (reference: model_source.go:229-287): pvc:// and file:// map to
    local paths (the cache dir layout); hf://org/name has no network here,
    so unresolved hf URLs fall back to preset names when they match.
    """
    url = model.spec.url
    if url.startswith("file://"):
        return url[len("file://") :]
    if url.startswith("pvc://"):
        rest = url[len("pvc://") :]
        return "/model/" + rest.split("/", 1)[1] if "/" in rest else "/model"
    if url.startswith("hf://"):
        name = url[len("hf://") :]
        from kubeai_amd.models.config import PRESETS

        short = name.split("/")[-1].lower()
        for preset in PRESETS:
            if preset in short:
                return preset
        return name
    return url
