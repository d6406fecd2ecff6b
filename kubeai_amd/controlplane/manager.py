"""Composition root (reference: internal/manager/run.go).

Wires store + controller + load balancer + autoscaler + proxy + gateway +
messengers + leader election and serves the OpenAI API. Usable two ways:

  - CLI:   python -m kubeai_amd.controlplane.manager --config config.yaml
  - tests: Manager(cfg, runtime=FakeRuntime(...)) -> await mgr.start()
           (the envtest-style in-process harness, main_test.go:132-157)
"""
from __future__ import annotations

import argparse
import asyncio
from typing import Optional

from .autoscaler import Autoscaler
from .config import SystemConfig, load_config
from .controller import CacheManager, ModelController
from .leader import Election
from .loadbalancer import LoadBalancer
from .messenger import MemBroker, Messenger
from .modelclient import ModelClient
from .openaiserver import build_gateway_app
from .proxy import ProxyHandler
from .runtime import FakeRuntime, LocalProcessRuntime
from .store import Store


class Manager:
    def __init__(
        self,
        cfg: Optional[SystemConfig] = None,
        runtime=None,
        broker: Optional[MemBroker] = None,
    ):
        self.cfg = cfg or SystemConfig()
        kube = self.cfg.kubernetes
        if kube is not None:
            # K8s substrate: Models as CRs, replicas as Pods, Lease
            # election (reference run.go:142-174 topology)
            from .kubeclient import KubeClient
            from .kubestore import KubeRuntime, KubeStore, LeaseElection

            kc = KubeClient(
                api_url=kube.get("apiUrl") or None,
                namespace=kube.get("namespace"),
            )
            self.store = KubeStore(kc)
            self.runtime = runtime or KubeRuntime(
                self.store,
                image=kube.get("engineImage", "kubeai-amd-engine:latest"),
                gpu_resource=kube.get("gpuResource", "amd.com/gpu"),
                engine_images=kube.get("engineImages"),
            )
            self.election = LeaseElection(
                kc, lease_name=kube.get("leaseName", "kubeai.org")
            )
            from .kubeclient import KubeCacheManager
            from .kubestore import ConfigMapStateStore

            kube_state = ConfigMapStateStore(
                kc, name=kube.get("autoscalerStateConfigMap",
                                  "kubeai-autoscaler-state"),
            )
            kube_cache = KubeCacheManager(
                kc,
                cache_profiles=self.cfg.cacheProfiles,
                loader_image=kube.get(
                    "modelLoaderImage",
                    "substratusai/huggingface-model-loader:main",
                ),
            )
        else:
            kube_cache = None
            kube_state = None
            self.store = Store()
            self.runtime = runtime or LocalProcessRuntime(
                self.store, n_gpus=self.cfg.n_gpus
            )
            self.election = Election(self.cfg.leader_lock_path)
        self.model_client = ModelClient(
            self.store,
            required_consecutive_scale_downs=self.cfg.autoscaling.required_consecutive_scale_downs,
            autoscaling_interval=self.cfg.autoscaling.interval_seconds,
        )
        self.lb = LoadBalancer(self.store)
        self.controller = ModelController(
            self.store,
            self.runtime,
            resource_profiles=self.cfg.resource_profiles,
            cache=kube_cache or CacheManager(self.cfg.cache_dir),
        )
        self.autoscaler = Autoscaler(
            self.store,
            self.model_client,
            interval=self.cfg.autoscaling.interval_seconds,
            time_window=self.cfg.autoscaling.time_window_seconds,
            self_metric_addrs=self.cfg.fixed_self_metric_addrs,
            state_path=self.cfg.autoscaling.state_path,
            state_store=kube_state,
            is_leader=self.election.is_leader,
        )
        self.proxy = ProxyHandler(
            self.model_client, self.lb, priority_classes=self.cfg.priority_classes
        )
        self.app = build_gateway_app(self.model_client, self.proxy)
        # broker driver selected per stream by URL scheme (mem:// shares
        # one in-process broker; file:// is durable); an injected broker
        # overrides for tests
        from .messenger import stream_transport

        self.broker = broker or MemBroker()
        self.messengers = []
        for s in self.cfg.messaging:
            if broker is None and not s.requests_url.startswith("mem"):
                b, req_t, resp_t = stream_transport(
                    s.requests_url, s.responses_url
                )
            else:
                b = self.broker
                req_t = s.requests_url.split("://", 1)[-1]
                resp_t = s.responses_url.split("://", 1)[-1]
            self.messengers.append(
                Messenger(
                    b, req_t, resp_t, self.model_client, self.lb,
                    max_handlers=s.max_handlers,
                )
            )

    async def start(self) -> None:
        if hasattr(self.store, "start"):
            await self.store.start()  # warm the informer cache first
        self.lb.start()
        self.controller.start()
        self.election.start()
        self.autoscaler.start()
        for m in self.messengers:
            m.start()

    async def stop(self) -> None:
        for m in self.messengers:
            await m.stop()
        await self.autoscaler.stop()
        await self.election.stop()
        await self.controller.stop()
        await self.lb.stop()
        await self.proxy.close()
        if hasattr(self.runtime, "shutdown"):
            await self.runtime.shutdown()
        if hasattr(self.store, "stop"):
            self.store.stop()


async def run(cfg: SystemConfig) -> None:
    import uvicorn

    mgr = Manager(cfg)
    await mgr.start()
    server = uvicorn.Server(
        uvicorn.Config(
            mgr.app, host="0.0.0.0", port=cfg.api_port, log_level="warning"
        )
    )
    try:
        await server.serve()
    finally:
        await mgr.stop()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--config", default=None)
    args = p.parse_args()
    cfg = load_config(args.config)
    asyncio.run(run(cfg))


if __name__ == "__main__":
    main()
