"""Submit a Model to a running in-process control plane (library usage).

For cluster deployments, apply the CRD form instead:
    kubectl apply -f deploy/crds/kubeai.org_models.yaml
    helm install kubeai-amd deploy/kubeai-amd
    # then kubectl apply a kubeai.org/v1 Model
"""
import asyncio

from kubeai_amd.controlplane.config import SystemConfig
from kubeai_amd.controlplane.crd import LoadBalancingSpec, Model, ModelSpec
from kubeai_amd.controlplane.manager import Manager


async def main() -> None:
    mgr = Manager(SystemConfig())
    await mgr.start()
    mgr.store.apply_model(
        Model(
            name="llama-3-8b",
            spec=ModelSpec(
                url="hf://meta-llama/Meta-Llama-3-8B-Instruct",
                resource_profile="amd-gpu-mi355x:1",
                min_replicas=0,  # scale-from-zero on first request
                max_replicas=8,
                target_requests=64,
                load_balancing=LoadBalancingSpec(strategy="PrefixHash"),
            ),
        )
    )
    print("model applied; gateway at :8000 once a replica is ready")
    # serve the gateway (manager.run() does this when used as a CLI)
    import uvicorn

    await uvicorn.Server(
        uvicorn.Config(mgr.app, host="0.0.0.0", port=8000, log_level="info")
    ).serve()


if __name__ == "__main__":
    asyncio.run(main())
