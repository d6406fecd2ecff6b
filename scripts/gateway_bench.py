"""Full-stack gateway benchmark: boots the control plane (LocalProcessRuntime)
with one Model on this node's GPU(s), waits for the engine replica, then
drives benchmarks/multi_turn_chat.py's workload through the OpenAI gateway.

This is the closest single-node analog of the reference's k6-vs-gateway
setup (operator + engine pods + k6): every request crosses the proxy, the
CHWBL/LeastLoad balancer, and the engine HTTP server.

  python scripts/gateway_bench.py --vus 40 --iterations 200
"""
import argparse
import asyncio
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from kubeai_amd.controlplane.config import AutoscalingConfig, SystemConfig
from kubeai_amd.controlplane.crd import LoadBalancingSpec, Model, ModelSpec
from kubeai_amd.controlplane.manager import Manager


async def main_async(args):
    cfg = SystemConfig(
        autoscaling=AutoscalingConfig(
            interval_seconds=5, time_window_seconds=60, state_path=None
        ),
        leader_lock_path="/tmp/kubeai-gwbench-leader.lock",
    )
    mgr = Manager(cfg)
    mgr.store.apply_model(
        Model(
            name=args.model_name,
            spec=ModelSpec(
                url=f"hf://bench/{args.model}",
                resource_profile=args.resource_profile,
                min_replicas=args.replicas,
                max_replicas=max(args.replicas, 2),
                load_balancing=LoadBalancingSpec(strategy="PrefixHash"),
                args=["--max-model-len", "8192"],
            ),
        )
    )
    await mgr.start()
    import uvicorn

    server = uvicorn.Server(
        uvicorn.Config(mgr.app, host="127.0.0.1", port=args.port, log_level="warning")
    )
    serve_task = asyncio.create_task(server.serve())
    try:
        for i in range(2400):
            reps = mgr.store.list_replicas(args.model_name)
            if reps and sum(r.ready for r in reps) >= args.replicas:
                break
            await asyncio.sleep(0.5)
        else:
            raise RuntimeError(f"replicas never ready: {reps}")
        print(f"[gateway_bench] {args.replicas} replica(s) ready", file=sys.stderr)

        if args.mode == "chat":
            import benchmarks.multi_turn_chat as mtc

            bench_args = argparse.Namespace(
                base_url=f"http://127.0.0.1:{args.port}",
                model=args.model_name,
                vus=args.vus,
                iterations=args.iterations,
                max_tokens=args.max_tokens,
                user_words=48,
                max_history=40,
                timeout=300.0,
            )
            await mtc.main_async(bench_args)
        else:  # bench_serving-style sweep
            import benchmarks.serving_sweep as sweep

            bench_args = argparse.Namespace(
                base_url=f"http://127.0.0.1:{args.port}",
                model=args.model_name,
                num_prompts=args.iterations,
                concurrency=args.vus,
                max_tokens=args.max_tokens,
                prefix_words=256,
                suffix_words=32,
                timeout=300.0,
            )
            await sweep.main_async(bench_args)
    finally:
        server.should_exit = True
        await asyncio.sleep(0.5)
        await mgr.stop()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--model-name", default="llama-3-8b")
    p.add_argument("--replicas", type=int, default=1)
    p.add_argument("--vus", type=int, default=40)
    p.add_argument("--iterations", type=int, default=200)
    p.add_argument("--max-tokens", type=int, default=32)
    p.add_argument("--port", type=int, default=18080)
    p.add_argument("--resource-profile", default="amd-gpu-mi355x:1")
    p.add_argument("--mode", choices=["chat", "sweep"], default="chat")
    asyncio.run(main_async(p.parse_args()))


if __name__ == "__main__":
    main()
