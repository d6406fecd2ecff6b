"""bench_serving-style concurrency sweep against the gateway.

Reference parity: benchmarks/chat-py/benchmark_serving.py (TTFT / ITL /
throughput at fixed concurrency over shared prompts). Re-designed small:
N prompts with a shared long prefix (ShareGPT-like prefix reuse), fixed
concurrency via a semaphore, streaming responses timed per token.

  python benchmarks/serving_sweep.py --base-url http://127.0.0.1:8000 \
      --model m1 --num-prompts 200 --concurrency 64
"""
from __future__ import annotations

import argparse
import asyncio
import json
import random
import statistics
import time

import httpx


def pctile(xs, p):
    if not xs:
        return None
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(p / 100 * len(xs)))]


async def one_request(client, args, prompt, metrics):
    t0 = time.monotonic()
    first = None
    inter = []
    last = None
    n_tokens = 0
    try:
        async with client.stream(
            "POST",
            f"{args.base_url}/openai/v1/completions",
            json={
                "model": args.model,
                "prompt": prompt,
                "max_tokens": args.max_tokens,
                "temperature": 0,
                "stream": True,
            },
            timeout=args.timeout,
        ) as r:
            if r.status_code != 200:
                metrics["errors"] += 1
                return
            async for line in r.aiter_lines():
                if not line.startswith("data: ") or line == "data: [DONE]":
                    continue
                now = time.monotonic()
                if first is None:
                    first = now
                    metrics["ttft"].append((now - t0) * 1000)
                elif last is not None:
                    inter.append((now - last) * 1000)
                last = now
                n_tokens += 1
    except httpx.HTTPError:
        metrics["errors"] += 1
        return
    metrics["itl"].extend(inter)
    metrics["output_tokens"] += n_tokens
    metrics["latency"].append((time.monotonic() - t0) * 1000)


async def main_async(args):
    rng = random.Random(0)
    shared_prefix = " ".join(f"ctx{rng.randrange(10**6)}" for _ in range(args.prefix_words))
    prompts = [
        shared_prefix + " " + " ".join(f"q{rng.randrange(10**6)}" for _ in range(args.suffix_words))
        for _ in range(args.num_prompts)
    ]
    metrics = {"ttft": [], "itl": [], "latency": [], "output_tokens": 0, "errors": 0}
    sem = asyncio.Semaphore(args.concurrency)

    async def guarded(client, p):
        async with sem:
            await one_request(client, args, p, metrics)

    async with httpx.AsyncClient() as client:
        t0 = time.monotonic()
        await asyncio.gather(*[guarded(client, p) for p in prompts])
        dur = time.monotonic() - t0

    print(
        json.dumps(
            {
                "num_prompts": args.num_prompts,
                "concurrency": args.concurrency,
                "duration_s": round(dur, 2),
                "errors": metrics["errors"],
                "output_tokens_per_s": round(metrics["output_tokens"] / dur, 2),
                "mean_ttft_ms": round(statistics.mean(metrics["ttft"]), 2)
                if metrics["ttft"]
                else None,
                "p50_ttft_ms": round(pctile(metrics["ttft"], 50), 2)
                if metrics["ttft"]
                else None,
                "p99_ttft_ms": round(pctile(metrics["ttft"], 99), 2)
                if metrics["ttft"]
                else None,
                "mean_itl_ms": round(statistics.mean(metrics["itl"]), 2)
                if metrics["itl"]
                else None,
                "p99_itl_ms": round(pctile(metrics["itl"], 99), 2)
                if metrics["itl"]
                else None,
                "mean_latency_ms": round(statistics.mean(metrics["latency"]), 2)
                if metrics["latency"]
                else None,
            }
        )
    )


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--base-url", default="http://127.0.0.1:8000")
    p.add_argument("--model", required=True)
    p.add_argument("--num-prompts", type=int, default=200)
    p.add_argument("--concurrency", type=int, default=64)
    p.add_argument("--max-tokens", type=int, default=64)
    p.add_argument("--prefix-words", type=int, default=256)
    p.add_argument("--suffix-words", type=int, default=32)
    p.add_argument("--timeout", type=float, default=300.0)
    asyncio.run(main_async(p.parse_args()))


if __name__ == "__main__":
    main()
