"""Multi-turn chat benchmark against the HTTP gateway (k6-analog).

Reference parity: benchmarks/multi-turn-chat-k6/k6.js — per-VU chat threads
whose history grows with each assistant reply, shared-iterations executor,
metrics time_per_token / new_tokens / input_tokens / http_req_duration read
from body.usage. k6 itself is not in this image, so this is the same
workload in asyncio + httpx.

Usage:
  python benchmarks/multi_turn_chat.py --base-url http://127.0.0.1:8000 \
      --model m1 --vus 80 --iterations 1000 --max-tokens 32
"""
from __future__ import annotations

import argparse
import asyncio
import json
import random
import statistics
import time

import httpx


async def vu_loop(vu_id, args, state, client):
    rng = random.Random(1000 + vu_id)
    messages = [{"role": "system", "content": "You are a helpful assistant."}]
    while True:
        async with state["iter_lock"]:
            if state["iterations_left"] <= 0:
                return
            state["iterations_left"] -= 1
        words = " ".join(f"w{rng.randrange(10_000)}" for _ in range(args.user_words))
        messages.append({"role": "user", "content": words})
        t0 = time.monotonic()
        try:
            r = await client.post(
                f"{args.base_url}/openai/v1/chat/completions",
                json={
                    "model": args.model,
                    "messages": messages,
                    "max_tokens": args.max_tokens,
                    "temperature": 0,
                },
                timeout=args.timeout,
            )
            dur = time.monotonic() - t0
            if r.status_code != 200:
                state["errors"] += 1
                messages.pop()
                continue
            body = r.json()
            usage = body.get("usage", {})
            new_toks = usage.get("completion_tokens", 0)
            state["durations"].append(dur)
            state["new_tokens"] += new_toks
            state["input_tokens"] += usage.get("prompt_tokens", 0)
            if new_toks:
                state["time_per_token"].append(dur / new_toks)
            reply = body["choices"][0]["message"]["content"]
            messages.append({"role": "assistant", "content": reply})
            if len(messages) > args.max_history:
                messages = messages[:1]
        except httpx.HTTPError:
            state["errors"] += 1
            messages.pop()


async def main_async(args):
    state = {
        "iterations_left": args.iterations,
        "iter_lock": asyncio.Lock(),
        "durations": [],
        "time_per_token": [],
        "new_tokens": 0,
        "input_tokens": 0,
        "errors": 0,
    }
    async with httpx.AsyncClient() as client:
        t0 = time.monotonic()
        await asyncio.gather(*[vu_loop(v, args, state, client) for v in range(args.vus)])
        elapsed = time.monotonic() - t0
    tpt = state["time_per_token"]
    out = {
        "vus": args.vus,
        "iterations": args.iterations - state["iterations_left"],
        "errors": state["errors"],
        "duration_s": round(elapsed, 2),
        "time_per_token_avg_ms": round(1000 * statistics.mean(tpt), 2) if tpt else None,
        "time_per_token_p90_ms": round(
            1000 * statistics.quantiles(tpt, n=10)[-1], 2
        ) if len(tpt) >= 10 else None,
        "new_tokens_per_s": round(state["new_tokens"] / elapsed, 2),
        "input_tokens_per_s": round(state["input_tokens"] / elapsed, 2),
        "http_req_duration_avg_s": round(statistics.mean(state["durations"]), 3)
        if state["durations"]
        else None,
    }
    print(json.dumps(out))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--base-url", default="http://127.0.0.1:8000")
    p.add_argument("--model", required=True)
    p.add_argument("--vus", type=int, default=80)
    p.add_argument("--iterations", type=int, default=1000)
    p.add_argument("--max-tokens", type=int, default=32)
    p.add_argument("--user-words", type=int, default=48)
    p.add_argument("--max-history", type=int, default=40)
    p.add_argument("--timeout", type=float, default=300.0)
    asyncio.run(main_async(p.parse_args()))


if __name__ == "__main__":
    main()
