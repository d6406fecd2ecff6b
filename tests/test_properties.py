"""Property-based invariants (hypothesis) for the KV block manager and
the scheduler/engine loop.

Reference analog: the reference leans on Go's race detector plus heavy
unit suites for its bookkeeping; the engine's paged-KV
bookkeeping is the part a single missed refcount turns into silent
cross-request corruption, so it gets generative testing here.
"""
import random

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from kubeai_amd.engine.kvcache import BlockManager, NoFreeBlocks
from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams


# ---------------------------------------------------------- block manager
@settings(max_examples=60, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.data())
def test_block_manager_refcount_invariants(data):
    """Random allocate/seal/free interleavings keep refcounts exact:
    - every block's ref_count equals the number of live tables holding it
    - free blocks are reusable; the pool never leaks or double-frees
    - prefix matches only ever return sealed (hashed) blocks
    """
    bs = 4
    n_blocks = 32
    bm = BlockManager(num_blocks=n_blocks, block_size=bs)
    live: dict[int, list[int]] = {}  # handle -> table
    sealed: dict[int, list[int]] = {}  # handle -> hash chain
    tokens: dict[int, list[int]] = {}
    next_h = 0

    n_steps = data.draw(st.integers(20, 120))
    for _ in range(n_steps):
        ops = ["alloc", "free", "seal"]
        op = data.draw(st.sampled_from(ops))
        if op == "alloc":
            n_tok = data.draw(st.integers(1, 24))
            toks = data.draw(
                st.lists(st.integers(0, 7), min_size=n_tok, max_size=n_tok)
            )
            try:
                table, _cached = bm.allocate(
                    toks, salt=data.draw(st.integers(0, 1))
                )
            except NoFreeBlocks:
                continue  # pool full: expected under random load
            h = next_h
            next_h += 1
            live[h] = list(table)
            sealed[h] = []
            tokens[h] = list(toks)
        elif op == "free" and live:
            h = data.draw(st.sampled_from(sorted(live)))
            bm.free(live.pop(h))
            sealed.pop(h)
            tokens.pop(h)
        elif op == "seal" and live:
            h = data.draw(st.sampled_from(sorted(live)))
            table = live[h]
            i = len(sealed[h])
            if (i + 1) * bs <= len(tokens[h]) and i < len(table):
                parent = sealed[h][-1] if sealed[h] else None
                hh = bm.seal_block(
                    table, i, tuple(tokens[h][i * bs : (i + 1) * bs]),
                    parent, 0,
                )
                sealed[h].append(hh)

        # ---- invariants after every step ----
        counts: dict[int, int] = {}
        for t in live.values():
            for b in t:
                counts[b] = counts.get(b, 0) + 1
        for bid, blk in enumerate(bm.blocks):
            held = counts.get(bid, 0)
            assert blk.ref_count == held, (
                f"block {bid}: ref_count {blk.ref_count} != live holders "
                f"{held}"
            )

    # free everything; the whole pool must come back
    for t in live.values():
        bm.free(t)
    assert all(b.ref_count == 0 for b in bm.blocks)


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(seed=st.integers(0, 2**16))
def test_engine_random_traffic_invariants(seed):
    """Random request traffic (varied lengths, aborts, shared prefixes)
    never wedges the engine: every admitted request finishes or aborts,
    KV usage returns to zero, and shared prefixes reproduce greedily."""
    rng = random.Random(seed)
    eng = LLMEngine(EngineConfig(
        model="llama-tiny", device="cpu", num_gpu_blocks=48,
        enable_graphs=False, max_model_len=256, max_num_seqs=8,
    ))
    shared = [1] + [rng.randrange(100, 900) for _ in range(rng.randrange(4, 40))]
    reqs = []
    for i in range(rng.randrange(3, 9)):
        prompt = list(shared[: rng.randrange(1, len(shared))])
        prompt += [rng.randrange(100, 900) for _ in range(rng.randrange(1, 30))]
        p = SamplingParams(max_tokens=rng.randrange(1, 8), ignore_eos=True)
        reqs.append(eng.add_request(prompt, p, request_id=f"r{i}"))
    aborted = set()
    finished = {}
    for step in range(400):
        if not eng.has_work():
            break
        if step == 2 and len(reqs) > 3 and rng.random() < 0.5:
            eng.abort_request("r1")
            aborted.add("r1")
        for o in eng.step():
            if o.finished:
                finished[o.request_id] = o
    for r in reqs:
        rid = r.request_id
        assert rid in finished or rid in aborted, f"{rid} never completed"
    s = eng.stats()
    assert s["num_waiting"] == 0 and s["num_running"] == 0
    # all KV blocks returned
    assert all(b.ref_count == 0 for b in eng.scheduler.bm.blocks)


@settings(max_examples=20, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(seed=st.integers(0, 2**16))
def test_priority_admission_order(seed):
    """With a single running slot, strictly lower `priority` values are
    admitted first regardless of arrival order (FIFO within a class)."""
    rng = random.Random(seed)
    eng = LLMEngine(EngineConfig(
        model="llama-tiny", device="cpu", num_gpu_blocks=48,
        enable_graphs=False, max_model_len=128, max_num_seqs=1,
    ))
    prios = [rng.choice([0, 1, 5]) for _ in range(5)]
    for i, p in enumerate(prios):
        eng.add_request(
            [1] + [100 + i], SamplingParams(max_tokens=2, ignore_eos=True,
                                            priority=p),
            request_id=f"p{i}",
        )
    finish_order = []
    for _ in range(200):
        if not eng.has_work():
            break
        for o in eng.step():
            if o.finished:
                finish_order.append(o.request_id)
    assert len(finish_order) == 5
    finished_prios = [prios[int(r[1:])] for r in finish_order]
    assert finished_prios == sorted(finished_prios), (
        f"priority order violated: {finished_prios}"
    )


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(seed=st.integers(0, 2**16),
       budget=st.sampled_from([8, 16, 32, 64]))
def test_chunked_prefill_determinism(seed, budget):
    """Greedy output is invariant to the prefill chunking schedule
    (max_num_batched_tokens) — chunk boundaries must not leak into
    attention results."""
    rng = random.Random(seed)
    prompt = [1] + [rng.randrange(100, 900) for _ in range(rng.randrange(20, 90))]

    def run(b):
        eng = LLMEngine(EngineConfig(
            model="llama-tiny", device="cpu", num_gpu_blocks=64,
            enable_graphs=False, max_model_len=256,
            max_num_batched_tokens=b, seed=0,
        ))
        eng.add_request(prompt, SamplingParams(max_tokens=5, ignore_eos=True),
                        request_id="c")
        for _ in range(300):
            if not eng.has_work():
                break
            for o in eng.step():
                if o.finished:
                    return o.output_token_ids
        raise AssertionError("did not finish")

    assert run(budget) == run(4096)
