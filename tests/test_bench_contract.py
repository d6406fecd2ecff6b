"""bench.py driver contract: single- and multi-rank (gloo) CPU sanity.

The driver launches `torch.distributed.run --nproc-per-node N bench.py` for
the scaling benchmark; this guards that path (rank env handling, gloo init,
max-over-ranks timing, single rank-0 JSON line) without a GPU.
"""
import json
import os
import socket
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _parse_json_line(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {stdout!r}"
    return json.loads(lines[0])


def test_bench_single_rank_cpu():
    r = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--steps", "3",
         "--warmup", "2", "--vus", "4", "--max-turns", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    d = _parse_json_line(r.stdout)
    assert d["n_gpus"] == 1 and d["steps"] == 3
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert set(d) >= {"metric", "value", "unit", "warmup", "dtype", "data",
                      "config", "vs_baseline"}


def test_bench_two_ranks_gloo():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", "2",
         "--device", "cpu", "--steps", "3", "--warmup", "2", "--vus", "4",
         "--max-turns", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    d = _parse_json_line(r.stdout)
    assert d["n_gpus"] == 2
    # whole-job aggregate: two replicas of fixed per-replica work
    assert d["value"] > 0
