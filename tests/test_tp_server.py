"""Tensor-parallel HTTP serving: --tensor-parallel-size 2 on CPU (gloo)
must produce identical greedy completions to a TP=1 server."""
import socket
import subprocess
import sys
import time

import httpx
import pytest


def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def start_server(tp: int):
    port = free_port()
    import os

    env = dict(os.environ)
    if tp == 1:
        # route TP=1 through the TP model class too: its rank-identical
        # init is what TP=2 must reproduce
        env["KUBEAI_FORCE_TP"] = "1"
    cmd = [
        sys.executable, "-m", "kubeai_amd.engine.server",
        "--model", "llama-tiny-tp", "--served-model-name", "tiny",
        "--host", "127.0.0.1", "--port", str(port),
        "--device", "cpu", "--num-gpu-blocks", "64", "--max-model-len", "512",
        "--tensor-parallel-size", str(tp),
    ]
    proc = subprocess.Popen(cmd, env=env)
    return proc, port


def wait_ready(port, timeout=180):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            r = httpx.get(f"http://127.0.0.1:{port}/health", timeout=2)
            if r.status_code == 200:
                return
        except Exception:
            pass
        time.sleep(0.3)
    raise AssertionError("server never became healthy")


def completion(port):
    r = httpx.post(
        f"http://127.0.0.1:{port}/v1/completions",
        json={"prompt": "alpha beta gamma delta", "max_tokens": 6,
              "temperature": 0},
        timeout=120,
    )
    assert r.status_code == 200, r.text
    return r.json()["choices"][0]["text"]


@pytest.mark.timeout(600)
def test_tp2_server_matches_tp1():
    texts = {}
    for tp in (1, 2):
        proc, port = start_server(tp)
        try:
            wait_ready(port)
            texts[tp] = completion(port)
        finally:
            proc.terminate()
            try:
                proc.wait(timeout=20)
            except subprocess.TimeoutExpired:
                proc.kill()
    assert texts[1] == texts[2], texts
