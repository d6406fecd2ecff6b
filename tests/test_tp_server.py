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


def test_tp2_multimodal_matches_tp1():
    """Images travel rank0 -> workers via the broadcast intent; each rank
    runs the replicated vision tower deterministically, so TP=2 greedy
    output must equal TP=1."""
    import base64

    import torch

    from kubeai_amd.utils import imaging

    g = torch.Generator().manual_seed(5)
    img = torch.randint(0, 256, (24, 36, 3), generator=g, dtype=torch.uint8)
    url = "data:image/png;base64," + base64.b64encode(
        imaging.encode_png(img)
    ).decode()
    body = {
        "model": "tiny",
        "messages": [{
            "role": "user",
            "content": [
                {"type": "text", "text": "describe"},
                {"type": "image_url", "image_url": {"url": url}},
            ],
        }],
        "max_tokens": 5,
        "temperature": 0,
    }
    outs = {}
    for tp in (1, 2):
        proc, port = start_server_model(tp, "llava-tiny-tp")
        try:
            wait_ready(port)
            r = httpx.post(
                f"http://127.0.0.1:{port}/v1/chat/completions", json=body,
                timeout=60,
            )
            assert r.status_code == 200, r.text
            outs[tp] = r.json()["choices"][0]["message"]["content"]
        finally:
            proc.terminate()
            proc.wait(timeout=30)
    assert outs[1] == outs[2]


def start_server_model(tp: int, model: str):
    import os

    port = free_port()
    env = dict(os.environ)
    if tp == 1:
        env["KUBEAI_FORCE_TP"] = "1"
    cmd = [
        sys.executable, "-m", "kubeai_amd.engine.server",
        "--model", model, "--served-model-name", "tiny",
        "--host", "127.0.0.1", "--port", str(port),
        "--device", "cpu", "--num-gpu-blocks", "64", "--max-model-len", "512",
        "--tensor-parallel-size", str(tp),
    ]
    return subprocess.Popen(cmd, env=env), port
