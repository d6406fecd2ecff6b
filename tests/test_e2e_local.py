"""Full-stack e2e on CPU: LocalProcessRuntime spawns a REAL engine server
subprocess; the controller reconciles it to Ready; the gateway proxies an
OpenAI request end to end (reference analog: test/e2e quickstart, which
boots a real engine in kind — here the engine is in-process-node).
"""
import asyncio

import httpx
import pytest

from kubeai_amd.controlplane.config import AutoscalingConfig, SystemConfig
from kubeai_amd.controlplane.crd import Model, ModelSpec
from kubeai_amd.controlplane.manager import Manager
from kubeai_amd.controlplane.runtime import LocalProcessRuntime


def test_local_runtime_end_to_end():
    async def body():
        cfg = SystemConfig(
            autoscaling=AutoscalingConfig(
                interval_seconds=3600, time_window_seconds=7200, state_path=None
            ),
            leader_lock_path="/tmp/kubeai-e2e-leader.lock",
            n_gpus=0,  # CPU engines
        )
        mgr = Manager(cfg)
        assert isinstance(mgr.runtime, LocalProcessRuntime)
        m = Model(
            name="tiny",
            spec=ModelSpec(
                url="hf://test/llama-tiny",
                resource_profile="cpu:1",
                min_replicas=1,
                max_replicas=1,
                args=["--max-model-len", "512", "--num-gpu-blocks", "128"],
            ),
        )
        mgr.store.apply_model(m)
        await mgr.start()
        try:
            # wait for the subprocess engine to become Ready (first torch
            # import in a cold subprocess can take a while)
            for _ in range(1200):
                reps = mgr.store.list_replicas("tiny")
                if reps and reps[0].ready:
                    break
                await asyncio.sleep(0.1)
            else:
                raise AssertionError(f"replica never ready: {reps}")

            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw", timeout=60
            ) as client:
                r = await client.post(
                    "/openai/v1/chat/completions",
                    json={
                        "model": "tiny",
                        "messages": [{"role": "user", "content": "hello there"}],
                        "max_tokens": 4,
                        "temperature": 0,
                    },
                )
                assert r.status_code == 200, r.text
                body = r.json()
                assert body["usage"]["completion_tokens"] >= 1
                assert body["choices"][0]["message"]["content"]
        finally:
            await mgr.stop()

    asyncio.run(body())


def test_local_runtime_speech_to_text():
    """SpeechToText model: runtime launches the engine with task=transcribe;
    the gateway proxies a multipart transcription request end to end."""
    import io

    import numpy as np
    from scipy.io import wavfile

    async def body():
        cfg = SystemConfig(
            autoscaling=AutoscalingConfig(
                interval_seconds=3600, time_window_seconds=7200, state_path=None
            ),
            leader_lock_path="/tmp/kubeai-e2e-stt-leader.lock",
            n_gpus=0,
        )
        mgr = Manager(cfg)
        m = Model(
            name="stt",
            spec=ModelSpec(
                url="hf://test/whisper-tiny",
                features=["SpeechToText"],
                resource_profile="cpu:1",
                min_replicas=1,
                max_replicas=1,
            ),
        )
        mgr.store.apply_model(m)
        await mgr.start()
        try:
            for _ in range(1200):
                reps = mgr.store.list_replicas("stt")
                if reps and reps[0].ready:
                    break
                await asyncio.sleep(0.1)
            else:
                raise AssertionError(f"replica never ready: {reps}")

            t = np.arange(16_000) / 16_000.0
            tone = (0.5 * np.sin(2 * np.pi * 440 * t) * 32767).astype(np.int16)
            buf = io.BytesIO()
            wavfile.write(buf, 16_000, tone)

            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw", timeout=60
            ) as client:
                r = await client.post(
                    "/openai/v1/audio/transcriptions",
                    files={"file": ("tone.wav", buf.getvalue(), "audio/wav")},
                    data={"model": "stt"},
                )
                assert r.status_code == 200, r.text
                assert isinstance(r.json()["text"], str)
        finally:
            await mgr.stop()

    asyncio.run(body())


def test_local_runtime_bert_embeddings():
    """Pure TextEmbedding model: runtime launches the engine with
    task=embed (BERT encoder); the gateway proxies /v1/embeddings."""
    async def body():
        cfg = SystemConfig(
            autoscaling=AutoscalingConfig(
                interval_seconds=3600, time_window_seconds=7200, state_path=None
            ),
            leader_lock_path="/tmp/kubeai-e2e-emb-leader.lock",
            n_gpus=0,
        )
        mgr = Manager(cfg)
        m = Model(
            name="embedder",
            spec=ModelSpec(
                url="hf://test/bert-tiny",
                features=["TextEmbedding", "Reranking"],
                resource_profile="cpu:1",
                min_replicas=1,
                max_replicas=1,
            ),
        )
        mgr.store.apply_model(m)
        await mgr.start()
        try:
            for _ in range(1200):
                reps = mgr.store.list_replicas("embedder")
                if reps and reps[0].ready:
                    break
                await asyncio.sleep(0.1)
            else:
                raise AssertionError(f"replica never ready: {reps}")
            transport = httpx.ASGITransport(app=mgr.app)
            async with httpx.AsyncClient(
                transport=transport, base_url="http://gw", timeout=60
            ) as client:
                r = await client.post(
                    "/openai/v1/embeddings",
                    json={"model": "embedder", "input": "embed me"},
                )
                assert r.status_code == 200, r.text
                assert len(r.json()["data"][0]["embedding"]) == 128
        finally:
            await mgr.stop()

    asyncio.run(body())
