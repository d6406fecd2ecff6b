"""Transcribe a WAV file through the kubeai-amd gateway (SpeechToText).

Apply a speech model first (see deploy/models/catalog.yaml `whisper-large`,
or process-level):

    store.apply_model(Model(name="whisper", spec=ModelSpec(
        url="hf://test/whisper-large", features=["SpeechToText"],
        resource_profile="amd-gpu-mi355x:1", min_replicas=0, max_replicas=2,
    )))

Then:  python examples/transcribe_client.py path/to/audio.wav
"""
import os
import sys

import httpx

BASE = os.environ.get("KUBEAI_BASE_URL", "http://localhost:8000/openai/v1")
MODEL = os.environ.get("KUBEAI_MODEL", "whisper")


def main() -> None:
    path = sys.argv[1] if len(sys.argv) > 1 else "audio.wav"
    with open(path, "rb") as f:
        wav = f.read()
    r = httpx.post(
        f"{BASE}/audio/transcriptions",
        files={"file": (os.path.basename(path), wav, "audio/wav")},
        data={"model": MODEL, "response_format": "verbose_json"},
        timeout=120,
    )
    r.raise_for_status()
    body = r.json()
    print(f"[{body.get('duration', '?')}s] {body['text']}")


if __name__ == "__main__":
    main()
