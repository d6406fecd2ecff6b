"""Use the official OpenAI SDK against the kubeai-amd gateway.

Reference analog: examples/k8s-api-clients + test/e2e/openai-python-client.
Requires a running control plane (python -m kubeai_amd.controlplane.manager)
with a Model applied — see examples/submit_model.py.
"""
import os

try:
    from openai import OpenAI  # any OpenAI-compatible client works
except ImportError:  # the image ships httpx; fall back to raw HTTP
    OpenAI = None

BASE = os.environ.get("KUBEAI_BASE_URL", "http://localhost:8000/openai/v1")
MODEL = os.environ.get("KUBEAI_MODEL", "llama-3-8b")


def main() -> None:
    if OpenAI is not None:
        client = OpenAI(base_url=BASE, api_key="not-needed")
        resp = client.chat.completions.create(
            model=MODEL,
            messages=[{"role": "user", "content": "Hello from the OpenAI SDK"}],
            max_tokens=32,
        )
        print(resp.choices[0].message.content)
        return
    import httpx

    r = httpx.post(
        f"{BASE}/chat/completions",
        json={
            "model": MODEL,
            "messages": [{"role": "user", "content": "Hello over raw HTTP"}],
            "max_tokens": 32,
        },
        timeout=120,
    )
    r.raise_for_status()
    print(r.json()["choices"][0]["message"]["content"])


if __name__ == "__main__":
    main()
