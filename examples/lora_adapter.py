"""Hot-swapped LoRA adapters through the control plane (reference analog:
examples + test/e2e lora-adapter flow).

The Model spec lists adapters; the controller loads them onto every ready
replica via the engine admin API; requests select one with the
`model_adapter` name form; CHWBL keys include the adapter so cache
affinity holds per adapter.
"""
import os

import httpx

BASE = os.environ.get("KUBEAI_BASE_URL", "http://localhost:8000/openai/v1")


def main() -> None:
    # Process-level model submission (cluster: the CRD `adapters:` field):
    #
    #   store.apply_model(Model(name="llama-3-8b", spec=ModelSpec(
    #       url="hf://meta-llama/Llama-3.1-8B-Instruct",
    #       resource_profile="amd-gpu-mi355x:1", min_replicas=1,
    #       adapters=[Adapter(name="sql", url="hf://org/sql-lora")],
    #   )))
    #
    # The gateway lists the adapter as "<model>_<adapter>":
    models = httpx.get(f"{BASE}/models").json()["data"]
    print("available:", [m["id"] for m in models])

    # select the adapter by the model_adapter name form
    r = httpx.post(
        f"{BASE}/completions",
        json={"model": "llama-3-8b_sql", "prompt": "SELECT", "max_tokens": 32},
        timeout=120,
    )
    r.raise_for_status()
    print(r.json()["choices"][0]["text"])


if __name__ == "__main__":
    main()
