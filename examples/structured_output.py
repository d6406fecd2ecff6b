"""Schema-guided decoding (OpenAI response_format json_schema).

The sampler enforces the schema token by token: types, required keys,
enum values (with streaming prefix rejection), closed objects and local
$ref — jsonmode.SchemaValidator. Start any text model, then:

    python examples/structured_output.py [--url http://127.0.0.1:8000]
"""
import argparse
import json

import httpx


SCHEMA = {
    "type": "object",
    "properties": {
        "sentiment": {"enum": ["positive", "negative", "neutral"]},
        "confidence": {"type": "number"},
        "entities": {"type": "array", "items": {"type": "string"}},
    },
    "required": ["sentiment", "confidence"],
    "additionalProperties": False,
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--url", default="http://127.0.0.1:8000")
    p.add_argument("--model", default="llama-3-8b")
    args = p.parse_args()
    r = httpx.post(
        f"{args.url}/v1/chat/completions",
        json={
            "model": args.model,
            "messages": [{
                "role": "user",
                "content": "Classify: 'The rollout went great!' "
                           "Reply as JSON.",
            }],
            "max_tokens": 128,
            "response_format": {
                "type": "json_schema",
                "json_schema": {"name": "classification", "schema": SCHEMA},
            },
        },
        timeout=120,
    )
    r.raise_for_status()
    text = r.json()["choices"][0]["message"]["content"]
    print(text)
    obj = json.loads(text)  # guaranteed parseable and schema-shaped
    assert obj["sentiment"] in ("positive", "negative", "neutral")


if __name__ == "__main__":
    main()
