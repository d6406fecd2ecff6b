"""Multimodal chat against a llava-family Model (OpenAI content parts).

The engine accepts data: image URLs (PNG or binary PPM — decoded
in-house, no egress). Run a vision model first, e.g.:

    python -m kubeai_amd.engine.server --model llava-tiny --port 8000

then:

    python examples/vision_chat.py [--url http://127.0.0.1:8000]
"""
import argparse
import base64

import httpx
import torch

from kubeai_amd.utils import imaging


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--url", default="http://127.0.0.1:8000")
    p.add_argument("--model", default="llava-tiny")
    p.add_argument("--image", default=None,
                   help="path to a PNG; omit for a synthetic test card")
    args = p.parse_args()

    if args.image:
        raw = open(args.image, "rb").read()
    else:
        # synthetic 64x64 test card
        g = torch.Generator().manual_seed(0)
        img = torch.randint(0, 256, (64, 64, 3), generator=g,
                            dtype=torch.uint8)
        raw = imaging.encode_png(img)
    url = "data:image/png;base64," + base64.b64encode(raw).decode()

    r = httpx.post(
        f"{args.url}/v1/chat/completions",
        json={
            "model": args.model,
            "messages": [{
                "role": "user",
                "content": [
                    {"type": "text", "text": "What is in this image?"},
                    {"type": "image_url", "image_url": {"url": url}},
                ],
            }],
            "max_tokens": 64,
        },
        timeout=120,
    )
    r.raise_for_status()
    body = r.json()
    print(body["choices"][0]["message"]["content"])
    print("usage:", body["usage"])


if __name__ == "__main__":
    main()
