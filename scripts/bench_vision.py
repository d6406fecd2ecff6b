"""LLaVA-1.5-7B-shape vision serving latencies on one MI355X.

Measures (random-init weights, synthetic images):
  - vision tower encode latency per image (CLIP ViT-L/14-336, 576
    patches -> projector)
  - end-to-end chat request with one image: TTFT and decode rate
    through the engine (prompt = text + 576 spliced image positions)
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from kubeai_amd.engine import EngineConfig, LLMEngine, SamplingParams
from kubeai_amd.utils import imaging


def main():
    dev = "cuda"
    eng = LLMEngine(EngineConfig(model="llava-1.5-7b", device=dev,
                                 max_model_len=4096))
    tower = eng.runner.vision
    g = torch.Generator().manual_seed(0)
    img = torch.randint(0, 256, (480, 640, 3), generator=g,
                        dtype=torch.uint8)
    t0 = time.monotonic()
    px = imaging.preprocess(img, tower.image_size)
    t_pre = time.monotonic() - t0

    # tower encode: warm then timed
    for _ in range(3):
        tower.encode(px.unsqueeze(0))
    torch.cuda.synchronize()
    t0 = time.monotonic()
    n = 10
    for _ in range(n):
        tower.encode(px.unsqueeze(0))
    torch.cuda.synchronize()
    t_enc = (time.monotonic() - t0) / n

    # batched encode (4 images at once)
    px4 = px.unsqueeze(0).expand(4, -1, -1, -1).contiguous()
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(n):
        tower.encode(px4)
    torch.cuda.synchronize()
    t_enc4 = (time.monotonic() - t0) / n

    # e2e: one image + short text, measure TTFT + decode
    img_id = eng.arch.image_token_id
    prompt = [1] + list(range(1000, 1030)) + [img_id] + list(range(2000, 2010))
    t0 = time.monotonic()
    eng.add_request(prompt, SamplingParams(max_tokens=32, ignore_eos=True),
                    request_id="v", images=[px])
    ttft = None
    toks = 0
    t_first = None
    while eng.has_work():
        for o in eng.step():
            if o.new_token_ids and ttft is None:
                ttft = time.monotonic() - t0
                t_first = time.monotonic()
            toks += len(o.new_token_ids)
    t_total = time.monotonic() - (t_first or t0)
    print(f"preprocess(CPU): {t_pre*1e3:.1f} ms")
    print(f"tower encode (1 img, 576 patches): {t_enc*1e3:.2f} ms")
    print(f"tower encode (4 imgs): {t_enc4*1e3:.2f} ms ({t_enc4/4*1e3:.2f}/img)")
    print(f"e2e 1-image chat: TTFT {ttft*1e3:.1f} ms, "
          f"{(toks-1)/max(t_total,1e-9):.0f} tok/s decode")


if __name__ == "__main__":
    main()
