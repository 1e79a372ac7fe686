#!/usr/bin/env python3
"""Flux-class MMDiT 1024px bench: img/s at schnell shape, hipGraph on."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from modal_examples_amd.models.flux import FluxPipeline, MMDiTConfig  # noqa: E402


def main(batch=1, steps=4, iters=5, warmup=2):
    t0 = time.time()
    pipe = FluxPipeline(MMDiTConfig.schnell(), latent_size=128)
    init_s = time.time() - t0
    prompts = ["a lighthouse in a storm"] * batch
    for _ in range(warmup):
        pipe.generate(prompts, steps=steps)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        pipe.generate(prompts, steps=steps)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(json.dumps({
        "metric": "flux-schnell-class 1024px images/sec", "value": round(batch / dt, 3),
        "ms_per_image": round(dt / batch * 1e3, 1), "params_b": round(pipe.param_count() / 1e9, 2),
        "batch": batch, "steps": steps, "init_s": round(init_s, 2),
        "hipgraph": True, "dtype": "bf16", "data": "synthetic prompts, random-init weights"}))


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--steps", type=int, default=4)
    ap.add_argument("--iters", type=int, default=5)
    main(**vars(ap.parse_args()))
