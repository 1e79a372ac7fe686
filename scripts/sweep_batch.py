"""Sweep SDXL bench batch size on one GPU: find the throughput-optimal
`--batch` for bench.py (NOTES_ROUND2 item 6).  Loads the pipeline once and
re-times generate() per batch (each batch captures its own hipGraph)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from modal_examples_amd.models.sdxl import pipeline as P


def main():
    batches = [int(b) for b in (sys.argv[1:] or [2, 4, 6, 8, 12, 16])]
    pipe = P.SDXLPipeline()
    out = {}
    for B in batches:
        prompts = [f"sweep prompt {i}" for i in range(B)]
        for _ in range(2):  # capture + settle
            pipe.generate(prompts, steps=4, guidance=0.0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        steps = 3
        for _ in range(steps):
            pipe.generate(prompts, steps=4, guidance=0.0)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        ips = steps * B / dt
        out[B] = {"ms_per_gen": round(1e3 * dt / steps, 1), "img_per_s": round(ips, 2)}
        print(f"batch {B:3d}: {out[B]['ms_per_gen']:8.1f} ms/gen  {ips:6.2f} img/s",
              flush=True)
    best = max(out, key=lambda b: out[b]["img_per_s"])
    print("BEST", best, out[best])
    with open("gpurun_out/batch_sweep.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
