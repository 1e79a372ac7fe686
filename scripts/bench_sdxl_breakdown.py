#!/usr/bin/env python3
"""Where does the SDXL step go? Times UNet fwd vs VAE decode separately and
prints the torch.profiler op table for one warm UNet call."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import torch  # noqa: E402

from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline  # noqa: E402
from modal_examples_amd.models.sdxl.unet import UNetConfig  # noqa: E402


def timeit(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    batch = int(sys.argv[1]) if len(sys.argv) > 1 else 4
    pipe = SDXLPipeline(UNetConfig.sdxl(), device="cuda", latent_size=128,
                        use_graph=False)
    x = torch.randn(batch, 4, 128, 128, device="cuda", dtype=torch.bfloat16)
    t = torch.full((batch,), 500.0, device="cuda")
    ctx, add = pipe.encode(["breakdown"] * batch)

    with torch.no_grad():
        dt_unet = timeit(lambda: pipe.unet(x, t, ctx, add))
        dt_vae = timeit(lambda: pipe.vae(x))
    print(f"batch={batch}: unet_fwd={dt_unet*1e3:.1f}ms  vae_decode={dt_vae*1e3:.1f}ms")
    print(f"denoise(4 steps)+decode estimate: {(4*dt_unet+dt_vae)*1e3:.1f}ms")

    from torch.profiler import ProfilerActivity, profile

    with torch.no_grad(), profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        pipe.unet(x, t, ctx, add)
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=18,
                                    max_name_column_width=60))
    with torch.no_grad(), profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof2:
        pipe.vae(x)
        torch.cuda.synchronize()
    print(prof2.key_averages().table(sort_by="cuda_time_total", row_limit=12,
                                     max_name_column_width=60))


if __name__ == "__main__":
    main()
