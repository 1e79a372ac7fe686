#!/usr/bin/env python3
"""Conv memory-format experiment: NCHW vs channels_last for the SDXL/VAE
conv shapes, plus a TunableOp check for the Llama decode GEMM shapes."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import torch  # noqa: E402


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def conv_exp():
    from modal_examples_amd.gpu import kernel_cache

    kernel_cache.restore()
    shapes = [  # (B, Cin, Cout, H, W) — VAE + UNet hot convs
        (4, 128, 128, 1024, 1024),
        (4, 256, 256, 512, 512),
        (4, 512, 512, 256, 256),
        (4, 512, 512, 128, 128),
        (4, 320, 320, 128, 128),
        (4, 640, 640, 64, 64),
        (4, 1280, 1280, 32, 32),
    ]
    print(f"{'shape':28s} {'nchw_ms':>8s} {'nhwc_ms':>8s} {'speedup':>8s}")
    for B, Ci, Co, H, W in shapes:
        x = torch.randn(B, Ci, H, W, device="cuda", dtype=torch.bfloat16)
        conv = torch.nn.Conv2d(Ci, Co, 3, padding=1).to("cuda", torch.bfloat16)
        t_nchw = timeit(lambda: conv(x))
        xc = x.to(memory_format=torch.channels_last)
        convc = conv.to(memory_format=torch.channels_last)
        t_nhwc = timeit(lambda: convc(xc))
        print(f"{B}x{Ci}->{Co}@{H}x{W:<12} {t_nchw:8.2f} {t_nhwc:8.2f} "
              f"{t_nchw / t_nhwc:7.2f}x")


def gemm_exp():
    print("\nskinny decode GEMMs (batch=64 tokens):")
    for n, k in [(6144, 4096), (4096, 4096), (28672, 4096), (4096, 14336),
                 (128256, 4096)]:
        a = torch.randn(64, k, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: a @ w.T, iters=20)
        gb = (64 * k + n * k) * 2 / 1e9
        print(f"  64x{k} @ {k}x{n}: {dt:6.3f} ms  {gb/dt*1000:6.0f} GB/s")


if __name__ == "__main__":
    conv_exp()
    gemm_exp()
