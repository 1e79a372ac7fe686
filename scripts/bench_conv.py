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




def kernel_exp():
    """Hand conv3x3 kernel (K3) vs MIOpen NCHW per hot shape + TF rate."""
    import math

    from modal_examples_amd.gpu import kernel_cache
    from modal_examples_amd.ops import functional as F

    kernel_cache.restore()
    shapes = [  # (B, Cin, Cout, H, W)
        (4, 128, 128, 1024, 1024),
        (4, 256, 256, 512, 512),
        (4, 512, 512, 256, 256),
        (4, 512, 512, 128, 128),
        (4, 512, 256, 512, 512),
        (4, 320, 320, 128, 128),
        (4, 640, 640, 64, 64),
        (4, 1280, 1280, 32, 32),
        (4, 4, 512, 128, 128),
    ]
    print(f"{'shape':26s} {'miopen':>8s} {'hand':>8s} {'speedup':>8s} {'TF':>6s}")
    tot_mi = tot_hand = 0.0
    for B, Ci, Co, H, W in shapes:
        x = torch.randn(B, Ci, H, W, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(Co, Ci, 3, 3, device="cuda",
                        dtype=torch.bfloat16) / math.sqrt(Ci * 9)
        b = torch.randn(Co, device="cuda", dtype=torch.float32)
        conv = torch.nn.Conv2d(Ci, Co, 3, padding=1).to("cuda", torch.bfloat16)
        with torch.no_grad():
            conv.weight.copy_(w)
            conv.bias.copy_(b.to(torch.bfloat16))
        t_mi = timeit(lambda: conv(x))
        wr = F.repack_conv3x3_weight(w)
        t_k = timeit(lambda: F.conv3x3(x, wr, b, Co, raw_weight=w))
        flops = 2.0 * B * Ci * Co * 9 * H * W
        tf = flops / (t_k * 1e-3) / 1e12
        tot_mi += t_mi
        tot_hand += t_k
        print(f"{B}x{Ci}->{Co}@{H:<4}x{W:<6} {t_mi:8.3f} {t_k:8.3f} "
              f"{t_mi / t_k:7.2f}x {tf:6.0f}")
    print(f"{'TOTAL':26s} {tot_mi:8.3f} {tot_hand:8.3f} {tot_mi/tot_hand:7.2f}x")


if __name__ == "__main__":
    import sys as _sys

    if "--kernel" in _sys.argv:
        kernel_exp()
    else:
        conv_exp()
        gemm_exp()
