#!/usr/bin/env python3
"""Per-kernel micro-benchmarks vs speed-of-light on MI355X.

Ceilings (MI355X_MICROARCH.md): bf16 MFMA dense ≈2.5 PF, HBM ≈8 TB/s peak
(~6.3 achievable).  Prints TFLOP/s for compute kernels and GB/s for
memory-bound ones, with % of the relevant ceiling.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import modal_examples_amd.ops.functional as F  # noqa: E402

HBM_CEIL_GBS = 6300.0
MFMA_CEIL_TF = 2500.0


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_attention(results):
    shapes = [
        ("sdxl_self_s4096_d64", 4, 10, 4096, 4096, 64, False),
        ("sdxl_cross_s4096_kv77", 4, 10, 4096, 77, 64, False),
        ("sdxl_self_s1024_d64", 4, 20, 1024, 1024, 64, False),
        ("whisper_enc_s1500_d64", 8, 20, 1500, 1500, 64, False),
        ("llama_prefill_s2048_d128", 1, 32, 2048, 2048, 128, True),
        ("llama_prefill_s8192_d128", 1, 32, 8192, 8192, 128, True),
    ]
    for name, B, H, Sq, Sk, D, causal in shapes:
        q = torch.randn(B, H, Sq, D, device="cuda", dtype=torch.bfloat16)
        kv_h = H if "llama" not in name else 8
        k = torch.randn(B, kv_h, Sk, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, kv_h, Sk, D, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: F.attention(q, k, v, causal=causal))
        flops = 4.0 * B * H * Sq * Sk * D * (0.5 if causal else 1.0)
        tf = flops / dt / 1e12
        results[f"attn/{name}"] = {"ms": round(dt * 1e3, 3), "TF": round(tf, 1),
                                   "pct_peak": round(100 * tf / MFMA_CEIL_TF, 1)}


def bench_decode(results):
    for name, B, Hq, Hkv, S, D in [
        ("llama_b64_s1024", 64, 32, 8, 1024, 128),
        ("llama_b1_s4096", 1, 32, 8, 4096, 128),
        ("whisper_dec_b16_s448", 16, 20, 20, 448, 64),
    ]:
        q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
        kc = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
        vc = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16)
        lens = torch.full((B,), S, device="cuda", dtype=torch.int32)
        dt = timeit(lambda: F.paged_decode(q, kc, vc, None, lens))
        gb = 2 * B * Hkv * S * D * 2 / 1e9  # K+V bytes actually read
        results[f"decode/{name}"] = {
            "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
            "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}


def bench_memory_ops(results):
    x = torch.randn(16, 1280, 64, 64, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(1280, device="cuda")
    b = torch.randn(1280, device="cuda")
    dt = timeit(lambda: F.groupnorm_silu(x, g, b, 32))
    gb = 2 * x.numel() * 2 / 1e9
    results["gn_silu/16x1280x64x64"] = {
        "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
        "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}

    x2 = torch.randn(65536, 4096, device="cuda", dtype=torch.bfloat16)
    g2 = torch.randn(4096, device="cuda")
    dt = timeit(lambda: F.rmsnorm(x2, g2))
    gb = 2 * x2.numel() * 2 / 1e9
    results["rmsnorm/64k_rows_4096"] = {
        "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
        "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}

    x3 = torch.randn(8192, 1280, device="cuda", dtype=torch.bfloat16)
    g3 = torch.randn(1280, device="cuda")
    b3 = torch.randn(1280, device="cuda")
    dt = timeit(lambda: F.layernorm(x3, g3, b3))
    gb = 2 * x3.numel() * 2 / 1e9
    results["layernorm/8192x1280"] = {
        "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
        "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}

    a = torch.randn(64 * 1024 * 1024, device="cuda", dtype=torch.bfloat16)
    bb = torch.randn_like(a)
    dt = timeit(lambda: F.silu_mul(a, bb))
    gb = 3 * a.numel() * 2 / 1e9
    results["silu_mul/64M"] = {
        "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
        "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}

    lo = torch.randn(64, 128256, device="cuda")
    dt = timeit(lambda: F.sample(lo, 1.0, seed=1))
    gb = lo.numel() * 4 / 1e9
    results["sample/64x128k"] = {
        "ms": round(dt * 1e3, 3), "GBps": round(gb / dt, 0),
        "pct_hbm": round(100 * gb / dt / HBM_CEIL_GBS, 1)}


def bench_gemm_reference(results):
    """hipBLASLt via torch for context."""
    for n in (4096, 8192):
        a = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, n, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: a @ b)
        tf = 2 * n**3 / dt / 1e12
        results[f"hipblaslt_gemm/{n}"] = {"ms": round(dt * 1e3, 3),
                                          "TF": round(tf, 1),
                                          "pct_peak": round(100 * tf / MFMA_CEIL_TF, 1)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    results = {}
    bench_attention(results)
    bench_decode(results)
    bench_memory_ops(results)
    bench_gemm_reference(results)
    text = json.dumps(results, indent=1)
    print(text)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
