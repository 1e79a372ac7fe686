#!/usr/bin/env python3
"""Isolate attention-kernel bugs with structured inputs (GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

import modal_examples_amd.ops.functional as F  # noqa: E402
import modal_examples_amd.ops.reference as ref  # noqa: E402


def check(name, out, exp, tol=3e-2):
    err = (out.float() - exp.float()).abs().max().item()
    print(f"{name:40s} max_err={err:.5f} {'OK' if err < tol else '** FAIL **'}")
    return err < tol


def main():
    torch.manual_seed(0)
    dev = "cuda"
    B, H, Sq, Sk, D = 1, 1, 16, 32, 64

    # 1. Q=0 → P uniform → O = mean over kv of V  (isolates PV path)
    q = torch.zeros(B, H, Sq, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, H, Sk, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, H, Sk, D, device=dev, dtype=torch.bfloat16)
    out = F.attention(q, k, v)
    exp = v.float().mean(dim=2, keepdim=True).expand(B, H, Sq, D)
    check("Q=0 uniform-P (PV path)", out, exp)

    # 2. V[kv][d] = kv (kv-axis integrity through PV)
    v2 = torch.arange(Sk, device=dev, dtype=torch.bfloat16)[None, None, :, None] \
        .expand(B, H, Sk, D).contiguous() / Sk
    out = F.attention(q, k, v2)
    exp = ref.attention_ref(q, k, v2)
    check("V=f(kv) (kv axis)", out, exp)

    # 3. V[kv][d] = d (d-axis integrity)
    v3 = torch.arange(D, device=dev, dtype=torch.bfloat16)[None, None, None, :] \
        .expand(B, H, Sk, D).contiguous() / D
    out = F.attention(q, k, v3)
    exp = ref.attention_ref(q, k, v3)
    check("V=f(d) (d axis)", out, exp)

    # 4. single kv block exactly (Sk=64), random
    k4 = torch.randn(B, H, 64, D, device=dev, dtype=torch.bfloat16)
    v4 = torch.randn(B, H, 64, D, device=dev, dtype=torch.bfloat16)
    q4 = torch.randn(B, H, 64, D, device=dev, dtype=torch.bfloat16)
    check("random S=64 single block", F.attention(q4, k4, v4),
          ref.attention_ref(q4, k4, v4))

    # 5. multi block S=128 (online-softmax across blocks)
    k5 = torch.randn(B, H, 128, D, device=dev, dtype=torch.bfloat16)
    v5 = torch.randn(B, H, 128, D, device=dev, dtype=torch.bfloat16)
    q5 = torch.randn(B, H, 128, D, device=dev, dtype=torch.bfloat16)
    check("random S=128 two blocks", F.attention(q5, k5, v5),
          ref.attention_ref(q5, k5, v5))

    # 6. QK^T isolation: V=Id-like won't work (D!=Sk); use K=0 → s=0 uniform
    k6 = torch.zeros(B, H, 64, D, device=dev, dtype=torch.bfloat16)
    check("K=0 uniform (QK path bypassed)", F.attention(q4, k6, v4),
          v4.float().mean(2, keepdim=True).expand(B, H, 64, D))

    # 7. D=128 variants
    for mt in ("1", "2"):
        os.environ["MODAL_AMD_FA_MT128"] = mt
        q7 = torch.randn(1, 2, 96, 128, device=dev, dtype=torch.bfloat16)
        k7 = torch.randn(1, 2, 96, 128, device=dev, dtype=torch.bfloat16)
        v7 = torch.randn(1, 2, 96, 128, device=dev, dtype=torch.bfloat16)
        check(f"random D=128 MT={mt}", F.attention(q7, k7, v7),
              ref.attention_ref(q7, k7, v7))


if __name__ == "__main__":
    main()
