#!/usr/bin/env python3
"""Prefix-cache TTFT: time-to-first-token for a request whose 2k-token
system prompt is already cached vs cold. Usage: python scripts/bench_prefix_cache.py"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from modal_examples_amd.models.llama.engine import LlamaEngine  # noqa: E402
from modal_examples_amd.models.llama.model import LlamaConfig  # noqa: E402


def ttft(eng, prompt):
    t0 = time.perf_counter()
    rid = eng.add_request(prompt, max_new_tokens=1, temperature=0.0)
    while eng.has_work:
        eng.step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    assert eng.finished[rid].out_tokens
    return time.perf_counter() - t0


def main():
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = LlamaConfig.llama3_8b() if dev == "cuda" else LlamaConfig.small()
    eng = LlamaEngine(cfg, device=dev, dtype=torch.bfloat16,
                      use_graph=False, prefix_cache=True, eos_id=-1)
    g = torch.Generator().manual_seed(0)
    n_sys = 2048 if dev == "cuda" else 256
    system = torch.randint(0, cfg.vocab_size, (n_sys,), generator=g).tolist()
    cold = ttft(eng, system + [7, 8, 9])        # fills the cache
    hits = [ttft(eng, system + [10 + i]) for i in range(3)]
    print(json.dumps({
        "metric": f"prefill TTFT, {n_sys}+3-token prompt ({n_sys} shared)",
        "cold_s": round(cold, 4),
        "prefix_hit_s": round(min(hits), 4),
        "speedup": round(cold / min(hits), 1),
        "hit_tokens": eng.prefix_hit_tokens, "device": dev}))



if __name__ == "__main__":
    main()
