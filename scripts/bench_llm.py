#!/usr/bin/env python3
"""LLM serving throughput bench (config 4): Llama-3-8B continuous batching.

Measures offline throughput the way the reference quotes vLLM's
(vllm_throughput.py:27-40: ~30k input tok/s + ~2k output tok/s per H100):
submit N requests (prompt P tokens, generate G tokens each), run the engine
to completion, report input/output tok/s.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=128)
    ap.add_argument("--gen-len", type=int, default=128)
    ap.add_argument("--layers", type=int, default=0, help="0 = full 32")
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--small", action="store_true")
    ap.add_argument("--kv-dtype", choices=("bf16", "fp8"), default="bf16")
    ap.add_argument("--spec", type=int, default=0,
                    help="ngram speculative tokens (eager decode path)")
    ap.add_argument("--prefix-cache", action="store_true")
    ap.add_argument("--chunked", type=int, default=0,
                    help="chunked-prefill size (0 = monolithic)")
    ap.add_argument("--ragged", action="store_true",
                    help="mixed prompt lengths (uniform prompt_len/4 .. "
                         "prompt_len) — exercises the one-padded-forward "
                         "ragged prefill")
    args = ap.parse_args()

    import torch

    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig

    cfg = LlamaConfig.small() if args.small else LlamaConfig.llama3_8b()
    if args.layers:
        cfg.n_layers = args.layers
    device = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    t0 = time.perf_counter()
    eng = LlamaEngine(cfg, device=device, dtype=dtype,
                      use_graph=(not args.no_graph and device == "cuda"
                                 and args.spec == 0),
                      max_batch=args.requests, kv_dtype=args.kv_dtype,
                      spec_tokens=args.spec,
                      prefix_cache=args.prefix_cache,
                      chunked_prefill=args.chunked)
    eng.warmup()  # decode-graph capture is cold-start work
    if device == "cuda":
        torch.cuda.synchronize()
    init_s = time.perf_counter() - t0

    import random

    rng = random.Random(7)
    for r in range(args.requests):
        plen = (rng.randint(args.prompt_len // 4, args.prompt_len)
                if args.ragged else args.prompt_len)
        prompt = [((i * 31 + r) % 1000) + 10 for i in range(plen)]
        eng.add_request(prompt, max_new_tokens=args.gen_len, temperature=0.0)
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.run_until_done()
    if device == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    out_toks = sum(len(r.out_tokens) for r in eng.finished.values())
    in_toks = sum(len(r.prompt) for r in eng.finished.values())
    print(json.dumps({
        "metric": "llama decode throughput",
        "model": f"llama3-8b-class ({cfg.n_layers}L)",
        "requests": args.requests,
        "prompt_len": args.prompt_len,
        "ragged": args.ragged,
        "gen_len": args.gen_len,
        "elapsed_s": round(dt, 3),
        "init_s": round(init_s, 2),
        "input_tok_per_s": round(in_toks / dt, 1),
        "output_tok_per_s": round(out_toks / dt, 1),
        "kv_blocks": eng.num_blocks,
        "kv_dtype": args.kv_dtype,
        "hipgraph": eng.use_graph,
        "spec": {"proposed": eng.spec_proposed, "accepted": eng.spec_accepted}
        if args.spec else None,
        "prefix_hit_tokens": eng.prefix_hit_tokens if args.prefix_cache else None,
    }), flush=True)


if __name__ == "__main__":
    main()
