#!/usr/bin/env python3
"""Decompose the Llama decode step: graph replay vs full engine.step."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from modal_examples_amd.models.llama.engine import LlamaEngine  # noqa: E402
from modal_examples_amd.models.llama.model import LlamaConfig  # noqa: E402


def main():
    batch = int(sys.argv[1]) if len(sys.argv) > 1 else 64
    eng = LlamaEngine(LlamaConfig.llama3_8b(), device="cuda", max_batch=batch)
    for i in range(batch):
        eng.add_request(list(range(10, 138)), max_new_tokens=4096)
    eng.step()  # prefill + first decode (captures graph)
    for _ in range(3):
        eng.step()

    # 1. raw graph replay
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        eng._graph.replay()
    torch.cuda.synchronize()
    t_replay = (time.perf_counter() - t0) / 50

    # 2. decode batch (replay + sample + bookkeeping)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        eng._decode_batch()
    torch.cuda.synchronize()
    t_decode = (time.perf_counter() - t0) / 50

    # 3. full step (admission scan + block mgmt + decode)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        eng.step()
    torch.cuda.synchronize()
    t_step = (time.perf_counter() - t0) / 50

    print(f"batch={batch}: replay={t_replay*1e3:.2f}ms "
          f"decode={t_decode*1e3:.2f}ms step={t_step*1e3:.2f}ms "
          f"-> {batch/t_step:.0f} tok/s")


if __name__ == "__main__":
    main()
