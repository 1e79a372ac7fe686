#!/usr/bin/env python3
"""Whisper-large-v3-class batch transcription throughput (config 3).

Reference anchor: 2.8x batched-vs-unbatched on A10G (batched_whisper.py:5-6).
Measures clips/sec at several batch sizes on one MI355X.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "FAST")

import torch  # noqa: E402

from modal_examples_amd.models.whisper.model import WhisperConfig  # noqa: E402
from modal_examples_amd.models.whisper.pipeline import WhisperPipeline  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batches", default="1,8,32")
    ap.add_argument("--clip-seconds", type=float, default=30.0)
    ap.add_argument("--max-tokens", type=int, default=32)
    ap.add_argument("--small", action="store_true")
    args = ap.parse_args()

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = WhisperConfig.small_test() if args.small or dev == "cpu" else WhisperConfig.large_v3()
    t0 = time.perf_counter()
    pipe = WhisperPipeline(cfg, device=dev,
                           dtype=torch.bfloat16 if dev == "cuda" else torch.float32)
    init_s = time.perf_counter() - t0

    results = {}
    for bs in [int(b) for b in args.batches.split(",")]:
        clips = [torch.randn(int(16000 * args.clip_seconds)) for _ in range(bs)]
        pipe.transcribe(clips[:1], max_tokens=4)  # warm
        if dev == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        outs = pipe.transcribe(clips, max_tokens=args.max_tokens)
        if dev == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        results[f"batch_{bs}"] = {
            "clips_per_s": round(bs / dt, 2),
            "audio_s_per_s": round(bs * args.clip_seconds / dt, 1),
            "ms_per_batch": round(dt * 1e3, 1),
            "tokens": sum(len(o) for o in outs),
        }
    print(json.dumps({"metric": "whisper batch transcription",
                      "model": "whisper-large-v3-class" if not args.small else "small",
                      "init_s": round(init_s, 2), **results}))


if __name__ == "__main__":
    main()
