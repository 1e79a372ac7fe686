#!/usr/bin/env python3
"""Synthetic monitoring: run one RANDOM example from scratch (the hourly-cron
monitor tier; reference build-and-run-example.yml:5-29).

Usage: python tools/monitor.py [--seed N]
"""
import random
import subprocess
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))
from example_utils import get_examples  # noqa: E402

# examples needing a GPU or long runtimes are excluded from the random pool
EXCLUDE = {"text_to_image", "batched_whisper", "openai_compatible_server",
           "lora_finetune", "hp_sweep_gpt", "long_training", "gpu_snapshot",
           "image_to_image", "streaming_whisper", "load_test", "protein_folding",
           "simple_torch_cluster", "torch_profiling", "bulk_embeddings",
           "image_embeddings", "render_farm", "webrtc_stream", "import_torch",
           "inference_map", "gpu_fallbacks"}


def main():
    seed = int(sys.argv[sys.argv.index("--seed") + 1]) if "--seed" in sys.argv else None
    rng = random.Random(seed)
    pool = [e for e in get_examples() if e.stem not in EXCLUDE]
    ex = rng.choice(pool)
    print(f"monitoring run: {ex.stem}")
    r = subprocess.run([sys.executable, str(Path(__file__).parent / "run_example.py"),
                        ex.stem, "--timeout", "300"])
    return r.returncode


if __name__ == "__main__":
    sys.exit(main())
