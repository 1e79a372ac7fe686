# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/01_getting_started/inference_map.py"]
# ---
# GPU inference fanned out over prompts with `.for_each` and
# `ignore_exceptions` (bad inputs are skipped, the rest proceed).

import modal_examples_amd as modal

app = modal.App("example-inference-map")

prompts = ["a cat", "a dog", "", "a spaceship", "a forest"]


@app.function(gpu="mi355x")
def classify(prompt: int):
    if not prompt:
        raise ValueError("empty prompt")
    import torch

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    score = torch.randn(4, device=dev).softmax(-1).max().item()
    print(f"{prompt!r}: {score:.3f}")
    return score


@app.local_entrypoint()
def main():
    classify.for_each(prompts, ignore_exceptions=True)
