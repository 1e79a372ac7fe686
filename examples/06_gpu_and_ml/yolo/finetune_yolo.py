# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/yolo/finetune_yolo.py"]
# ---
# # Fine-tune a detector, then stream inference (yolo/finetune_yolo.py role)
#
# The reference fine-tunes YOLO on a custom dataset and then streams frames
# through the tuned model at ~60 im/s (finetune_yolo.py:176).  Analog: a
# conv detector head fine-tuned on synthetic "bright square" targets with
# the fused-AdamW kernel (K9), checkpointed to a Volume, then a streamed
# `.map` inference pass that reports measured images/sec and localization
# accuracy.

import modal_examples_amd as modal

app = modal.App("example-yolo-finetune")

weights = modal.Volume.from_name("yolo-weights", create_if_missing=True)

GRID = 8  # detection grid (each cell predicts objectness)


def make_scene(rng, size=128):
    """Image with one bright square; label = grid cell containing it."""
    import numpy as np

    img = rng.standard_normal((3, size, size)).astype("float32") * 0.1
    cell = size // GRID
    gy, gx = rng.integers(0, GRID), rng.integers(0, GRID)
    y = gy * cell + rng.integers(0, cell - 8 + 1)
    x = gx * cell + rng.integers(0, cell - 8 + 1)
    img[:, y:y + 8, x:x + 8] += 2.0
    return img, gy * GRID + gx


def build_model(torch):
    import torch.nn as nn

    return nn.Sequential(
        nn.Conv2d(3, 32, 3, stride=2, padding=1), nn.SiLU(),
        nn.Conv2d(32, 64, 3, stride=2, padding=1), nn.SiLU(),
        nn.Conv2d(64, 64, 3, stride=2, padding=1), nn.SiLU(),
        nn.Conv2d(64, 64, 3, stride=2, padding=1), nn.SiLU(),
        nn.Conv2d(64, 1, 1),  # [B,1,GRID,GRID] objectness
    )


@app.function(gpu="mi355x", timeout=1200)
def finetune(steps: int = 60) -> dict:
    import numpy as np
    import torch

    from modal_examples_amd.train.lora import FusedAdamW

    device = "cuda" if torch.cuda.is_available() else "cpu"
    rng = np.random.default_rng(0)
    torch.manual_seed(0)
    model = build_model(torch).to(device)
    opt = FusedAdamW(list(model.parameters()), lr=2e-3)
    for step in range(steps):
        imgs, labels = zip(*(make_scene(rng) for _ in range(16)))
        x = torch.as_tensor(np.stack(imgs), device=device)
        y = torch.as_tensor(labels, device=device)
        logits = model(x).flatten(1)  # [B, GRID*GRID]
        loss = torch.nn.functional.cross_entropy(logits, y)
        loss.backward()
        opt.step()
        opt.zero_grad()
    torch.save(model.state_dict(), weights.path / "detector.pt")
    weights.commit()
    return {"steps": steps, "final_loss": round(float(loss), 4)}


@app.cls(gpu="mi355x")
@modal.concurrent(max_inputs=4)
class Detector:
    @modal.enter()
    def load(self):
        import torch

        weights.reload()
        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        self.model = build_model(torch).to(self.device).eval()
        self.model.load_state_dict(
            torch.load(weights.path / "detector.pt", map_location=self.device))

    @modal.method()
    def detect_batch(self, frames) -> list:
        import numpy as np

        x = self.torch.as_tensor(np.stack(frames), device=self.device)
        with self.torch.no_grad():
            cells = self.model(x).flatten(1).argmax(-1)
        return cells.tolist()


@app.local_entrypoint()
def main():
    import time

    import numpy as np

    print("finetune:", finetune.remote(steps=60))
    rng = np.random.default_rng(1)
    batches, labels = [], []
    for _ in range(8):
        scenes = [make_scene(rng) for _ in range(16)]
        batches.append([s[0] for s in scenes])
        labels.append([s[1] for s in scenes])
    det = Detector()
    det.detect_batch.remote(batches[0])  # warm (load weights)
    t0 = time.monotonic()
    preds = list(det.detect_batch.map(batches))
    dt = time.monotonic() - t0
    n = sum(len(b) for b in batches)
    correct = sum(p == l for ps, ls in zip(preds, labels) for p, l in zip(ps, ls))
    acc = correct / n
    print(f"streamed {n} frames in {dt:.2f}s = {n/dt:.0f} im/s; cell accuracy {acc:.2f}")
    assert acc > 0.8, acc
