# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/embeddings/image_embeddings.py"]
# ---
# # High-throughput image embeddings
#
# The infinity-engine recipe: request-level concurrency (`@modal.concurrent`)
# + batched encode inside the container; throughput scales with batch size.

import modal_examples_amd as modal

app = modal.App("example-image-embeddings")


@app.cls(gpu="mi355x")
@modal.concurrent(max_inputs=4)
class ImageEncoder:
    @modal.enter()
    def load(self):
        import torch
        import torch.nn as nn

        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        dtype = torch.bfloat16 if self.device == "cuda" else torch.float32
        torch.manual_seed(0)
        # compact conv encoder (ViT-free: conv stem → pooled embedding)
        self.net = nn.Sequential(
            nn.Conv2d(3, 64, 4, stride=4), nn.SiLU(),
            nn.Conv2d(64, 128, 4, stride=4), nn.SiLU(),
            nn.Conv2d(128, 256, 4, stride=4), nn.SiLU(),
            nn.AdaptiveAvgPool2d(1), nn.Flatten(),
        ).to(self.device, dtype)
        self.torch = torch

    @modal.batched(max_batch_size=50, wait_ms=300)
    def embed(self, images: list) -> list:
        torch = self.torch
        x = torch.stack([torch.as_tensor(im) for im in images]).to(
            self.device, next(self.net.parameters()).dtype)
        with torch.no_grad():
            e = self.net(x).float()
        e = e / e.norm(dim=-1, keepdim=True)
        return e.cpu().tolist()


@app.local_entrypoint()
def main(n: int = 32):
    import time

    import numpy as np

    rng = np.random.default_rng(0)
    imgs = [rng.standard_normal((3, 64, 64)).astype("float32") for _ in range(n)]
    enc = ImageEncoder()
    t0 = time.perf_counter()
    embs = list(enc.embed.map(imgs))
    dt = time.perf_counter() - t0
    print(f"embedded {len(embs)} images in {dt:.2f}s ({n / dt:.0f} im/s), "
          f"dim {len(embs[0])}")
