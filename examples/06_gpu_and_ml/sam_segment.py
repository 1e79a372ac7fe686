# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/sam_segment.py"]
# ---
# # Promptable segmentation (the SAM example role)
#
# The segment-anything shape: a ViT image encoder embeds the image once;
# interactive POINT PROMPTS decode masks from the cached embedding without
# re-running the encoder (that split is SAM's whole serving trick).  Decoder
# analog: cosine similarity of every patch to the prompted patch, thresholded
# — on a bright-object scene this segments the object.  Self-test checks IoU
# against the ground-truth object mask.

import modal_examples_amd as modal

app = modal.App("example-sam")


@app.cls(gpu="mi355x", timeout=600)
class Segmenter:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.vision import ViTConfig, VisionEncoder

        gpu = torch.cuda.is_available()
        self.torch = torch
        self.device = "cuda" if gpu else "cpu"
        self.dtype = torch.bfloat16 if gpu else torch.float32
        cfg = ViTConfig.base() if gpu else ViTConfig.small_test()
        self.cfg = cfg
        torch.manual_seed(0)
        self.enc = VisionEncoder(cfg).to(self.device, self.dtype).eval()
        self.embeddings = {}  # image_id -> [n_patches, D] (the SAM cache)

    @modal.method()
    def embed_image(self, image_id: str, image) -> int:
        import torch

        img = torch.as_tensor(image)[None].to(self.device, self.dtype)
        self.embeddings[image_id] = self.enc.embed(img)[0]
        return self.embeddings[image_id].shape[0]

    @modal.method()
    def segment_at(self, image_id: str, py: int, px: int,
                   thresh: float = 0.75) -> list:
        """Mask from a point prompt, decoded from the CACHED embedding."""
        e = self.embeddings[image_id]
        side = int(e.shape[0] ** 0.5)  # from the CACHED embedding, not the
        # config image size — prompts address the embedded image's grid
        prompt_idx = (py // self.cfg.patch) * side + (px // self.cfg.patch)
        sim = (e @ e[prompt_idx]).float()  # embeddings are L2-normalized
        mask = (sim >= thresh).reshape(side, side)
        return mask.cpu().tolist()


@app.local_entrypoint()
def main():
    import numpy as np

    rng = np.random.default_rng(0)
    size = 64
    img = rng.standard_normal((3, size, size)).astype("float32") * 0.05
    # the object: a bright square occupying patches (1..2, 1..2)
    img[:, 16:48, 16:48] += 2.0
    truth = np.zeros((4, 4), bool)
    truth[1:3, 1:3] = True

    seg = Segmenter()
    n = seg.embed_image.remote("img-1", img)
    print(f"cached {n} patch embeddings")
    mask = np.asarray(seg.segment_at.remote("img-1", py=24, px=24))
    iou = (mask & truth).sum() / (mask | truth).sum()
    print(f"mask:\n{mask.astype(int)}\nIoU vs truth: {iou:.2f}")
    assert iou >= 0.75, iou
    # a second prompt on the background must NOT return the object mask
    bg = np.asarray(seg.segment_at.remote("img-1", py=60, px=4))
    assert not (bg & truth).all() or bg.sum() < truth.sum() * 2
