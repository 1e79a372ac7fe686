# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/stable_diffusion/controlnet_edges.py"]
# ---
# # Structure-conditioned generation (the ControlNet example role)
#
# ControlNet steers diffusion with a conditioning image (edge map / pose).
# Analog on the SDXL pipeline: the conditioning map is downsampled into
# latent space and blended into the initial noise with a conditioning scale —
# generation is deterministic per (prompt, control) and the control map
# measurably steers the output.

import modal_examples_amd as modal

app = modal.App("example-controlnet")


@app.cls(gpu="mi355x", timeout=900, scaledown_window=120)
class ControlledSDXL:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
        from modal_examples_amd.models.sdxl.unet import UNetConfig

        gpu = torch.cuda.is_available()
        self.torch = torch
        cfg = UNetConfig.sdxl() if gpu else UNetConfig.small()
        self.pipe = SDXLPipeline(cfg, device="cuda" if gpu else "cpu",
                                 dtype=torch.bfloat16 if gpu else torch.float32,
                                 latent_size=128 if gpu else 16)

    @modal.method()
    def generate(self, prompt: str, control, scale: float = 0.6,
                 steps: int = 4, seed: int = 42) -> dict:
        torch = self.torch
        lat = self.pipe.latent
        c = torch.as_tensor(control, dtype=torch.float32)[None, None]
        c = torch.nn.functional.interpolate(c, size=(lat, lat), mode="nearest")
        c = (c - c.mean()) / (c.std() + 1e-5)
        gen = torch.Generator(device="cpu").manual_seed(seed)
        noise = torch.randn(1, 4, lat, lat, generator=gen)
        cond = (1 - scale) * noise + scale * c.expand(1, 4, lat, lat)
        # the pipeline consumes the blended latent through its seed path:
        # run eager denoise from the conditioned start
        from modal_examples_amd.models.sdxl.pipeline import euler_sigmas

        sigmas, timesteps = euler_sigmas(steps)
        x = (cond * float(sigmas[0])).to(self.pipe.device, self.pipe.dtype)
        ctx, add = self.pipe.encode([prompt])
        x = self.pipe._denoise_eager(x, ctx, add, sigmas, timesteps, 0.0)
        img = self.pipe.vae(x)
        img = ((img.float().clamp(-1, 1) + 1) * 127.5).round().to(torch.uint8)
        return {"checksum": int(img.sum()), "shape": list(img.shape)}


@app.local_entrypoint()
def main():
    import numpy as np

    edges_a = np.zeros((64, 64), "float32")
    edges_a[:, 28:36] = 1.0  # vertical bar
    edges_b = np.zeros((64, 64), "float32")
    edges_b[28:36, :] = 1.0  # horizontal bar

    m = ControlledSDXL()
    r1 = m.generate.remote("a neon sign", edges_a)
    r1_again = m.generate.remote("a neon sign", edges_a)
    r2 = m.generate.remote("a neon sign", edges_b)
    # repeatable up to the GroupNorm stats kernel's float atomics (a few
    # uint8 rounding flips); a different control map moves the output far more
    drift = abs(r1["checksum"] - r1_again["checksum"])  # GN float atomics
    tol = max(1000, int(0.0005 * r1["checksum"]))
    assert drift <= tol, (r1, r1_again)
    diff = abs(r1["checksum"] - r2["checksum"])
    assert diff > 10 * max(drift, 100), \
        f"control map must steer the output (diff {diff}, drift {drift})"
    print(f"control steers output: {r1['checksum']} vs {r2['checksum']}, "
          f"image {r1['shape']}")
