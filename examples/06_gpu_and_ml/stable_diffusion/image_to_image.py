# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/stable_diffusion/image_to_image.py"]
# ---
# # Image-to-image refinement
#
# SDXL-refiner-style img2img: encode strength as a partial noise level, start
# the denoise loop from the noised input latent instead of pure noise.

import modal_examples_amd as modal

app = modal.App("example-image-to-image")


@app.cls(gpu="mi355x")
class Refiner:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline, euler_sigmas
        from modal_examples_amd.models.sdxl.unet import UNetConfig

        gpu = torch.cuda.is_available()
        self.torch = torch
        cfg = UNetConfig.sdxl() if gpu else UNetConfig.small()
        self.latent = 128 if gpu else 16
        self.pipe = SDXLPipeline(cfg, device="cuda" if gpu else "cpu",
                                 latent_size=self.latent, use_graph=False)
        self.euler_sigmas = euler_sigmas

    @modal.method()
    def refine(self, strength: float = 0.4, steps: int = 4, seed: int = 7) -> list:
        """Takes a synthetic 'input image' latent, refines the last
        `strength` fraction of the schedule."""
        torch = self.torch
        pipe = self.pipe
        g = torch.Generator().manual_seed(seed)
        x0 = torch.randn(1, 4, self.latent, self.latent, generator=g).to(
            pipe.device, pipe.dtype)  # stand-in encoded input image
        sigmas, timesteps = self.euler_sigmas(steps)
        start = max(1, int(steps * (1 - strength)))
        sig0 = float(sigmas[start])
        noise = torch.randn(x0.shape, generator=g).to(pipe.device, pipe.dtype)
        x = x0 + sig0 * noise
        ctx, add = pipe.encode(["refined detail"])
        x = pipe._denoise_eager(x, ctx, add, sigmas[start:], timesteps[start:], 0.0)
        img = pipe.vae(x)
        return [tuple(img.shape)]


@app.local_entrypoint()
def main(strength: float = 0.4):
    out = Refiner().refine.remote(strength)
    print("refined image tensor:", out)
