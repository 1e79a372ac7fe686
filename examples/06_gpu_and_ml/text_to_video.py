# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/text_to_video.py", "--frames", "4"]
# ---
# # Text-to-video
#
# The LTX/mochi-shaped workload: a clip generated as a latent walk — prompt
# conditioning interpolated across frames, each frame denoised from a shared
# seed so the sequence is temporally coherent, frames assembled on a Volume.

import modal_examples_amd as modal

app = modal.App("example-text-to-video")

clips = modal.Volume.from_name("t2v-clips", create_if_missing=True)


@app.cls(gpu="mi355x", timeout=1200)
class VideoGen:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
        from modal_examples_amd.models.sdxl.unet import UNetConfig

        gpu = torch.cuda.is_available()
        self.torch = torch
        self.pipe = SDXLPipeline(
            UNetConfig.sdxl() if gpu else UNetConfig.small(),
            device="cuda" if gpu else "cpu",
            latent_size=128 if gpu else 16, use_graph=False)

    @modal.method()
    def clip(self, prompt_a: str, prompt_b: str, frames: int = 4,
             steps: int = 4, seed: int = 3) -> dict:
        """Interpolate conditioning a→b over `frames`; shared initial noise."""
        torch = self.torch
        pipe = self.pipe
        from modal_examples_amd.models.sdxl.pipeline import euler_sigmas

        ctx_a, add_a = pipe.encode([prompt_a])
        ctx_b, add_b = pipe.encode([prompt_b])
        g = torch.Generator().manual_seed(seed)
        sigmas, timesteps = euler_sigmas(steps)
        x0 = (torch.randn(1, 4, pipe.latent, pipe.latent, generator=g)
              * float(sigmas[0])).to(pipe.device, pipe.dtype)
        outs = []
        with torch.no_grad():
            for i in range(frames):
                t = i / max(1, frames - 1)
                ctx = (1 - t) * ctx_a + t * ctx_b
                add = (1 - t) * add_a + t * add_b
                lat = pipe._denoise_eager(x0.clone(), ctx, add, sigmas,
                                          timesteps, 0.0)
                img = pipe.vae(lat)
                img = ((img.float().clamp(-1, 1) + 1) * 127.5).to(torch.uint8)
                outs.append(img[0].cpu())
        video = torch.stack(outs)  # [T, 3, H, W]
        path = clips.path / "clip.pt"
        torch.save(video, path)
        clips.commit()
        return {"frames": frames, "shape": tuple(video.shape),
                "file": path.name}


@app.local_entrypoint()
def main(frames: int = 4):
    out = VideoGen().clip.remote("a foggy harbor at dawn",
                                 "the same harbor at sunset", frames)
    print("clip:", out)
    (clips.path / out["file"]).unlink()
