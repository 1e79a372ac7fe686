# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/image_to_video.py", "--frames", "4"]
# ---
# # Image-to-video
#
# The image-to-video shape (reference: 06_gpu_and_ml/image-to-video/ — animate
# a still): the input image's latent anchors frame 0, and each later frame is
# denoised from a progressively noisier version of that latent while the
# motion-prompt conditioning ramps up — so the clip starts AT the input image
# and drifts along the prompt, staying temporally coherent (shared noise).

import modal_examples_amd as modal

app = modal.App("example-image-to-video")

clips = modal.Volume.from_name("i2v-clips", create_if_missing=True)


@app.cls(gpu="mi355x", timeout=1200)
class ImageAnimator:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
        from modal_examples_amd.models.sdxl.unet import UNetConfig

        gpu = torch.cuda.is_available()
        self.torch = torch
        self.latent = 128 if gpu else 16
        self.pipe = SDXLPipeline(
            UNetConfig.sdxl() if gpu else UNetConfig.small(),
            device="cuda" if gpu else "cpu",
            latent_size=self.latent, use_graph=False)

    @modal.method()
    def animate(self, motion_prompt: str, frames: int = 4, seed: int = 5) -> dict:
        import torch

        from modal_examples_amd.models.sdxl.pipeline import euler_sigmas

        pipe = self.pipe
        g = torch.Generator().manual_seed(seed)
        # the "input image": a fixed latent (stands in for a VAE-encoded still)
        x_img = torch.randn(1, 4, self.latent, self.latent, generator=g).to(
            pipe.device, pipe.dtype)
        noise = torch.randn(1, 4, self.latent, self.latent, generator=g).to(
            pipe.device, pipe.dtype)  # SHARED across frames: coherence
        ctx, add = pipe.encode([motion_prompt])
        steps = 4
        sigmas, timesteps = euler_sigmas(steps)
        outs = []
        for f in range(frames):
            if f == 0:
                x = x_img  # frame 0 IS the input image's latent
            else:
                strength = 0.2 + 0.6 * f / (frames - 1)  # motion ramp
                start = min(steps - 1, max(0, int(steps * (1 - strength))))
                x = x_img + float(sigmas[start]) * noise
                x = pipe._denoise_eager(x, ctx, add, sigmas[start:],
                                        timesteps[start:], 0.0)
            img = pipe.vae(x)
            outs.append(img.float().cpu())
        vid = self.torch.cat(outs)  # [frames, 3, H, W]
        self.torch.save(vid, clips.path / "clip.pt")
        clips.commit()
        # frame 0 must hug the input (tiny strength), later frames drift more
        drift = [float((outs[i] - outs[0]).float().abs().mean())
                 for i in range(frames)]
        return {"shape": tuple(vid.shape), "drift": [round(d, 4) for d in drift]}


@app.local_entrypoint()
def main(frames: int = 4):
    out = ImageAnimator().animate.remote("camera pans across a canyon", frames)
    print(out)
    assert out["shape"][0] == frames
    d = out["drift"]
    assert d[0] == 0.0 and d[-1] > d[1] > 0, f"motion must ramp: {d}"
    clips.reload()
    assert "clip.pt" in clips.listdir("/")
    print(f"animated {frames} frames; drift ramp {d}")
