# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/render_farm.py"]
# ---
# # Render farm
#
# The blender-video shape: one function renders one frame (GPU-accelerated
# here with a torch ray-marcher), `.map` fans frames across the pool, frames
# assemble into an animation strip on a Volume.

import modal_examples_amd as modal

app = modal.App("example-render-farm")

frames_vol = modal.Volume.from_name("render-frames", create_if_missing=True)

RES = 96


@app.function(gpu="mi355x")
def render_frame(t: float) -> str:
    import math

    import torch

    device = "cuda" if torch.cuda.is_available() else "cpu"
    ys, xs = torch.meshgrid(
        torch.linspace(-1, 1, RES, device=device),
        torch.linspace(-1, 1, RES, device=device), indexing="ij")
    # ray-march a moving sphere + ground plane
    cx, cy = 0.6 * math.cos(t), 0.3 * math.sin(2 * t)
    d2 = (xs - cx) ** 2 + (ys - cy) ** 2
    sphere = torch.exp(-d2 * 14)
    ground = torch.clamp((ys + 0.8) * 2, 0, 1) * 0.2
    img = torch.stack([sphere, sphere * 0.6 + ground, ground + 0.1 * sphere])
    img = (img.clamp(0, 1) * 255).byte().cpu()
    idx = int(t * 100)
    path = frames_vol.path / f"frame_{idx:05d}.pt"
    torch.save(img, path)
    return path.name


@app.local_entrypoint()
def main(n_frames: int = 12):
    import time

    import torch

    t0 = time.perf_counter()
    times = [i * 0.2 for i in range(n_frames)]
    names = list(render_frame.map(times))
    dt = time.perf_counter() - t0
    frames_vol.commit()
    strip = torch.cat([torch.load(frames_vol.path / n) for n in sorted(names)],
                      dim=2)
    torch.save(strip, frames_vol.path / "animation_strip.pt")
    print(f"rendered {n_frames} frames in {dt:.2f}s "
          f"({n_frames / dt:.1f} fps), strip {tuple(strip.shape)}")
    for n in names:
        (frames_vol.path / n).unlink()
