# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/stable_diffusion/text_to_image.py", "--prompt", "a bicycle on the moon", "--batch", "2"]
# ---
# # Text-to-image on MI355X (the canonical serving example)
#
# An SDXL-class pipeline served from a container class: weights load once in
# `@modal.enter` (with a pinned-host snapshot so later cold starts restore via
# hipMemcpyAsync), `@modal.method` runs generation through the hipGraph-captured
# denoise loop, and a `@modal.fastapi_endpoint` exposes it over HTTP.

import io
import time

import modal_examples_amd as modal

app = modal.App("example-text-to-image")

image = (
    modal.Image.debian_slim(python_version="3.10")
    .env({"MIOPEN_FIND_MODE": "FAST"})
)

outputs = modal.Volume.from_name("txt2img-outputs", create_if_missing=True)


@app.cls(gpu="mi355x", image=image, scaledown_window=120,
         enable_memory_snapshot=True,
         experimental_options={"enable_gpu_snapshot": True})
class Inference:
    steps: int = modal.parameter(default=4)

    @modal.enter(snap=True)
    def load(self):
        import torch

        from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
        from modal_examples_amd.models.sdxl.unet import UNetConfig

        device = "cuda" if torch.cuda.is_available() else "cpu"
        cfg = UNetConfig.sdxl() if device == "cuda" else UNetConfig.small()
        latent = 128 if device == "cuda" else 16
        self.pipe = SDXLPipeline(cfg, device=device, latent_size=latent)

    @modal.enter(snap=False)
    def wake(self):
        pass  # post-restore hook (stream/graph state re-established lazily)

    @modal.method()
    def run(self, prompt: str, batch: int = 1, seed: int = 42) -> list:
        t0 = time.time()
        imgs = self.pipe.generate([prompt] * batch, steps=self.steps, seed=seed)
        print(f"generated {batch} image(s) in {time.time() - t0:.2f}s")
        return [png_bytes(img) for img in imgs.cpu()]

    @modal.fastapi_endpoint(method="GET", label="generate")
    def web(self, prompt: str = "a watercolor city", seed: int = 42):
        return {"prompt": prompt, "png_base64_bytes": len(self.run.local(prompt, 1, seed)[0])}


def png_bytes(img_hwc_uint8) -> bytes:
    """Minimal PNG writer (no PIL in the base env)."""
    import struct
    import zlib

    import numpy as np

    arr = np.asarray(img_hwc_uint8)
    h, w, _ = arr.shape
    raw = b"".join(b"\x00" + arr[y].tobytes() for y in range(h))

    def chunk(tag, data):
        c = struct.pack(">I", len(data)) + tag + data
        return c + struct.pack(">I", zlib.crc32(tag + data) & 0xFFFFFFFF)

    return (b"\x89PNG\r\n\x1a\n"
            + chunk(b"IHDR", struct.pack(">IIBBBBB", w, h, 8, 2, 0, 0, 0))
            + chunk(b"IDAT", zlib.compress(raw, 6))
            + chunk(b"IEND", b""))


@app.function()
@modal.asgi_app(label="ui")
def ui():
    """Minimal browser frontend (the Alpine.js-UI pattern, served inline)."""
    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse

    web = FastAPI()

    @web.get("/")
    def index():
        return HTMLResponse("""<!doctype html>
<title>MI355X txt2img</title>
<body style='font-family:sans-serif;max-width:40em;margin:2em auto'>
<h2>SDXL on MI355X</h2>
<form action='/generate' method='get'>
  <input name='prompt' size='40' value='a watercolor city'/>
  <button>Generate</button>
</form>
<p>POSTs hit the <code>generate</code> endpoint; images land on the
<code>txt2img-outputs</code> volume.</p>
</body>""")

    return web


@app.local_entrypoint()
def main(prompt: str = "a bicycle on the moon", batch: int = 1,
         steps: int = 4, seed: int = 42):
    model = Inference(steps=steps)
    pngs = model.run.remote(prompt, batch, seed)
    for i, png in enumerate(pngs):
        path = outputs.path / f"img_{i}.png"
        path.write_bytes(png)
        print(f"wrote {path} ({len(png)} bytes)")
    outputs.commit()
