# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/flux_endpoint.py"]
# ---
# # Flux behind one HTTP endpoint (misc/flux_endpoint.py role): the minimal
# # deployable image API over the flow-matching MMDiT pipeline.

import modal_examples_amd as modal

app = modal.App("example-flux-endpoint")


@app.cls(gpu="mi355x", scaledown_window=120)
class FluxAPI:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.flux import FluxPipeline, MMDiTConfig

        gpu = torch.cuda.is_available()
        self.pipe = FluxPipeline(
            MMDiTConfig.schnell() if gpu else MMDiTConfig.small(),
            device="cuda" if gpu else "cpu",
            dtype=torch.bfloat16 if gpu else torch.float32,
            latent_size=128 if gpu else 8)

    @modal.fastapi_endpoint(method="GET", label="flux")
    def generate(self, prompt: str = "a tiny robot", steps: int = 4):
        img = self.pipe.generate([prompt], steps=steps)
        return {"prompt": prompt, "height": int(img.shape[1]),
                "width": int(img.shape[2])}


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    async def go():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            r = await c.get("/flux", params={"prompt": "a tiny robot", "steps": 2})
            return r.json()

    out = asyncio.run(go())
    assert out["height"] == out["width"] > 0
    print("endpoint returned:", out)
