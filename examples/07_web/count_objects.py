# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/07_web/count_objects.py"]
# ---
# # Binary-upload vision endpoint
#
# The count-faces shape (reference: 07_web/count_faces.py — POST an image,
# get detections back): an `@modal.asgi_app` accepts a raw-bytes body,
# decodes it into a tensor, and runs a GPU blob detector (difference-of-
# means + threshold — dependency-free).  Raw-bytes POST avoids multipart so
# the endpoint needs nothing beyond fastapi itself.

import modal_examples_amd as modal

app = modal.App("example-count-objects")

SIZE = 64  # the endpoint accepts SIZE*SIZE grayscale uint8 payloads


def make_test_image(n_blobs: int, seed: int = 0) -> bytes:
    """A synthetic grayscale image with n bright blobs."""
    import numpy as np

    rng = np.random.default_rng(seed)
    img = rng.normal(30, 5, (SIZE, SIZE))
    # well-separated grid cells with small jitter, so counts are unambiguous
    cells = [(cy, cx) for cy in (12, 32, 52) for cx in (12, 32, 52)]
    rng.shuffle(cells)
    for cy, cx in cells[:n_blobs]:
        cy, cx = cy + int(rng.integers(-3, 4)), cx + int(rng.integers(-3, 4))
        y, x = np.ogrid[:SIZE, :SIZE]
        img += 220 * np.exp(-((y - cy) ** 2 + (x - cx) ** 2) / 8.0)
    return img.clip(0, 255).astype("uint8").tobytes()


@app.function(gpu="mi355x")
@modal.asgi_app(label="vision")
def vision_app():
    import torch
    from fastapi import FastAPI, Request

    api = FastAPI()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    @api.post("/count")
    async def count(request: Request) -> dict:
        raw = await request.body()
        if len(raw) != SIZE * SIZE:
            return {"error": f"expected {SIZE * SIZE} bytes, got {len(raw)}"}
        img = torch.frombuffer(bytearray(raw), dtype=torch.uint8)
        img = img.float().view(1, 1, SIZE, SIZE).to(device)
        # blob detector: local mean minus wide mean, threshold, count islands
        local = torch.nn.functional.avg_pool2d(img, 5, 1, 2)
        wide = torch.nn.functional.avg_pool2d(img, 31, 1, 15)
        hot = (local - wide) > 40
        # count connected components by iterated max-pool label spreading
        lab = torch.arange(SIZE * SIZE, device=device, dtype=torch.float32)
        lab = (lab.view(1, 1, SIZE, SIZE) + 1) * hot
        for _ in range(SIZE):
            lab = torch.nn.functional.max_pool2d(lab, 3, 1, 1) * hot
        n = int(lab.unique().numel()) - 1  # minus background 0
        return {"objects": n, "device": device}

    return api


@app.local_entrypoint()
def main():
    import asyncio

    import httpx

    asgi = vision_app.raw()

    async def go():
        transport = httpx.ASGITransport(app=asgi)
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://v") as c:
            for want in (1, 3, 5):
                r = await c.post(
                    "/count", content=make_test_image(want, seed=want),
                    headers={"Content-Type": "application/octet-stream"})
                got = r.json()["objects"]
                print(f"image with {want} blobs → counted {got}")
                assert got == want, (want, r.json())

    asyncio.run(go())
    print("binary-upload vision endpoint OK")
