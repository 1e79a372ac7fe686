# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/12_datasets/coco.py"]
# ---
# # COCO-style dataset onto a Volume (12_datasets/coco.py role)
#
# Image dataset + annotation index ingested with a multithreaded copy, then
# integrity-verified by parallel consumers (checksums + annotation joins).

import modal_examples_amd as modal

app = modal.App("example-coco")

coco = modal.Volume.from_name("coco-mini", create_if_missing=True)


@app.function()
def ingest(n_images: int = 48) -> dict:
    import concurrent.futures
    import hashlib
    import json

    import numpy as np

    (coco.path / "images").mkdir(parents=True, exist_ok=True)
    ann = {"images": [], "annotations": []}

    def write_one(i: int):
        rng = np.random.default_rng(i)
        img = rng.integers(0, 255, (3, 32, 32), dtype=np.uint8)
        raw = img.tobytes()
        p = coco.path / "images" / f"{i:06d}.bin"
        p.write_bytes(raw)
        return i, hashlib.sha256(raw).hexdigest(), int(rng.integers(1, 5))

    with concurrent.futures.ThreadPoolExecutor(8) as pool:
        for i, digest, n_boxes in pool.map(write_one, range(n_images)):
            ann["images"].append({"id": i, "file": f"{i:06d}.bin",
                                  "sha256": digest})
            for b in range(n_boxes):
                ann["annotations"].append({"image_id": i, "bbox": [b, b, 8, 8]})
    (coco.path / "instances.json").write_text(json.dumps(ann))
    coco.commit()
    return {"images": n_images, "annotations": len(ann["annotations"])}


@app.function()
def verify_shard(ids: list) -> dict:
    import hashlib
    import json

    coco.reload()
    ann = json.loads((coco.path / "instances.json").read_text())
    by_id = {im["id"]: im for im in ann["images"]}
    ok = 0
    for i in ids:
        im = by_id[i]
        raw = (coco.path / "images" / im["file"]).read_bytes()
        if hashlib.sha256(raw).hexdigest() == im["sha256"]:
            ok += 1
    return {"checked": len(ids), "ok": ok}


@app.local_entrypoint()
def main():
    stats = ingest.remote(48)
    print("ingested:", stats)
    shards = [list(range(s, 48, 4)) for s in range(4)]
    out = list(verify_shard.map(shards))
    assert all(r["ok"] == r["checked"] for r in out), out
    print(f"verified {sum(r['checked'] for r in out)} images across "
          f"{len(out)} parallel consumers")
