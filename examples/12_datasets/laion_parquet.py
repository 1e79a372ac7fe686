# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/12_datasets/laion_parquet.py"]
# ---
# # Parquet-shard dataset pipeline (the LAION-400M ingestion role)
#
# LAION ships as parquet shards; the reference streams them onto Volumes and
# fans consumption out over containers.  Here: writer functions produce
# parquet shards in parallel onto a Volume (pyarrow), consumers `.map` over
# shards computing per-shard stats, and the driver validates global counts.

import modal_examples_amd as modal

app = modal.App("example-laion-parquet")

shards_vol = modal.Volume.from_name("laion-shards", create_if_missing=True)

ROWS_PER_SHARD = 1000


@app.function()
def write_shard(i: int) -> str:
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq

    rng = np.random.default_rng(i)
    tbl = pa.table({
        "url": [f"https://img.example/{i}/{j}.jpg" for j in range(ROWS_PER_SHARD)],
        "caption": [f"caption {i}-{j}" for j in range(ROWS_PER_SHARD)],
        "similarity": rng.random(ROWS_PER_SHARD),
        "width": rng.integers(64, 2048, ROWS_PER_SHARD),
    })
    path = shards_vol.path / f"part-{i:05d}.parquet"
    pq.write_table(tbl, path)
    shards_vol.commit()
    return path.name


@app.function()
def shard_stats(name: str) -> dict:
    import pyarrow.parquet as pq

    shards_vol.reload()
    tbl = pq.read_table(shards_vol.path / name)
    sim = tbl.column("similarity").to_numpy()
    return {"shard": name, "rows": tbl.num_rows,
            "keep": int((sim > 0.5).sum())}


@app.local_entrypoint()
def main(shards: int = 8):
    names = list(write_shard.map(range(shards)))
    stats = list(shard_stats.map(names))
    total = sum(s["rows"] for s in stats)
    kept = sum(s["keep"] for s in stats)
    assert total == shards * ROWS_PER_SHARD
    print(f"{shards} parquet shards, {total} rows, {kept} above similarity 0.5")
