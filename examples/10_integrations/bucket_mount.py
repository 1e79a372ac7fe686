# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/bucket_mount.py"]
# ---
# # Cloud bucket mounts
#
# `CloudBucketMount` exposes an object-store prefix as a directory (locally a
# named directory; on a network-connected deployment an S3/GCS prefix).  The
# analysis pattern: write parquet-ish shards, query them from workers.

import modal_examples_amd as modal

app = modal.App("example-bucket-mount")

bucket = modal.CloudBucketMount("demo-datalake", key_prefix="year=2026/")


@app.function()
def write_shard(day: int) -> str:
    import json

    p = bucket.path / f"day_{day:02d}.jsonl"
    rows = [{"day": day, "metric": day * 1.5 + i} for i in range(10)]
    p.write_text("\n".join(json.dumps(r) for r in rows))
    return p.name


@app.function()
def query_total() -> float:
    import json

    total = 0.0
    for f in sorted(bucket.path.glob("day_*.jsonl")):
        for line in f.read_text().splitlines():
            total += json.loads(line)["metric"]
    return total


@app.local_entrypoint()
def main():
    names = list(write_shard.map(range(5)))
    print("wrote", names)
    print("aggregate metric:", query_total.remote())
    for f in bucket.path.glob("day_*.jsonl"):
        f.unlink()
