# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/bucket_mount.py"]
# ---
# # Analyze parquet in an S3 bucket mount (s3_bucket_mount.py role)
#
# The reference mounts an S3 bucket and runs DuckDB over its parquet
# (10_integrations/s3_bucket_mount.py:63-80).  Same shape here against the
# framework's LOCAL S3-compatible endpoint: seed parquet objects over the S3
# API, mount the bucket READ-ONLY into an analysis function (the worker
# prefix-syncs the objects over HTTP — not a shared directory), aggregate
# with pandas/pyarrow (the DuckDB role), and write the result back through a
# writable mount prefix.

import modal_examples_amd as modal

app = modal.App("example-bucket-mount")

BUCKET = "demo-datalake"
taxi = modal.CloudBucketMount(BUCKET, key_prefix="yellow/2026", read_only=True)
out = modal.CloudBucketMount(BUCKET, key_prefix="reports")


@app.function(volumes={"/mnt/taxi": taxi, "/mnt/reports": out},
              scaledown_window=0.5)
def aggregate() -> dict:
    """Reads the mounted parquet prefix, writes a report object back."""
    import json
    from pathlib import Path

    import pandas as pd

    frames = [pd.read_parquet(p) for p in sorted(Path("/mnt/taxi").glob("*.parquet"))]
    df = pd.concat(frames)
    by_day = df.groupby("day")["fare"].agg(["count", "mean"])
    report = {str(d): {"rides": int(r["count"]), "avg_fare": round(r["mean"], 2)}
              for d, r in by_day.iterrows()}
    Path("/mnt/reports/summary.json").write_text(json.dumps(report))
    return {"files": len(frames), "rows": len(df), "days": len(report)}


@app.local_entrypoint()
def main():
    import time

    import numpy as np
    import pandas as pd

    from modal_examples_amd.resources.s3local import S3Client, start_s3_server

    # seed the bucket over the S3 REST API (the upstream producer role)
    c = S3Client(start_s3_server())
    rng = np.random.default_rng(0)
    for m in range(3):
        df = pd.DataFrame({
            "day": rng.integers(1, 8, 500),
            "fare": rng.gamma(2.0, 9.0, 500).round(2),
        })
        import io

        buf = io.BytesIO()
        df.to_parquet(buf)
        c.put(BUCKET, f"yellow/2026/month-{m:02d}.parquet", buf.getvalue())

    stats = aggregate.remote()
    assert stats == {"files": 3, "rows": 1500, "days": 7}, stats
    # the report landed back in the bucket (written via the mount, synced up
    # at worker exit)
    deadline = time.time() + 20
    while time.time() < deadline:
        if "reports/summary.json" in c.list(BUCKET, "reports/"):
            break
        time.sleep(0.3)
    import json

    report = json.loads(c.get(BUCKET, "reports/summary.json"))
    assert len(report) == 7
    print(f"aggregated {stats['rows']} rows from {stats['files']} parquet "
          f"objects; report day-1: {report.get('1')}")
