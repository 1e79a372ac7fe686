# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/12_datasets/volume_ingest.py"]
# ---
# # Dataset ingestion onto a Volume
#
# Multithreaded ingest of many shards into a shared Volume (the
# imagenet-to-volume pattern), then parallel consumption with `.map`.

import concurrent.futures

import modal_examples_amd as modal

app = modal.App("example-volume-ingest")

dataset = modal.Volume.from_name("demo-dataset", create_if_missing=True)

N_SHARDS = 16


@app.function()
def ingest() -> int:
    import hashlib

    def write_shard(i: int) -> int:
        payload = hashlib.sha256(str(i).encode()).hexdigest().encode() * 64
        (dataset.path / f"shard_{i:04d}.bin").write_bytes(payload)
        return len(payload)

    with concurrent.futures.ThreadPoolExecutor(8) as pool:
        sizes = list(pool.map(write_shard, range(N_SHARDS)))
    dataset.commit()
    return sum(sizes)


@app.function()
def process_shard(name: str) -> int:
    dataset.reload()
    return len((dataset.path / name).read_bytes())


@app.local_entrypoint()
def main():
    total = ingest.remote()
    print(f"ingested {N_SHARDS} shards, {total} bytes")
    names = sorted(dataset.listdir("/"))
    sizes = list(process_shard.map(names))
    assert sum(sizes) == total
    print(f"verified {len(sizes)} shards in parallel")
    for n in names:
        dataset.remove_file(n)
