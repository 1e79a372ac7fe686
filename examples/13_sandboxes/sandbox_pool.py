# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/sandbox_pool.py"]
# ---
# A warm pool of sandboxes: a Queue of ready sandbox ids, TTL-based culling,
# health checks, and scheduled repair — so checkout latency is near zero.

import time

import modal_examples_amd as modal

app = modal.App("example-sandbox-pool")

POOL_SIZE = 2
TTL_S = 300

pool = modal.Queue.from_name("sandbox-pool", create_if_missing=True)


def make_sandbox() -> str:
    sb = modal.Sandbox.create(
        app=app, timeout=TTL_S,
        readiness_probe=modal.Probe.with_exec(["true"]),
    )
    return sb.object_id


@app.function()
def fill_pool() -> int:
    added = 0
    while pool.len() < POOL_SIZE:
        pool.put({"id": make_sandbox(), "born": time.time()})
        added += 1
    return added


@app.function(schedule=modal.Period(minutes=5))
def maintain_pool():
    """Cull expired entries, then refill (runs on a schedule when deployed)."""
    kept = []
    while True:
        entry = pool.get(block=False)
        if entry is None:
            break
        age = time.time() - entry["born"]
        healthy = False
        if age < TTL_S:
            try:
                sb = modal.Sandbox.from_id(entry["id"])
                healthy = sb.exec("true").wait() == 0
            except Exception:
                healthy = False
        if healthy:
            kept.append(entry)
        else:
            try:
                modal.Sandbox.from_id(entry["id"]).terminate()
            except Exception:
                pass
    pool.put_many(kept)
    fill_pool.local()


def checkout() -> modal.Sandbox:
    entry = pool.get(block=False)
    if entry is None:
        return modal.Sandbox.from_id(make_sandbox())
    return modal.Sandbox.from_id(entry["id"])


@app.local_entrypoint()
def main():
    pool.clear(all=True)
    # .local: sandboxes are owned by the creating process; a deployed pool
    # would run fill/maintain in the daemon (schedule above)
    print("filled", fill_pool.local(), "sandboxes")
    sb = checkout()
    p = sb.exec("echo", "from-warm-pool")
    p.wait()
    print(p.stdout.read().strip())
    maintain_pool.local()
    print("pool size after maintenance:", pool.len())
    # teardown
    while True:
        e = pool.get(block=False)
        if e is None:
            break
        modal.Sandbox.from_id(e["id"]).terminate()
