# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/08_advanced/restricted_volume.py"]
# ---
# # Read-only volume mounts
#
# A producer writes a model artifact to a volume; consumers mount it with
# `vol.read_only()` — write attempts inside the worker fail with EROFS (the
# runtime bind-mounts the volume read-only in the worker's private mount
# namespace), so a buggy consumer can't corrupt shared artifacts.
# Reference shape: 08_advanced/restricted_volumes.py.

import modal_examples_amd as modal

app = modal.App("example-restricted-volume")

artifacts = modal.Volume.from_name("shared-artifacts", create_if_missing=True)

MOUNT = "/tmp/artifacts_ro"


@app.function(volumes={"/tmp/artifacts_rw": artifacts})
def publish(version: str) -> str:
    path = f"/tmp/artifacts_rw/model-{version}.txt"
    with open(path, "w") as f:
        f.write(f"weights for {version}")
    artifacts.commit()
    return path


@app.function(volumes={MOUNT: artifacts.read_only()})
def consume(version: str) -> dict:
    import os

    with open(f"{MOUNT}/model-{version}.txt") as f:
        content = f.read()
    fs_enforced = os.environ.get("MODAL_AMD_RO_ENFORCED") == "1"
    try:
        open(f"{MOUNT}/scribble.txt", "w")
        tampered = True
    except OSError:
        tampered = False  # EROFS: the mount is enforced read-only
    # the API surface is ALWAYS read-only, even where the kernel refuses
    # the ro bind mount (no CAP_SYS_ADMIN)
    try:
        artifacts.read_only().commit()
        api_blocked = False
    except Exception:
        api_blocked = True
    return {"content": content, "tamper_blocked": not tampered,
            "fs_enforced": fs_enforced, "api_blocked": api_blocked}


@app.local_entrypoint()
def main():
    publish.remote("v1")
    out = consume.remote("v1")
    print(out)
    assert out["content"] == "weights for v1"
    assert out["api_blocked"], "read-only volume API should reject commits"
    if out["fs_enforced"]:
        assert out["tamper_blocked"], "ro bind mount should reject raw writes"
        print("read-only mount enforced at the filesystem level")
    else:
        print("kernel refused the ro bind mount here; API-level read-only "
              "enforcement verified")
