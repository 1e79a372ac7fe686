# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/03_scaling_out/cls_with_options.py"]
# ---
# Runtime resource override with `.with_options` — same class, different
# pool limits, without redeploying.

import os

import modal_examples_amd as modal

app = modal.App("example-with-options")


@app.cls(max_containers=1)
class Worker:
    @modal.method()
    def pid(self, i: int = 0) -> int:
        return os.getpid()


@app.local_entrypoint()
def main():
    small = Worker()
    pids_small = {small.pid.remote() for _ in range(4)}
    print("1-container pool pids:", pids_small)
    assert len(pids_small) == 1

    Big = Worker.with_options(max_containers=4)
    pids_big = set(Big().pid.map(range(8)))  # warms several workers
    print(f"with_options(max_containers=4) scaled to {len(pids_big)} workers")
