# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/gpu_fallbacks.py"]
# ---
# # GPU preference lists
#
# `gpu=` accepts a preference list: the scheduler takes the first satisfiable
# entry (on this node everything resolves onto the MI355X pool; counts still
# apply).  `single_use_containers` gives every input a fresh container.

import modal_examples_amd as modal

app = modal.App("example-gpu-fallbacks")


@app.function(gpu=["mi355x:8", "mi355x:4", "mi355x"],
              single_use_containers=True)
def any_gpu(i: int) -> dict:
    import os

    return {"input": i, "pid": os.getpid(),
            "visible": os.environ.get("HIP_VISIBLE_DEVICES", "cpu")}


@app.local_entrypoint()
def main():
    results = list(any_gpu.map(range(3)))
    for r in results:
        print(r)
    pids = {r["pid"] for r in results}
    assert len(pids) == 3, "single_use_containers must not reuse workers"
