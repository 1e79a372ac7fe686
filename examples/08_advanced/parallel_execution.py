# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/08_advanced/parallel_execution.py"]
# ---
# `spawn` + `gather`: launch calls without blocking, collect later;
# exceptions propagate through gather.

import modal_examples_amd as modal

app = modal.App("example-parallel")


@app.function()
def step(x: int) -> int:
    if x == 13:
        raise ValueError("unlucky input")
    return x + 1


@app.local_entrypoint()
def main():
    calls = [step.spawn(i) for i in range(5)]
    print("gathered:", modal.functions.gather(*calls))

    bad = step.spawn(13)
    try:
        bad.get()
    except ValueError as e:
        print("exception propagated:", e)
    else:
        raise AssertionError("expected ValueError")
