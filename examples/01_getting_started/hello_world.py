# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/01_getting_started/hello_world.py"]
# ---
# # Hello, MI355X node!
#
# The three ways to call a function on the local serverless runner:
# in-process (`.local`), on a worker process (`.remote`), and fanned out
# across the autoscaling pool (`.map`).

import sys

import modal_examples_amd as modal

app = modal.App("example-hello-world")


@app.function()
def f(i: int) -> int:
    if i % 2 == 0:
        print("hello", i)
    else:
        print("world", i, file=sys.stderr)
    return i * i


@app.local_entrypoint()
def main():
    # run locally, in this process
    print("local:", f.local(1000))

    # run remotely, in a worker process on this node
    print("remote:", f.remote(1000))

    # fan out across the worker pool, results stream back in order
    total = 0
    for ret in f.map(range(200)):
        total += ret
    print("map total:", total)
