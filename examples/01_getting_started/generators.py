# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/01_getting_started/generators.py"]
# ---
# Generator functions stream items back with `.remote_gen`.

import modal_examples_amd as modal

app = modal.App("example-generators")


@app.function()
def countdown(n: int):
    for i in range(n, 0, -1):
        yield i


@app.local_entrypoint()
def main():
    for x in countdown.remote_gen(5):
        print("tick", x)
