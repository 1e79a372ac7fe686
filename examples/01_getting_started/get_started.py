# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/01_getting_started/get_started.py"]
# ---
# # Get started
#
# Your first app (01_getting_started/get_started.py role): define a function,
# run it locally AND remotely, see that both agree.

import modal_examples_amd as modal

app = modal.App("example-get-started")


@app.function()
def square(x: int) -> int:
    print(f"computing {x}^2 inside a container")
    return x**2


@app.local_entrypoint()
def main():
    local = square.local(42)
    remote = square.remote(42)
    print(f"locally: {local}, remotely: {remote}")
    assert local == remote == 1764
