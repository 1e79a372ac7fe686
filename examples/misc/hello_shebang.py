#!/usr/bin/env python3
# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/hello_shebang.py"]
# ---
# # Hello with a shebang (misc/hello_shebang.py role): the smallest runnable app.

import modal_examples_amd as modal

app = modal.App("example-hello-shebang")


@app.function()
def hello(name: str) -> str:
    return f"hello, {name}!"


@app.local_entrypoint()
def main(name: str = "MI355X"):
    print(hello.remote(name))
