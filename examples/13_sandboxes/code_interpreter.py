# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/code_interpreter.py"]
# ---
# # Stateful sandboxed code interpreter
#
# The code-interpreter shape (reference: 13_sandboxes/simple_code_interpreter.py
# and jupyter_sandbox.py): a Sandbox runs a driver that reads JSON-framed code
# cells on stdin, `exec`s them in a PERSISTENT namespace, and replies with
# captured stdout/stderr and errors on stdout — a notebook-kernel protocol
# without a notebook, safe because the untrusted code lives in the sandbox.

import inspect
import json

import modal_examples_amd as modal

app = modal.App("example-code-interpreter")


def driver():
    """Runs INSIDE the sandbox: one JSON line in → one JSON line out."""
    import json
    import sys
    import traceback
    from contextlib import redirect_stderr, redirect_stdout
    from io import StringIO

    ns = {}
    for raw in sys.stdin:
        try:
            cell = json.loads(raw)
        except json.JSONDecodeError:
            print(json.dumps({"error": "bad frame"}), flush=True)
            continue
        out, err = StringIO(), StringIO()
        reply = {}
        try:
            with redirect_stdout(out), redirect_stderr(err):
                exec(cell.get("code", ""), ns)
        except BaseException:
            reply["error"] = traceback.format_exc().splitlines()[-1]
        reply["stdout"] = out.getvalue()
        reply["stderr"] = err.getvalue()
        print(json.dumps(reply), flush=True)


class Interpreter:
    """Client handle: a Sandbox running the driver, one cell at a time."""

    def __init__(self):
        import sys

        code = inspect.getsource(driver) + "\ndriver()\n"
        self.sb = modal.Sandbox.create("python3", "-u", "-c", code, timeout=120)
        self.proc = self.sb._main
        self._lines = iter(self.proc.stdout)

    def run(self, code: str) -> dict:
        self.proc.stdin.write(json.dumps({"code": code}) + "\n")
        self.proc.stdin.flush()
        return json.loads(next(self._lines))

    def close(self):
        self.sb.terminate()


@app.local_entrypoint()
def main():
    interp = Interpreter()
    try:
        r1 = interp.run("x = [i * i for i in range(10)]\nprint(len(x))")
        print("cell 1:", r1)
        assert r1["stdout"].strip() == "10" and "error" not in r1

        # state persists across cells (the namespace lives in the sandbox)
        r2 = interp.run("print(sum(x))")
        print("cell 2:", r2)
        assert r2["stdout"].strip() == "285"

        # errors come back structured; the interpreter survives them
        r3 = interp.run("1 / 0")
        print("cell 3:", r3)
        assert "ZeroDivisionError" in r3["error"]

        r4 = interp.run("print(x[-1])")
        assert r4["stdout"].strip() == "81", "state must survive the error"
    finally:
        interp.close()
    print("stateful code interpreter OK")
