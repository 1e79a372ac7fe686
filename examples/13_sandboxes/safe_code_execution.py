# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/safe_code_execution.py"]
# ---
# Run untrusted code in a Sandbox: isolated scratch directory, exec streams,
# exit codes, timeouts.

import modal_examples_amd as modal

app = modal.App("example-sandbox")


@app.local_entrypoint()
def main():
    sb = modal.Sandbox.create(app=app, timeout=60)

    p = sb.exec("python3", "-c", "print(sum(range(10)))")
    assert p.wait() == 0
    print("sandboxed python said:", p.stdout.read().strip())

    p2 = sb.exec("python3", "-c", "import sys; sys.exit(3)")
    assert p2.wait() == 3
    print("exit codes propagate:", p2.returncode)

    with sb.open("notes.txt", "w") as f:
        f.write("sandbox filesystem is scratch\n")
    assert "notes.txt" in sb.ls(".")
    sb.terminate()
    print("sandbox terminated")
