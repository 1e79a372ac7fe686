# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/13_sandboxes/harbor_evals.py"]
# ---
# # Coding-eval harness in sandboxes (the harbor_evals role)
#
# Run a suite of coding tasks, each candidate solution executed against its
# tests in an ISOLATED Sandbox (untrusted code never touches the harness
# process), fan the suite out with `.map`, aggregate a pass@1 report.

import modal_examples_amd as modal

app = modal.App("example-harbor-evals")

# (task, candidate solution, test) — one deliberately wrong, one crashing
SUITE = [
    ("add", "def add(a, b):\n    return a + b", "assert add(2, 3) == 5"),
    ("fib", "def fib(n):\n    a, b = 0, 1\n"
     "    for _ in range(n): a, b = b, a + b\n    return a",
     "assert fib(10) == 55"),
    ("rev", "def rev(s):\n    return s[::-1]", "assert rev('abc') == 'cba'"),
    ("bad", "def mul(a, b):\n    return a + b", "assert mul(3, 4) == 12"),
    ("boom", "def f():\n    raise RuntimeError('boom')", "f()"),
]


@app.function(timeout=120)
def run_eval(name: str, solution: str, test: str) -> dict:
    """One eval = one sandbox: write the program, execute, judge by rc."""
    program = f"{solution}\n\n{test}\nprint('PASS')\n"
    sb = modal.Sandbox.create(app=app, timeout=60)
    try:
        p = sb.exec("python", "-c", program)
        p.wait()
        out = p.stdout.read()
        return {"task": name, "passed": p.returncode == 0 and "PASS" in out,
                "rc": p.returncode}
    finally:
        sb.terminate()


@app.local_entrypoint()
def main():
    results = list(run_eval.starmap(SUITE))
    passed = [r["task"] for r in results if r["passed"]]
    failed = [r["task"] for r in results if not r["passed"]]
    rate = len(passed) / len(results)
    print(f"pass@1 = {rate:.2f}  passed={passed}  failed={failed}")
    assert set(passed) == {"add", "fib", "rev"}, results
    assert set(failed) == {"bad", "boom"}, results
