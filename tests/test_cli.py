"""CLI surface (`python -m modal_examples_amd …`): volume/dict/queue/app
subcommands against an isolated state dir, matching the `modal` CLI verbs."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture()
def cli_env(tmp_path):
    env = dict(os.environ)
    env["MODAL_AMD_STATE_DIR"] = str(tmp_path / "state")
    env["MODAL_AMD_NUM_GPUS"] = "0"
    return env


def run_cli(env, *args, **kw):
    return subprocess.run([sys.executable, "-m", "modal_examples_amd", *args],
                          capture_output=True, text=True, cwd=REPO, env=env,
                          timeout=kw.pop("timeout", 120))


def test_volume_put_ls_get_rm(cli_env, tmp_path):
    src = tmp_path / "weights.bin"
    src.write_bytes(b"W" * 32)
    assert run_cli(cli_env, "volume", "put", "cli-vol", str(src),
                   "models/w.bin").returncode == 0
    ls = run_cli(cli_env, "volume", "ls", "cli-vol", "models")
    assert "w.bin" in ls.stdout
    dst = tmp_path / "back.bin"
    assert run_cli(cli_env, "volume", "get", "cli-vol", "models/w.bin",
                   str(dst)).returncode == 0
    assert dst.read_bytes() == b"W" * 32
    assert run_cli(cli_env, "volume", "rm", "cli-vol",
                   "models/w.bin").returncode == 0
    assert "w.bin" not in run_cli(cli_env, "volume", "ls", "cli-vol",
                                  "models").stdout


def test_dict_and_queue_subcommands(cli_env):
    assert run_cli(cli_env, "dict", "set", "cli-d", "k1", "v1").returncode == 0
    assert run_cli(cli_env, "dict", "get", "cli-d", "k1").stdout.strip() == "v1"
    assert "k1\tv1" in run_cli(cli_env, "dict", "items", "cli-d").stdout
    assert run_cli(cli_env, "queue", "put", "cli-q", "job-1").returncode == 0
    assert run_cli(cli_env, "queue", "len", "cli-q").stdout.strip() == "1"
    assert run_cli(cli_env, "queue", "get", "cli-q").stdout.strip() == "job-1"


def test_app_list_empty_then_run(cli_env):
    out = run_cli(cli_env, "app", "list").stdout
    assert "no deployments" in out
    r = run_cli(cli_env, "run",
                "examples/01_getting_started/hello_world.py", timeout=180)
    assert r.returncode == 0, r.stderr[-1000:]


def test_unknown_command_errors(cli_env):
    r = run_cli(cli_env, "frobnicate")
    assert r.returncode != 0
    assert "unknown command" in (r.stderr + r.stdout)


def test_serve_binds_and_answers(cli_env):
    """`serve` boots the uvicorn ingress, answers HTTP, and exits after
    --timeout (the MODAL_SERVE_TIMEOUT self-termination contract)."""
    import socket
    import threading
    import time
    import urllib.request

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [sys.executable, "-m", "modal_examples_amd", "serve",
         "examples/07_web/basic_web.py", "--port", str(port),
         "--timeout", "20"],
        cwd=REPO, env=cli_env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        body, deadline = None, time.time() + 30
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/docs", timeout=2) as r:
                    body = r.read().decode()
                break
            except Exception:
                time.sleep(0.3)
        assert body and "swagger" in body.lower()
        assert proc.wait(timeout=40) == 0  # self-terminates after --timeout
    finally:
        if proc.poll() is None:
            proc.terminate()


def test_serve_hot_reload(tmp_path):
    """`modal serve` watches the target file: editing it swaps the served
    app in place (the reference serve behavior)."""
    import subprocess
    import sys
    import time
    import urllib.request
    from pathlib import Path

    src_v1 = '''
import modal_examples_amd as modal
app = modal.App("reload-demo")

@app.function()
@modal.fastapi_endpoint(method="GET", label="greet")
def greet():
    return {"version": 1}
'''
    f = tmp_path / "serve_mod.py"
    f.write_text(src_v1)
    repo = str(Path(__file__).resolve().parent.parent)
    import os

    env = dict(os.environ)
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    port = "18931"
    proc = subprocess.Popen(
        [sys.executable, "-m", "modal_examples_amd", "serve", str(f),
         "--port", port, "--timeout", "40"],
        cwd=repo, env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        def fetch(timeout=20):
            deadline = time.time() + timeout
            while time.time() < deadline:
                try:
                    with urllib.request.urlopen(
                            f"http://127.0.0.1:{port}/greet", timeout=2) as r:
                        import json

                        return json.loads(r.read())
                except Exception:
                    time.sleep(0.3)
            return None

        assert fetch() == {"version": 1}
        time.sleep(0.5)
        f.write_text(src_v1.replace('"version": 1', '"version": 2'))
        deadline = time.time() + 25
        got = None
        while time.time() < deadline:
            got = fetch(5)
            if got == {"version": 2}:
                break
            time.sleep(0.5)
        assert got == {"version": 2}, got
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_version_flag(cli_env):
    import subprocess
    import sys

    r = subprocess.run([sys.executable, "-m", "modal_examples_amd",
                        "--version"], capture_output=True, text=True,
                       timeout=60, env=cli_env)
    assert r.returncode == 0 and "modal_examples_amd" in r.stdout
