"""bench.py driver contract: the exact torchrun launch the driver uses must
produce ONE valid JSON line from rank 0 (gloo on CPU, RCCL on GPU boxes)."""
import json
import socket
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _parse_json_line(out: str) -> dict:
    lines = [ln for ln in out.splitlines() if ln.startswith('{"metric"')]
    assert len(lines) == 1, f"expected exactly one JSON line, got:\n{out[-2000:]}"
    return json.loads(lines[0])


def _check_contract(d: dict, n: int):
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, f"missing {key}"
    assert d["n_gpus"] == n
    assert d["dtype"] == "bf16"
    assert d["scaling"] == "weak"
    assert d["higher_is_better"] is True
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["config"]["parallelism"] == f"dp{n}"


def test_bench_single_process_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "1", "--warmup", "1",
         "--small", "--latent", "32", "--batch", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    _check_contract(_parse_json_line(r.stdout), 1)


import pytest


@pytest.mark.parametrize("n", [2, 4])
def test_bench_torchrun_contract(n):
    """The driver's N>1 launcher: torch.distributed.run, one rank per GPU."""
    port = str(_free_port())
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(n), "--master-addr", "127.0.0.1",
         "--master-port", port, "bench.py", "--gpus", str(n), "--steps", "1",
         "--warmup", "1", "--small", "--latent", "16", "--batch", "1"],
        capture_output=True, text=True, cwd=REPO, timeout=420)
    assert r.returncode == 0, r.stderr[-2000:]
    d = _parse_json_line(r.stdout)
    _check_contract(d, n)
    assert d["config"]["global_batch"] == n  # whole-job aggregate, not per-rank
