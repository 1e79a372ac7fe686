"""Compile-time kernel audit (scripts/check_kernel_resources.py): every .hip
source must cross-compile for gfx950 with ZERO scratch/spills (guide rule
#20) and hold the occupancy floors of the shipped kernels.  This is the
no-GPU guard against two real round-1 failure modes: source breakage hidden
by a stale prebuilt .so, and silent register spills from runtime-indexed
arrays (measured 3x slowdown on the decode kernel)."""
import shutil
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.skipif(shutil.which("hipcc") is None, reason="hipcc not on PATH")
def test_all_kernels_compile_within_budget():
    r = subprocess.run(
        [sys.executable, str(REPO / "scripts" / "check_kernel_resources.py")],
        capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, f"\n{r.stdout[-3000:]}\n{r.stderr[-1000:]}"
