"""Profiling wrappers: torch.profiler traces + rocprofv3 invocation.

Mirrors the reference's generic profiler Function
(06_gpu_and_ml/torch_profiling.py:116-177: wrap any function in
torch.profiler, save the trace to a Volume, print key_averages) — with the
MI355X addition of a rocprofv3 command builder for per-kernel counters
(SURVEY.md §5.1 plan).
"""
from __future__ import annotations

import os
import subprocess
from pathlib import Path
from typing import Callable, List, Optional


def profile_call(fn: Callable, *args, trace_dir: str = "traces",
                 steps: int = 3, warmup: int = 1, **kwargs) -> str:
    """Run fn under torch.profiler (CPU+GPU activities); returns the trace
    path (Perfetto/chrome-trace compatible) and prints the op table."""
    import torch
    from torch.profiler import ProfilerActivity, profile, schedule

    acts = [ProfilerActivity.CPU]
    if torch.cuda.is_available():
        acts.append(ProfilerActivity.CUDA)
    Path(trace_dir).mkdir(parents=True, exist_ok=True)
    sched = schedule(wait=0, warmup=warmup, active=steps, repeat=1)
    with profile(activities=acts, schedule=sched, record_shapes=True) as prof:
        for _ in range(warmup + steps):
            fn(*args, **kwargs)
            prof.step()
    sort_key = "cuda_time_total" if torch.cuda.is_available() else "cpu_time_total"
    print(prof.key_averages().table(sort_by=sort_key, row_limit=15))
    out = str(Path(trace_dir) / f"{getattr(fn, '__name__', 'fn')}.pt.trace.json")
    prof.export_chrome_trace(out)
    return out


def rocprof_stats_command(cmd: List[str], out_dir: str = "gpurun_out/prof",
                          name: str = "run") -> List[str]:
    """Build a rocprofv3 --stats invocation (kernel time table).  NOTE: never
    combine --pmc with trace domains in one run (node-stability rule)."""
    # ROCm 7.2 rocprofv3: --stats must combine with a tracing type, and the
    # default output is a rocpd DB whose post-processing takes many minutes —
    # csv keeps it seconds.
    return ["rocprofv3", "--kernel-trace", "--stats", "-f", "csv",
            "-d", out_dir, "-o", name, "--"] + cmd


def rocprof_counters_command(cmd: List[str], counters: Optional[List[str]] = None,
                             out_dir: str = "gpurun_out/prof",
                             name: str = "pmc") -> List[str]:
    counters = counters or ["SQ_BUSY_CYCLES", "SQ_INSTS_MFMA",
                            "SQ_LDS_BANK_CONFLICT", "TCC_EA0_RDREQ_sum"]
    return ["rocprofv3", "--pmc", ",".join(counters), "-d", out_dir, "-o",
            name, "--"] + cmd


def gpu_utilization() -> dict:
    """Sample GPU util/power/VRAM via rocm-smi (runner metrics source)."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showuse", "--showpower", "--showmeminfo", "vram",
             "--json"],
            capture_output=True, text=True, timeout=10).stdout
        import json

        return json.loads(out)
    except Exception as e:  # noqa: BLE001
        return {"error": str(e)}
