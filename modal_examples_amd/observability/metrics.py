"""Runner metrics: counters/histograms + Prometheus text exposition +
rocm-smi GPU sampling (SURVEY.md §5.5 plan: push-style metrics because
workers are ephemeral)."""
from __future__ import annotations

import threading
import time
from collections import defaultdict
from typing import Dict, Optional, Tuple

_lock = threading.Lock()
_counters: Dict[Tuple[str, tuple], float] = defaultdict(float)
_hists: Dict[Tuple[str, tuple], list] = defaultdict(list)


def _key(name: str, labels: Optional[dict]):
    return (name, tuple(sorted((labels or {}).items())))


def inc(name: str, value: float = 1.0, labels: Optional[dict] = None):
    with _lock:
        _counters[_key(name, labels)] += value


def observe(name: str, value: float, labels: Optional[dict] = None):
    with _lock:
        h = _hists[_key(name, labels)]
        h.append(value)
        if len(h) > 10000:
            del h[: len(h) // 2]


def render_prometheus() -> str:
    out = []
    with _lock:
        for (name, labels), v in sorted(_counters.items()):
            lbl = ",".join(f'{k}="{x}"' for k, x in labels)
            out.append(f"{name}{{{lbl}}} {v}")
        for (name, labels), h in sorted(_hists.items()):
            if not h:
                continue
            s = sorted(h)
            lbl = ",".join(f'{k}="{x}"' for k, x in labels)
            for q, suffix in ((0.5, "p50"), (0.9, "p90"), (0.99, "p99")):
                idx = min(len(s) - 1, int(q * len(s)))
                out.append(f'{name}_{suffix}{{{lbl}}} {s[idx]}')
            out.append(f"{name}_count{{{lbl}}} {len(s)}")
    return "\n".join(out) + "\n"


def reset():
    with _lock:
        _counters.clear()
        _hists.clear()


class GPUSampler:
    """Background rocm-smi sampler → gauges (util %, VRAM, power)."""

    def __init__(self, interval_s: float = 5.0):
        self.interval = interval_s
        self._stop = threading.Event()
        self.latest = {}
        self._t = None

    def start(self):
        self._t = threading.Thread(target=self._loop, daemon=True)
        self._t.start()
        return self

    def _loop(self):
        from .profiling import gpu_utilization

        while not self._stop.wait(self.interval):
            data = gpu_utilization()
            if "error" not in data:
                self.latest = data

    def stop(self):
        self._stop.set()
