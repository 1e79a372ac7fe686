"""Shared pool of the node's MI355X devices.

The reference requests GPUs with type strings like ``gpu="H100"`` / ``"H200:4"``
and fallback lists (06_gpu_and_ml/gpu_fallbacks.py:21).  Locally the only
hardware is one node of MI355X, so every type string resolves to a count drawn
from the visible device pool; ``"mi355x:4"`` (or ``"H100:4"``) asks for 4 GPUs.
A strict suffix ``!`` is accepted and ignored (single GPU type here).
"""
from __future__ import annotations

import threading
from typing import Optional, Tuple

from .. import config


def parse_gpu(gpu) -> int:
    """Decorator ``gpu=`` value → device count. Accepts str, "type:count",
    lists (fallbacks — first entry wins locally), or None."""
    if gpu is None or gpu == "":
        return 0
    if isinstance(gpu, (list, tuple)):
        return parse_gpu(gpu[0]) if gpu else 0
    if isinstance(gpu, int):
        return gpu
    s = str(gpu).strip().rstrip("!")
    if ":" in s:
        _, _, count = s.partition(":")
        return max(1, int(count))
    return 1


class DevicePool:
    """Tracks which physical device indices are leased to worker processes."""

    def __init__(self, n: Optional[int] = None):
        self.n = n if n is not None else config.num_gpus()
        self._free = set(range(self.n))
        self._lock = threading.Lock()
        self._reclaim_hooks = []  # callables that reap idle GPU workers

    def acquire(self, count: int) -> Optional[Tuple[int, ...]]:
        with self._lock:
            if count > self.n:
                from ..exception import GPUUnavailableError

                raise GPUUnavailableError(
                    f"requested {count} GPUs but the node pool has {self.n}"
                )
            if len(self._free) < count:
                return None
            got = tuple(sorted(self._free)[:count])
            self._free.difference_update(got)
            return got

    def release(self, devices: Tuple[int, ...]):
        with self._lock:
            self._free.update(devices)

    def register_reclaim_hook(self, fn):
        self._reclaim_hooks.append(fn)

    def request_reclaim(self):
        """Called when an acquire fails: ask idle pools to give GPUs back."""
        for fn in list(self._reclaim_hooks):
            try:
                fn()
            except Exception:
                pass

    @property
    def free_count(self) -> int:
        with self._lock:
            return len(self._free)
