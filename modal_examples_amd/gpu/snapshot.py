"""Weight snapshot engine (K13): pinned-host capture + hipMemcpyAsync restore.

This is the local implementation of the reference's GPU memory snapshots
(enable_gpu_snapshot, 06_gpu_and_ml/gpu_snapshot.py:41-53; warm/sleep/wake at
sglang_snapshot.py:176-218): a model's weights are captured ONCE into pinned
host DRAM; every later cold start restores them with overlapping H2D copies on
4 dedicated streams instead of re-reading/deserializing from disk — the p50
cold-start headline path (BASELINE.json).
"""
from __future__ import annotations

from typing import Dict, Iterable, Tuple

import torch

from ..ops._build import get_ext


def _align(n: int, a: int = 256) -> int:
    return (n + a - 1) // a * a


class WeightSnapshot:
    """Pinned-host copy of a named tensor set."""

    def __init__(self, handle: int, layout: Dict[str, Tuple[int, int]], total: int):
        self._h = handle
        self._layout = layout  # name -> (offset, nbytes)
        self.total_bytes = total

    @staticmethod
    def capture(tensors: Dict[str, torch.Tensor]) -> "WeightSnapshot":
        ext = get_ext(required=True)
        layout, off = {}, 0
        for name, t in tensors.items():
            n = t.numel() * t.element_size()
            layout[name] = (off, n)
            off = _align(off + n)
        h = ext.snap_create(max(off, 256))
        for name, t in tensors.items():
            ext.snap_save(h, layout[name][0], t.contiguous())
        ext.snap_sync(h)
        return WeightSnapshot(h, layout, off)

    @staticmethod
    def capture_module(module: torch.nn.Module) -> "WeightSnapshot":
        return WeightSnapshot.capture(dict(module.state_dict()))

    def restore(self, tensors: Dict[str, torch.Tensor], sync: bool = True):
        ext = get_ext(required=True)
        for name, t in tensors.items():
            off, n = self._layout[name]
            assert t.numel() * t.element_size() == n, f"shape drift for {name}"
            ext.snap_restore(self._h, off, t)
        if sync:
            ext.snap_sync(self._h)

    def restore_module(self, module: torch.nn.Module, sync: bool = True):
        self.restore(
            {k: v for k, v in module.state_dict().items()}, sync=sync
        )

    def names(self) -> Iterable[str]:
        return self._layout.keys()

    def close(self):
        if self._h is not None:
            get_ext(required=True).snap_free(self._h)
            self._h = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass
