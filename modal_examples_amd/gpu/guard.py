"""Debug-mode kernel-safety tooling (what the reference lacks).

Two affordances for hand-written HIP kernels:

- ``GuardBand``: canary-padded tensor allocation.  A kernel that writes out
  of bounds (bad tile math, wrong stride) corrupts the 0xAB guard bytes on
  either side of the payload; ``check()`` catches it at the call site
  instead of as a corrupted tensor three ops later.
- ``MODAL_AMD_DEBUG_SYNC=1``: every custom-kernel call synchronizes the
  device before returning (ops/functional wraps the extension in
  ``SyncProxy``), so an async fault (XNACK, illegal address) surfaces with
  the op that caused it rather than at the next blocking call — the
  HIP-event-ordering assert mode for the runner.
"""
from __future__ import annotations

import os

import torch

_PAT = 0xAB


class GuardBand:
    """Canary-padded tensor: ``t = GuardBand(shape, dtype, device); use
    t.tensor; t.check()`` after the kernel."""

    def __init__(self, shape, dtype=torch.bfloat16, device="cpu",
                 pad_bytes: int = 512):
        nbytes = int(torch.Size(shape).numel()) * torch.tensor([], dtype=dtype).element_size()
        # keep the payload aligned for any dtype the kernels use
        self.pad = pad_bytes
        self.buf = torch.full((self.pad + nbytes + self.pad,), _PAT,
                              dtype=torch.uint8, device=device)
        self.tensor = self.buf[self.pad:self.pad + nbytes].view(dtype).view(shape)
        self.shape = tuple(shape)

    def check(self) -> None:
        lo = self.buf[: self.pad]
        hi = self.buf[-self.pad:]
        bad_lo = int((lo != _PAT).sum())
        bad_hi = int((hi != _PAT).sum())
        if bad_lo or bad_hi:
            raise RuntimeError(
                f"guard-band corrupted around {self.shape} tensor: "
                f"{bad_lo} bytes before, {bad_hi} bytes after — the kernel "
                "wrote out of bounds")


class SyncProxy:
    """Wraps the kernel extension so every call device-synchronizes before
    returning (MODAL_AMD_DEBUG_SYNC=1): async kernel faults surface at the
    faulting op, and stream-ordering bugs become deterministic."""

    def __init__(self, ext):
        self._ext = ext

    def __getattr__(self, name):
        fn = getattr(self._ext, name)
        if not callable(fn):
            return fn

        def synced(*args, **kwargs):
            out = fn(*args, **kwargs)
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            return out

        return synced


def debug_sync_enabled() -> bool:
    return os.environ.get("MODAL_AMD_DEBUG_SYNC", "0") == "1"
