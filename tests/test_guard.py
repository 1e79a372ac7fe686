"""Debug-mode kernel-safety tooling (gpu/guard.py): guard-band OOB detection
and the sync-debug dispatch proxy."""
import pytest
import torch

from modal_examples_amd.gpu.guard import GuardBand, SyncProxy


def test_guard_band_clean_passes():
    g = GuardBand((4, 8), dtype=torch.bfloat16)
    g.tensor.fill_(1.0)  # in-bounds writes are fine
    g.check()
    assert g.tensor.shape == (4, 8)


def test_guard_band_catches_oob_write():
    g = GuardBand((16,), dtype=torch.float32, pad_bytes=64)
    # simulate a kernel writing one element past the end
    g.buf[g.pad + 16 * 4] = 0x00
    with pytest.raises(RuntimeError, match="wrote out of bounds"):
        g.check()


def test_guard_band_catches_underflow_write():
    g = GuardBand((16,), dtype=torch.float32, pad_bytes=64)
    g.buf[g.pad - 1] = 0x12
    with pytest.raises(RuntimeError, match="bytes before"):
        g.check()


def test_sync_proxy_forwards_calls_and_attrs():
    class FakeExt:
        version = 3

        def op(self, x):
            return x * 2

    p = SyncProxy(FakeExt())
    assert p.op(21) == 42
    assert p.version == 3


def test_debug_sync_env_switch(monkeypatch):
    from modal_examples_amd.gpu import guard

    monkeypatch.setenv("MODAL_AMD_DEBUG_SYNC", "1")
    assert guard.debug_sync_enabled()
    monkeypatch.setenv("MODAL_AMD_DEBUG_SYNC", "0")
    assert not guard.debug_sync_enabled()
