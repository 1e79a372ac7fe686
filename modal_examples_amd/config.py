"""Global configuration for the local MI355X runner.

The reference platform (modal-labs/modal-examples) configures everything through
decorator kwargs and ``MODAL_*`` environment variables.  We
mirror that: all knobs here are overridable through ``MODAL_AMD_*`` env vars so
tests and the CLI can retarget state directories and GPU pool sizes.
"""
from __future__ import annotations

import os
from pathlib import Path


def _env(name: str, default: str) -> str:
    return os.environ.get(f"MODAL_AMD_{name}", default)


def state_dir() -> Path:
    """Root directory for durable local state (volumes, dicts, queues, results)."""
    d = Path(_env("STATE_DIR", os.path.join(os.path.expanduser("~"), ".modal_amd")))
    d.mkdir(parents=True, exist_ok=True)
    return d


def num_gpus() -> int:
    """Size of the local GPU pool.

    Defaults to the visible device count when torch sees GPUs, else 0.  A node of
    8× MI355X reports 8.  ``MODAL_AMD_NUM_GPUS`` overrides (used by CPU tests to
    simulate a pool).
    """
    override = os.environ.get("MODAL_AMD_NUM_GPUS")
    if override is not None:
        return int(override)
    try:
        import torch

        if torch.cuda.is_available():
            return torch.cuda.device_count()
    except Exception:
        pass
    return 0


def default_timeout() -> float:
    return float(_env("DEFAULT_TIMEOUT", "300"))


def scaledown_window() -> float:
    return float(_env("SCALEDOWN_WINDOW", "60"))


def worker_start_timeout() -> float:
    return float(_env("WORKER_START_TIMEOUT", "120"))


def is_verbose() -> bool:
    return _env("VERBOSE", "0") not in ("0", "", "false")
