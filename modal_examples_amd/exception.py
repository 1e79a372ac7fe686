"""Exception types mirroring the surface the reference examples catch.

The reference relies on ``modal.exception.FunctionTimeoutError`` style errors for
its retry/long-training patterns (reference: 06_gpu_and_ml/long-training.py:146-153,
08_advanced/parallel_execution.py:42-48).
"""
from __future__ import annotations


class Error(Exception):
    """Base class for runner errors."""


class RemoteError(Error):
    """An exception was raised inside a worker; carries the remote traceback."""

    def __init__(self, message: str, remote_traceback: str = ""):
        super().__init__(message)
        self.remote_traceback = remote_traceback


class FunctionTimeoutError(Error):
    """The function exceeded its configured ``timeout=``."""


class InvalidError(Error):
    """Misuse of the API (bad decorator combination, unknown resource, ...)."""


class NotFoundError(Error):
    """Named resource (Volume/Dict/Queue/FunctionCall) does not exist."""


class ExecutionError(Error):
    """Worker died or the runtime failed internally while executing a call."""


class FunctionCancelledError(Error):
    """The call was cancelled via ``FunctionCall.cancel()``
    (08_advanced/poll_delayed_result.py relies on cancellation semantics)."""


class DeserializationError(Error):
    """Result or argument could not be (un)pickled."""


class OutputExpiredError(Error):
    """Spawned FunctionCall result was garbage collected from the store."""


class SandboxTimeoutError(Error):
    """Sandbox exceeded its timeout."""


class GPUUnavailableError(Error):
    """The requested GPU count cannot be satisfied by the local pool."""
