"""Request tracing: per-call spans from the runner, OTel-style export.

The reference exports OTLP traces/logs with ids injected (misc/parseable_otel.py:14-90)
and streams container logs to the client (hello_world.py:73-76).  No OTel SDK
ships in this environment, so spans are emitted in OTLP-compatible JSON lines
to ``<state>/traces.jsonl`` (and optionally to any callable sink).  The worker
pool emits a span per dispatched call; applications can nest their own.
"""
from __future__ import annotations

import json
import os
import threading
import time
import uuid
from contextlib import contextmanager
from typing import Optional

from .. import config

_lock = threading.Lock()
_local = threading.local()
_sinks = []


def _trace_file():
    return config.state_dir() / "traces.jsonl"


def add_sink(fn):
    _sinks.append(fn)


def enabled() -> bool:
    return os.environ.get("MODAL_AMD_TRACING", "1") not in ("0", "false")


def current_trace_id() -> Optional[str]:
    return getattr(_local, "trace_id", None)


@contextmanager
def span(name: str, **attrs):
    """Emit one span; nests under the thread's current trace."""
    if not enabled():
        yield {}
        return
    trace_id = getattr(_local, "trace_id", None) or uuid.uuid4().hex
    parent = getattr(_local, "span_id", None)
    span_id = uuid.uuid4().hex[:16]
    prev = (getattr(_local, "trace_id", None), getattr(_local, "span_id", None))
    _local.trace_id, _local.span_id = trace_id, span_id
    t0 = time.time()
    rec = {"trace_id": trace_id, "span_id": span_id, "parent_id": parent,
           "name": name, "start": t0, "attrs": attrs}
    err = None
    try:
        yield rec
    except BaseException as e:  # noqa: BLE001
        err = repr(e)
        raise
    finally:
        rec["end"] = time.time()
        rec["duration_ms"] = round((rec["end"] - t0) * 1000, 3)
        if err:
            rec["error"] = err
        _emit(rec)
        _local.trace_id, _local.span_id = prev


def _emit(rec):
    line = json.dumps(rec)
    try:
        with _lock, open(_trace_file(), "a") as f:
            f.write(line + "\n")
    except Exception:
        pass
    for s in _sinks:
        try:
            s(rec)
        except Exception:
            pass


def read_traces(limit: int = 1000):
    p = _trace_file()
    if not p.exists():
        return []
    lines = p.read_text().splitlines()[-limit:]
    return [json.loads(ln) for ln in lines]


def clear():
    p = _trace_file()
    if p.exists():
        p.unlink()
