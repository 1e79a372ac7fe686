"""Client-side dynamic batcher for ``@modal.batched``.

Semantics from the reference (03_scaling_out/dynamic_batching.py:29,
06_gpu_and_ml/speech-to-text/batched_whisper.py:127): individual ``.remote`` /
``.map`` calls carrying single inputs are collected for up to ``wait_ms`` or
until ``max_batch_size`` inputs are waiting, then executed as ONE function call
whose arguments are lists; the returned list is split back per caller.
"""
from __future__ import annotations

import threading
import time
from typing import List, Tuple


class Batcher:
    def __init__(self, pool, max_batch_size: int, wait_ms: float):
        self.pool = pool
        self.max_batch_size = max_batch_size
        self.wait_s = wait_ms / 1000.0
        self._lock = threading.Lock()
        self._waiting: List[Tuple[tuple, dict, "threading.Event", list]] = []
        self._flusher = None
        self._first_ts = 0.0

    def enqueue(self, args, kwargs):
        """Returns a pool Call once the batch this input joined is dispatched."""
        slot = {"call": None}
        ev = threading.Event()
        with self._lock:
            self._waiting.append((args, kwargs, ev, slot))
            if len(self._waiting) == 1:
                self._first_ts = time.monotonic()
                self._flusher = threading.Timer(self.wait_s, self._flush)
                self._flusher.daemon = True
                self._flusher.start()
            if len(self._waiting) >= self.max_batch_size:
                if self._flusher:
                    self._flusher.cancel()
                batch = self._take()
            else:
                batch = None
        if batch:
            self._dispatch(batch)
        ev.wait()
        return slot["call"]

    def _take(self):
        batch, self._waiting = self._waiting[: self.max_batch_size], self._waiting[self.max_batch_size:]
        if self._waiting:
            self._first_ts = time.monotonic()
            self._flusher = threading.Timer(self.wait_s, self._flush)
            self._flusher.daemon = True
            self._flusher.start()
        return batch

    def _flush(self):
        with self._lock:
            batch = self._take()
        if batch:
            self._dispatch(batch)

    def _dispatch(self, batch):
        calls = self.pool.submit_batch([(a, k) for a, k, _, _ in batch])
        for (a, k, ev, slot), call in zip(batch, calls):
            slot["call"] = call
            ev.set()
