"""modal.Dict / modal.Queue: named distributed state over the local store.

Reference semantics: 09_job_queues/dicts_and_queues.py:72-95 (crawler using
``Dict`` for dedup + ``Queue`` for the frontier), 13_sandboxes/sandbox_pool.py:80-292
(queue partitions, TTL pool).  Backed by sqlite (WAL) so every worker process
on the node shares them; ephemeral variants clean up on context exit.
"""
from __future__ import annotations

import contextlib
import uuid
from typing import Any, List, Optional

from ..runtime.store import DictStore, QueueStore, delete_named


class Dict:
    def __init__(self, name: str):
        self.name = name
        self._s = DictStore(name)

    @staticmethod
    def from_name(name: str, create_if_missing: bool = True) -> "Dict":
        return Dict(name)

    @staticmethod
    @contextlib.contextmanager
    def ephemeral():
        d = Dict(f"ephemeral-{uuid.uuid4().hex[:8]}")
        try:
            yield d
        finally:
            d.clear()

    @staticmethod
    def delete(name: str):
        delete_named("dict", name)

    def __getitem__(self, k):
        sentinel = object()
        v = self._s.get(k, sentinel)
        if v is sentinel:
            raise KeyError(k)
        return v

    def __setitem__(self, k, v):
        self._s.put(k, v)

    def __delitem__(self, k):
        self._s.delete(k)

    def __contains__(self, k):
        return self._s.contains(k)

    def __len__(self):
        return self._s.len()

    def get(self, k, default=None):
        return self._s.get(k, default)

    def put(self, k, v):
        self._s.put(k, v)

    def pop(self, k):
        return self._s.pop(k)

    def delete_key(self, k):
        self._s.delete(k)

    def contains(self, k):
        return self._s.contains(k)

    def put_if_absent(self, k, v) -> bool:
        """Atomic insert-if-missing; True when this caller won the claim."""
        return self._s.put_if_absent(k, v)

    def len(self):
        return self._s.len()

    def keys(self):
        return self._s.keys()

    def items(self):
        return self._s.items()

    def values(self):
        for _k, v in self._s.items():
            yield v

    def clear(self):
        self._s.clear()

    def update(self, other=None, **kw):
        if other:
            for k, v in (other.items() if hasattr(other, "items") else other):
                self._s.put(k, v)
        for k, v in kw.items():
            self._s.put(k, v)


class Queue:
    def __init__(self, name: str):
        self.name = name
        self._s = QueueStore(name)

    @staticmethod
    def from_name(name: str, create_if_missing: bool = True) -> "Queue":
        return Queue(name)

    @staticmethod
    @contextlib.contextmanager
    def ephemeral():
        q = Queue(f"ephemeral-{uuid.uuid4().hex[:8]}")
        try:
            yield q
        finally:
            q.clear(all=True)

    @staticmethod
    def delete(name: str):
        delete_named("queue", name)

    def put(self, v: Any, partition: Optional[str] = None, block=True, timeout=None):
        self._s.put_many([v], partition)

    def put_many(self, vs: List[Any], partition: Optional[str] = None):
        self._s.put_many(vs, partition)

    def get(self, partition: Optional[str] = None, block: bool = True,
            timeout: Optional[float] = None):
        got = self._s.get_many(1, partition, block=block, timeout=timeout)
        if not got:
            if block and timeout is not None:
                from queue import Empty

                raise Empty()
            return None
        return got[0]

    def get_many(self, n: int, partition: Optional[str] = None, block: bool = True,
                 timeout: Optional[float] = None) -> List[Any]:
        return self._s.get_many(n, partition, block=block, timeout=timeout)

    def len(self, partition: Optional[str] = None) -> int:
        return self._s.len(partition)

    def __len__(self):
        return self._s.len(None)

    def clear(self, partition: Optional[str] = None, all: bool = False):
        self._s.clear(partition, all=all)

    def iterate(self, partition: Optional[str] = None, item_poll_timeout: float = 0.0):
        return self._s.iterate(partition, item_poll_timeout)
