"""hipGraph cache with an LRU bound.

A long-lived server fed many (batch, steps, ...) shapes accumulates one
captured graph + static buffer set per key; unbounded that is a slow memory
leak (r1 VERDICT weak #11).  Eviction drops the oldest entry — its
torch.cuda.CUDAGraph and static tensors free once dereferenced (graph
destruction while the runtime is live is safe; only teardown-time destruction
aborts, see LlamaEngine.close)."""
from __future__ import annotations

from collections import OrderedDict


class GraphLRU:
    def __init__(self, capacity: int = 4):
        self.capacity = max(1, int(capacity))
        self._d: OrderedDict = OrderedDict()
        self.evictions = 0

    def get(self, key):
        if key in self._d:
            self._d.move_to_end(key)
            return self._d[key]
        return None

    def put(self, key, value):
        self._d[key] = value
        self._d.move_to_end(key)
        while len(self._d) > self.capacity:
            old_key, old = self._d.popitem(last=False)
            self.evictions += 1
            g = old.get("graph") if isinstance(old, dict) else None
            del old, g  # drop refs now; hipGraphExecDestroy runs on GC

    def __len__(self):
        return len(self._d)

    def __contains__(self, key):
        return key in self._d

    def clear(self):
        self._d.clear()
