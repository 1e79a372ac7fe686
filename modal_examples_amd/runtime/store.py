"""Durable local state: FunctionCall results, named Dicts and Queues.

The reference platform persists spawned-call results ("1M inputs, 7 days",
06_gpu_and_ml/embeddings/amazon_embeddings.py:17-18) and offers named Dict/Queue
distributed state (09_job_queues/dicts_and_queues.py:72-95).  Locally both are
sqlite-backed under the state dir so they survive process restarts and are
shared across worker processes.
"""
from __future__ import annotations

import os
import pickle
import sqlite3
import threading
import time
from typing import Any, List, Optional

import cloudpickle

from .. import config
from ..exception import NotFoundError


import contextlib


@contextlib.contextmanager
def _write_txn(conn: sqlite3.Connection):
    """BEGIN IMMEDIATE so a read-then-delete pair is one atomic write
    transaction.  Python's sqlite3 only opens the implicit transaction at the
    first DML statement — a bare ``with conn:`` around SELECT+DELETE lets two
    processes both read the same rows before either deletes (duplicate job
    delivery in the cross-process dispatcher)."""
    conn.execute("BEGIN IMMEDIATE")
    try:
        yield conn
    except BaseException:
        conn.rollback()
        raise
    else:
        conn.commit()


class _DB:
    """One sqlite connection per (process, path), WAL mode for cross-process use."""

    _local = threading.local()

    @classmethod
    def get(cls, path: Optional[str] = None) -> sqlite3.Connection:
        if path is None:
            path = str(config.state_dir() / "state.db")
        key = f"conn_{path}"
        conn = getattr(cls._local, key, None)
        if conn is None:
            conn = sqlite3.connect(path, timeout=30.0, check_same_thread=False)
            conn.execute("PRAGMA journal_mode=WAL")
            conn.execute("PRAGMA synchronous=NORMAL")
            conn.execute(
                "CREATE TABLE IF NOT EXISTS results ("
                "call_id TEXT PRIMARY KEY, status TEXT, payload BLOB, tb TEXT, ts REAL)"
            )
            conn.execute(
                "CREATE TABLE IF NOT EXISTS kv ("
                "ns TEXT, k BLOB, v BLOB, ts REAL, PRIMARY KEY (ns, k))"
            )
            conn.execute(
                "CREATE TABLE IF NOT EXISTS fifo ("
                "rowid INTEGER PRIMARY KEY AUTOINCREMENT, ns TEXT, partition TEXT,"
                " v BLOB, ts REAL)"
            )
            conn.commit()
            setattr(cls._local, key, conn)
        return conn


# ---------------- FunctionCall result store ----------------

def put_result(call_id: str, ok: bool, value: Any, tb: str = ""):
    conn = _DB.get()
    blob = cloudpickle.dumps(value)
    with conn:
        conn.execute(
            "INSERT OR REPLACE INTO results VALUES (?,?,?,?,?)",
            (call_id, "ok" if ok else "error", blob, tb, time.time()),
        )


def get_result(call_id: str):
    """Returns (found, ok, value, tb)."""
    conn = _DB.get()
    row = conn.execute(
        "SELECT status, payload, tb FROM results WHERE call_id=?", (call_id,)
    ).fetchone()
    if row is None:
        return False, False, None, ""
    status, payload, tb = row
    if status == "expired":
        from ..exception import OutputExpiredError

        return True, False, OutputExpiredError(
            f"result of {call_id} expired (older than the retention window)"), tb
    return True, status == "ok", pickle.loads(payload), tb


RESULT_TTL_S = 7 * 86400       # reference durability: "1M inputs, 7 days"
TOMBSTONE_TTL_S = 30 * 86400   # expired markers linger so get() can explain


def gc_results(now: Optional[float] = None) -> int:
    """Spawn-result retention sweep: payloads older than 7 days become
    lightweight 'expired' tombstones (FunctionCall.get then raises
    OutputExpiredError); tombstones older than 30 days are dropped.
    Returns rows touched.  Called periodically by the runtime daemon."""
    conn = _DB.get()
    t = time.time() if now is None else now
    with _write_txn(conn):
        cur = conn.execute(
            "UPDATE results SET status='expired', payload=? "
            "WHERE status != 'expired' AND ts < ?",
            (cloudpickle.dumps(None), t - RESULT_TTL_S))
        n = cur.rowcount
        cur = conn.execute("DELETE FROM results WHERE status='expired' AND ts < ?",
                           (t - TOMBSTONE_TTL_S,))
        n += cur.rowcount
    return n


# ---------------- named Dict ----------------

class DictStore:
    def __init__(self, name: str):
        self.ns = f"dict:{name}"

    def put(self, k, v):
        conn = _DB.get()
        with conn:
            conn.execute(
                "INSERT OR REPLACE INTO kv VALUES (?,?,?,?)",
                (self.ns, cloudpickle.dumps(k), cloudpickle.dumps(v), time.time()),
            )

    def get(self, k, default=None):
        conn = _DB.get()
        row = conn.execute(
            "SELECT v FROM kv WHERE ns=? AND k=?", (self.ns, cloudpickle.dumps(k))
        ).fetchone()
        return default if row is None else pickle.loads(row[0])

    def put_if_absent(self, k, v) -> bool:
        """Atomic claim: True iff the key was newly inserted (INSERT OR IGNORE)."""
        conn = _DB.get()
        with conn:
            cur = conn.execute(
                "INSERT OR IGNORE INTO kv VALUES (?,?,?,?)",
                (self.ns, cloudpickle.dumps(k), cloudpickle.dumps(v), time.time()),
            )
        return cur.rowcount > 0

    def contains(self, k) -> bool:
        conn = _DB.get()
        row = conn.execute(
            "SELECT 1 FROM kv WHERE ns=? AND k=?", (self.ns, cloudpickle.dumps(k))
        ).fetchone()
        return row is not None

    def pop(self, k):
        conn = _DB.get()
        kb = cloudpickle.dumps(k)
        with _write_txn(conn):
            row = conn.execute(
                "SELECT v FROM kv WHERE ns=? AND k=?", (self.ns, kb)
            ).fetchone()
            if row is None:
                raise KeyError(k)
            conn.execute("DELETE FROM kv WHERE ns=? AND k=?", (self.ns, kb))
        return pickle.loads(row[0])

    def delete(self, k):
        conn = _DB.get()
        with conn:
            conn.execute(
                "DELETE FROM kv WHERE ns=? AND k=?", (self.ns, cloudpickle.dumps(k))
            )

    def len(self) -> int:
        conn = _DB.get()
        return conn.execute("SELECT COUNT(*) FROM kv WHERE ns=?", (self.ns,)).fetchone()[0]

    def keys(self):
        conn = _DB.get()
        for (kb,) in conn.execute("SELECT k FROM kv WHERE ns=?", (self.ns,)):
            yield pickle.loads(kb)

    def items(self):
        conn = _DB.get()
        for kb, vb in conn.execute("SELECT k, v FROM kv WHERE ns=?", (self.ns,)):
            yield pickle.loads(kb), pickle.loads(vb)

    def clear(self):
        conn = _DB.get()
        with conn:
            conn.execute("DELETE FROM kv WHERE ns=?", (self.ns,))


# ---------------- named Queue (FIFO with partitions) ----------------

class QueueStore:
    def __init__(self, name: str):
        self.ns = f"queue:{name}"

    def put_many(self, vs: List[Any], partition: Optional[str] = None):
        conn = _DB.get()
        with conn:
            conn.executemany(
                "INSERT INTO fifo (ns, partition, v, ts) VALUES (?,?,?,?)",
                [(self.ns, partition or "", cloudpickle.dumps(v), time.time()) for v in vs],
            )

    def get_many(self, n: int, partition: Optional[str] = None, block=True,
                 timeout: Optional[float] = None) -> List[Any]:
        deadline = None if timeout is None else time.monotonic() + timeout
        part = partition or ""
        while True:
            conn = _DB.get()
            with _write_txn(conn):
                rows = conn.execute(
                    "SELECT rowid, v FROM fifo WHERE ns=? AND partition=?"
                    " ORDER BY rowid LIMIT ?",
                    (self.ns, part, n),
                ).fetchall()
                if rows:
                    conn.executemany(
                        "DELETE FROM fifo WHERE rowid=?", [(r[0],) for r in rows]
                    )
            if rows:
                return [pickle.loads(r[1]) for r in rows]
            if not block:
                return []
            if deadline is not None and time.monotonic() > deadline:
                return []
            time.sleep(0.02)

    def len(self, partition: Optional[str] = None) -> int:
        conn = _DB.get()
        return conn.execute(
            "SELECT COUNT(*) FROM fifo WHERE ns=? AND partition=?",
            (self.ns, partition or ""),
        ).fetchone()[0]

    def clear(self, partition: Optional[str] = None, all: bool = False):
        conn = _DB.get()
        with conn:
            if all:
                conn.execute("DELETE FROM fifo WHERE ns=?", (self.ns,))
            else:
                conn.execute(
                    "DELETE FROM fifo WHERE ns=? AND partition=?",
                    (self.ns, partition or ""),
                )

    def iterate(self, partition: Optional[str] = None, item_poll_timeout: float = 0.0):
        while True:
            got = self.get_many(64, partition=partition, block=item_poll_timeout > 0,
                                timeout=item_poll_timeout)
            if not got:
                return
            yield from got


def delete_named(kind: str, name: str):
    conn = _DB.get()
    with conn:
        if kind == "dict":
            conn.execute("DELETE FROM kv WHERE ns=?", (f"dict:{name}",))
        elif kind == "queue":
            conn.execute("DELETE FROM fifo WHERE ns=?", (f"queue:{name}",))
        else:
            raise NotFoundError(kind)
