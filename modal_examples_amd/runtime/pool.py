"""Per-Function worker pool: autoscaling, dispatch, retries, timeouts.

This is the local analog of Modal's container scheduler (reference layer L2 at
SURVEY.md §1).  Semantics implemented here, with their reference anchors:

- autoscaling knobs ``min_containers``/``max_containers``/``scaledown_window``/
  ``buffer`` (06_gpu_and_ml/speech-to-text/batched_whisper.py:95,
  06_gpu_and_ml/gpu_fallbacks.py:20-23)
- ``single_use_containers`` — one input per container then exit
  (06_gpu_and_ml/long-training.py:129-135)
- retries + timeout-driven interruption (06_gpu_and_ml/long-training.py:108-137)
- dynamic batching window (03_scaling_out/dynamic_batching.py:29)
- GPU-count allocation from the shared 8-GPU pool (``gpu="mi355x:4"`` analog of
  ``gpu="H200:4"``)
"""
from __future__ import annotations

import itertools
import multiprocessing as mp
import sys
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .. import config
from ..exception import (
    ExecutionError,
    FunctionTimeoutError,
    RemoteError,
)
from . import ipc
from .worker import worker_main

_mp = mp.get_context("spawn")
_spawn_exe_lock = threading.Lock()


@dataclass
class RetryPolicy:
    max_retries: int = 0
    initial_delay: float = 1.0
    backoff_coefficient: float = 2.0
    max_delay: float = 60.0

    def delay(self, attempt: int) -> float:
        d = self.initial_delay * (self.backoff_coefficient ** max(0, attempt - 1))
        return min(d, self.max_delay)


class Call:
    """One in-flight invocation (also the client-side future)."""

    __slots__ = (
        "id",
        "method_name",
        "args_blob",
        "is_gen",
        "event",
        "value",
        "exc",
        "gen_q",
        "deadline",
        "timeout",
        "attempt",
        "retries",
        "worker_id",
        "spawned",
        "done",
        "t_submit",
        "sticky_key",
        "pool",
        "on_done",
    )

    def __init__(self, method_name, args_blob, is_gen, timeout, retries: RetryPolicy, spawned=False,
                 sticky_key=None):
        self.id = "fc-" + uuid.uuid4().hex[:16]
        self.method_name = method_name
        self.args_blob = args_blob
        self.is_gen = is_gen
        self.event = threading.Event()
        self.value = None
        self.exc: Optional[BaseException] = None
        self.gen_q: Optional[Any] = _SimpleQ() if is_gen else None
        self.timeout = timeout
        self.deadline = None  # set at dispatch
        self.attempt = 0
        self.retries = retries
        self.worker_id: Optional[int] = None
        self.spawned = spawned
        self.done = False
        self.t_submit = time.monotonic()
        self.sticky_key = sticky_key
        self.pool = None  # set at submit; lets FunctionCall.cancel reach us
        self.on_done = None  # optional completion callback (map wait-any)

    def wait(self, timeout: Optional[float] = None):
        if not self.event.wait(timeout):
            raise TimeoutError(f"result of {self.id} not ready")
        if self.exc is not None:
            raise self.exc
        return self.value


class _SimpleQ:
    """Tiny unbounded thread-safe queue for generator streaming."""

    def __init__(self):
        self._items: List[Any] = []
        self._cv = threading.Condition()
        self._closed = False

    def put(self, item):
        with self._cv:
            self._items.append(item)
            self._cv.notify()

    def close(self):
        with self._cv:
            self._closed = True
            self._cv.notify_all()

    def get(self):
        with self._cv:
            while not self._items and not self._closed:
                self._cv.wait(1.0)
            if self._items:
                return True, self._items.pop(0)
            return False, None


@dataclass
class WorkerHandle:
    id: int
    proc: Any
    task_q: Any
    gpu_devices: tuple
    ready: bool = False
    inflight: set = field(default_factory=set)
    last_active: float = field(default_factory=time.monotonic)
    shutting_down: bool = False
    inputs_served: int = 0


class Pool:
    """Autoscaling pool of worker processes hosting one service spec."""

    _ids = itertools.count(1)

    def __init__(
        self,
        name: str,
        make_spec,  # (gpu_devices: tuple) -> ipc.ServiceSpec
        gpu_count: int = 0,
        min_containers: int = 0,
        max_containers: int = 16,
        buffer_containers: int = 0,
        scaledown_window: Optional[float] = None,
        timeout: Optional[float] = None,
        retries: Optional[RetryPolicy] = None,
        single_use_containers: bool = False,
        max_inputs_per_worker: int = 1,
        target_inputs_per_worker: int = 0,
        device_pool=None,
        python_exe: Optional[str] = None,
    ):
        self.name = name
        self.make_spec = make_spec
        self.gpu_count = gpu_count
        self.min_containers = min_containers
        self.max_containers = max(1, max_containers)
        self.buffer_containers = buffer_containers
        self.scaledown_window = scaledown_window or config.scaledown_window()
        self.timeout = timeout if timeout is not None else config.default_timeout()
        self.retries = retries or RetryPolicy(max_retries=0)
        self.single_use = single_use_containers
        self.max_inputs_per_worker = max(1, max_inputs_per_worker)
        self.target_inputs_per_worker = target_inputs_per_worker or self.max_inputs_per_worker
        self.device_pool = device_pool
        self.python_exe = python_exe  # image-venv interpreter for workers

        self.result_q = _mp.Queue()
        self.workers: Dict[int, WorkerHandle] = {}
        self.calls: Dict[str, Call] = {}
        self.pending: List[Call] = []
        self.sticky: Dict[str, int] = {}  # session key -> worker id
        self.lock = threading.RLock()
        self.closed = False
        self._threads_started = False
        self._worker_seq = itertools.count(0)
        self.on_spawned_result = None  # hook: durable FunctionCall store

    # ---------------- public API ----------------

    def submit(self, method_name, args, kwargs, is_gen=False, spawned=False,
               timeout: Optional[float] = None, sticky_key: Optional[str] = None) -> Call:
        call = Call(
            method_name,
            ipc.dumps((args, kwargs)),
            is_gen,
            timeout if timeout is not None else self.timeout,
            self.retries,
            spawned,
            sticky_key=sticky_key,
        )
        call.pool = self
        with self.lock:
            self._ensure_threads()
            self.calls[call.id] = call
            self.pending.append(call)
            self._pump()
        return call

    def submit_batch(self, items, method_name: str = "") -> List[Call]:
        """items: list of (args, kwargs). Dispatched as ONE worker task
        (optionally against a named Cls method — @modal.batched on methods)."""
        calls = [
            Call(method_name, ipc.dumps((a, k)), False, self.timeout,
                 RetryPolicy())
            for a, k in items
        ]
        with self.lock:
            self._ensure_threads()
            for c in calls:
                self.calls[c.id] = c
            w = self._pick_worker(need_slots=1)
            if w is None:
                w = self._maybe_scale_up()
            target = w
            if target is None:
                # queue batch behind a ready worker later: simplest is block-spin
                target = self._wait_for_worker()
            for c in calls:
                c.worker_id = target.id
                c.deadline = time.monotonic() + c.timeout
                target.inflight.add(c.id)
            target.last_active = time.monotonic()
            target.task_q.put(
                (ipc.T_BATCH, [c.id for c in calls], method_name,
                 [c.args_blob for c in calls])
            )
        return calls

    def cancel(self, call_id: str) -> bool:
        """Cancel a submitted call: drop it from the queue, or — if already
        executing — terminate its worker (mid-flight work cannot be interrupted
        in-process; the container is torn down, matching the platform's input
        cancellation).  Returns True if a live call was cancelled."""
        from ..exception import FunctionCancelledError

        with self.lock:
            call = self.calls.get(call_id)
            if call is None or call.done:
                return False
            if call in self.pending:
                self.pending.remove(call)
                self._resolve(call, exc=FunctionCancelledError(
                    f"{self.name} call {call_id} cancelled"))
                return True
            w = self.workers.get(call.worker_id)
            if w is not None:
                w.inflight.discard(call.id)
                self._stop_worker(w, graceful=False)
                self._release_worker(w)
            self._resolve(call, exc=FunctionCancelledError(
                f"{self.name} call {call_id} cancelled"))
            return True

    def warm(self, n: Optional[int] = None):
        """Pre-start ``min_containers`` (or n) workers."""
        n = n if n is not None else self.min_containers
        with self.lock:
            self._ensure_threads()
            while len(self.workers) < min(n, self.max_containers):
                if self._start_worker() is None:
                    break

    def stats(self):
        with self.lock:
            return {
                "workers": len(self.workers),
                "ready": sum(1 for w in self.workers.values() if w.ready),
                "inflight": sum(len(w.inflight) for w in self.workers.values()),
                "pending": len(self.pending),
            }

    def shutdown(self):
        with self.lock:
            self.closed = True
            for w in list(self.workers.values()):
                self._stop_worker(w, graceful=True)
        deadline = time.monotonic() + 10
        for w in list(self.workers.values()):
            w.proc.join(max(0.1, deadline - time.monotonic()))
            if w.proc.is_alive():
                w.proc.terminate()
            if w.gpu_devices and self.device_pool:
                self.device_pool.release(w.gpu_devices)
        self.workers.clear()

    def reap_idle(self, force: bool = False):
        now = time.monotonic()
        with self.lock:
            if self.pending and not force:
                return  # queued inputs need every worker we have
            for w in list(self.workers.values()):
                if w.inflight or w.shutting_down:
                    continue
                if not w.ready and not force:
                    # still starting (@enter running): idle time is measured
                    # from READY, never from spawn — a sub-second
                    # scaledown_window must not reap a booting container
                    # (liveness: kill-on-boot + respawn would loop forever)
                    continue
                idle = now - w.last_active
                if force or idle > self.scaledown_window:
                    if len(self.workers) <= self.min_containers and not force:
                        continue
                    self._stop_worker(w, graceful=True)

    # ---------------- internals ----------------

    def _ensure_threads(self):
        if self._threads_started:
            return
        self._threads_started = True
        threading.Thread(target=self._collector, daemon=True, name=f"collect-{self.name}").start()
        threading.Thread(target=self._monitor, daemon=True, name=f"monitor-{self.name}").start()

    def _start_worker(self) -> Optional[WorkerHandle]:
        if len(self.workers) >= self.max_containers:
            return None
        devices = ()
        if self.gpu_count > 0:
            if self.device_pool is None:
                raise ExecutionError("GPU requested but no device pool configured")
            devices = self.device_pool.acquire(self.gpu_count)
            if devices is None:
                return None
        wid = next(self._worker_seq)
        spec = self.make_spec(devices)
        task_q = _mp.Queue()
        proc = _mp.Process(
            target=worker_main,
            args=(wid, ipc.dumps(spec), task_q, self.result_q),
            daemon=True,
            name=f"mxa-{self.name}-{wid}",
        )
        if self.python_exe:
            # image-built venv: the worker process execs the venv interpreter
            # (spawn replicates sys.path, so the package resolves; venv
            # site-packages take precedence over system for pip layers)
            with _spawn_exe_lock:
                _mp.set_executable(self.python_exe)
                try:
                    proc.start()
                finally:
                    _mp.set_executable(sys.executable)
        else:
            proc.start()
        w = WorkerHandle(wid, proc, task_q, devices)
        self.workers[wid] = w
        return w

    def _stop_worker(self, w: WorkerHandle, graceful: bool):
        w.shutting_down = True
        try:
            if graceful:
                w.task_q.put((ipc.T_SHUTDOWN,))
            else:
                w.proc.terminate()
        except Exception:
            pass
        # released fully in _reap_dead / shutdown

    def _pick_worker(self, need_slots=1) -> Optional[WorkerHandle]:
        best = None
        for w in self.workers.values():
            if not w.ready or w.shutting_down:
                continue
            if self.single_use and w.inputs_served > 0:
                continue
            if len(w.inflight) + need_slots <= self.max_inputs_per_worker:
                if best is None or len(w.inflight) < len(best.inflight):
                    best = w
        return best

    def _maybe_scale_up(self) -> Optional[WorkerHandle]:
        if len(self.workers) < self.max_containers:
            w = self._start_worker()
            if w is None and self.gpu_count and self.device_pool is not None:
                # ask other pools to give back idle GPU workers, then retry once
                self.device_pool.request_reclaim()
                w = self._start_worker()
            return w
        return None

    def _wait_for_worker(self) -> WorkerHandle:
        deadline = time.monotonic() + config.worker_start_timeout()
        while True:
            w = self._pick_worker()
            if w is not None:
                return w
            self._maybe_scale_up()
            self.lock.release()
            try:
                time.sleep(0.02)
            finally:
                self.lock.acquire()
            if time.monotonic() > deadline and not self.workers:
                raise ExecutionError(f"pool {self.name}: no worker available")

    def _sticky_worker(self, key: str):
        """(worker, wait): the worker bound to ``key``, or a fresh binding.
        wait=True means the bound worker exists but is at capacity — the call
        must queue for THAT worker (sticky semantics: same container serves
        the whole session; reference: 07_web/server_sticky.py:8-18)."""
        wid = self.sticky.get(key)
        w = self.workers.get(wid) if wid is not None else None
        if w is not None and w.ready and not w.shutting_down:
            if len(w.inflight) < self.max_inputs_per_worker:
                return w, False
            return None, True
        return self._pick_worker(), False  # no binding or stale: rebind below

    def _pump(self):
        """Dispatch pending calls onto ready workers; scale up when starved.

        Dispatch is FIFO.  A sticky call whose bound worker is at capacity
        parks the queue until that worker frees a slot (head-of-line by
        design: session ordering must hold, and a sticky pool is normally
        dedicated to its sessions)."""
        while self.pending:
            head = self.pending[0]
            if head.sticky_key is not None:
                w, wait = self._sticky_worker(head.sticky_key)
                if wait:
                    return  # re-pumped when the bound worker resolves a call
            else:
                w = self._pick_worker()
            if w is None:
                # demand-bound scale-up: workers already starting (spawned,
                # not yet READY) will absorb pending inputs — don't burst to
                # max_containers on a cold fan-out
                starting = sum(1 for h in self.workers.values()
                               if not h.ready and not h.shutting_down)
                if starting * self.max_inputs_per_worker >= len(self.pending):
                    return  # capacity is on the way; re-pumped on READY
                started = self._maybe_scale_up()
                if started is None:
                    return  # all busy / can't grow: leave pending
                continue  # newly started worker may not be ready; loop picks ready ones
            # target_inputs autoscale signal (@modal.concurrent): prefer
            # growing the pool over loading a worker past its target
            if (len(w.inflight) >= self.target_inputs_per_worker
                    and len(self.workers) < self.max_containers):
                self._maybe_scale_up()
            call = self.pending.pop(0)
            if call.sticky_key is not None:
                self.sticky[call.sticky_key] = w.id
            self._dispatch(call, w)
        # buffer containers: keep `buffer` idle warm workers beyond demand
        if self.buffer_containers:
            idle = sum(1 for w in self.workers.values() if w.ready and not w.inflight)
            while idle < self.buffer_containers and len(self.workers) < self.max_containers:
                if self._start_worker() is None:
                    break
                idle += 1

    def _dispatch(self, call: Call, w: WorkerHandle):
        call.worker_id = w.id
        call.attempt += 1
        call.deadline = time.monotonic() + call.timeout
        w.inflight.add(call.id)
        w.inputs_served += 1
        w.last_active = time.monotonic()
        w.task_q.put((ipc.T_CALL, call.id, call.method_name, call.args_blob))

    def _resolve(self, call: Call, value=None, exc: Optional[BaseException] = None):
        call.value = value
        call.exc = exc
        call.done = True
        try:
            from ..observability import metrics, tracing

            dur = time.monotonic() - call.t_submit
            metrics.inc("calls_total", 1, {"fn": self.name,
                                           "status": "error" if exc else "ok"})
            metrics.observe("call_duration_s", dur, {"fn": self.name})
            if tracing.enabled():
                tracing._emit({
                    "trace_id": call.id, "span_id": call.id[-16:],
                    "parent_id": None, "name": f"call:{self.name}",
                    "start": time.time() - dur, "end": time.time(),
                    "duration_ms": round(dur * 1000, 3),
                    "attrs": {"attempt": call.attempt,
                              "worker": call.worker_id,
                              "error": repr(exc) if exc else None},
                })
        except Exception:
            pass
        if call.gen_q is not None:
            call.gen_q.close()
        call.event.set()
        if call.on_done is not None:
            try:
                call.on_done(call)
            except Exception:
                pass
        if call.spawned and self.on_spawned_result is not None:
            try:
                self.on_spawned_result(call)
            except Exception:
                pass
        self.calls.pop(call.id, None)

    def _fail_or_retry(self, call: Call, exc: BaseException):
        if call.attempt <= call.retries.max_retries and not call.is_gen:
            delay = call.retries.delay(call.attempt)
            def _requeue():
                with self.lock:
                    if not self.closed and not call.done:
                        self.pending.append(call)
                        self._pump()
            t = threading.Timer(delay, _requeue)
            t.daemon = True
            t.start()
        else:
            self._resolve(call, exc=exc)

    def _collector(self):
        while not self.closed:
            try:
                msg: ipc.WorkerMsg = self.result_q.get(timeout=0.5)
            except Exception:
                continue
            with self.lock:
                w = self.workers.get(msg.worker_id)
                if msg.kind == ipc.READY:
                    if w:
                        w.ready = True
                        w.last_active = time.monotonic()
                        self._pump()
                elif msg.kind == ipc.EXITED:
                    if w:
                        self._release_worker(w)
                elif msg.kind == ipc.ERROR and msg.call_id is None:
                    # enter-hook failure: fail everything pending on this worker
                    if w:
                        self._release_worker(w)
                    err = RemoteError(
                        f"worker for {self.name} failed during startup", msg.text
                    )
                    for call in list(self.pending):
                        self.pending.remove(call)
                        self._resolve(call, exc=err)
                else:
                    call = self.calls.get(msg.call_id or "")
                    if call is None:
                        continue
                    if msg.kind == ipc.YIELD:
                        call.gen_q.put(ipc.loads(msg.payload))
                        if w:
                            w.last_active = time.monotonic()
                        continue
                    # terminal messages free the slot
                    if w and call.id in w.inflight:
                        w.inflight.discard(call.id)
                        w.last_active = time.monotonic()
                        if self.single_use and not w.inflight:
                            self._stop_worker(w, graceful=True)
                    if msg.kind == ipc.RESULT:
                        try:
                            self._resolve(call, value=ipc.loads(msg.payload))
                        except BaseException as e:  # noqa: BLE001
                            self._resolve(call, exc=e)
                    elif msg.kind == ipc.GEN_END:
                        self._resolve(call, value=None)
                    elif msg.kind == ipc.ERROR:
                        try:
                            exc = ipc.loads(msg.payload)
                        except BaseException:  # noqa: BLE001
                            exc = RemoteError("remote error", msg.text)
                        if not isinstance(exc, BaseException):
                            exc = RemoteError(str(exc), msg.text)
                        exc.remote_traceback = msg.text
                        self._fail_or_retry(call, exc)
                    self._pump()

    def _release_worker(self, w: WorkerHandle):
        self.workers.pop(w.id, None)
        if w.gpu_devices and self.device_pool:
            self.device_pool.release(w.gpu_devices)
        # orphaned inflight calls → crash
        for cid in list(w.inflight):
            call = self.calls.get(cid)
            if call:
                self._fail_or_retry(call, ExecutionError(f"worker for {self.name} exited mid-call"))

    def _monitor(self):
        while not self.closed:
            time.sleep(0.25)
            now = time.monotonic()
            with self.lock:
                # dead-process detection
                for w in list(self.workers.values()):
                    if not w.proc.is_alive():
                        self._release_worker(w)
                # timeouts: kill the worker hosting an overdue call
                for call in list(self.calls.values()):
                    if call.deadline is None or call.done:
                        continue
                    if now > call.deadline:
                        w = self.workers.get(call.worker_id)
                        if w:
                            w.inflight.discard(call.id)
                            self._stop_worker(w, graceful=False)
                            self._release_worker(w)
                        self._fail_or_retry(call, FunctionTimeoutError(
                            f"{self.name} exceeded timeout of {call.timeout:.0f}s"))
                # idle scaledown
                self.reap_idle()
                self._pump()
