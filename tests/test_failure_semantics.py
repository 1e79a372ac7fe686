"""Retries, timeouts, single-use containers, crash recovery.

Reference behavior spec: 06_gpu_and_ml/long-training.py:108-137 (Retries +
timeout interruption + resume), doc_ocr_jobs.py:89 (retries=3),
gpu_fallbacks.py:22 (single_use_containers)."""
import os
import time

import pytest

import modal_examples_amd as modal

app = modal.App("test-fail")

_attempt_dict_name = "test-fail-attempts"


@app.function(retries=modal.Retries(max_retries=3, initial_delay=0.05, backoff_coefficient=1.0))
def flaky(key):
    d = modal.Dict.from_name(_attempt_dict_name)
    n = d.get(key, 0) + 1
    d[key] = n
    if n < 3:
        raise RuntimeError(f"transient {n}")
    return n


@app.function(timeout=1.5, retries=modal.Retries(max_retries=0, initial_delay=0.0))
def sleeper(t):
    time.sleep(t)
    return "done"


@app.function(timeout=2.0, retries=modal.Retries(max_retries=2, initial_delay=0.05))
def checkpointed(key):
    """Simulates long-training.py: times out, retried, resumes from 'checkpoint'."""
    d = modal.Dict.from_name(_attempt_dict_name)
    step = d.get(key, 0)
    if step < 1:
        d[key] = step + 1
        time.sleep(30)  # will be killed by timeout
    return f"resumed-from-{d.get(key)}"


@app.function()
def crasher():
    os._exit(17)  # hard worker death


@app.function(single_use_containers=True)
def single_use():
    return os.getpid()


def test_retries_until_success():
    modal.Dict.from_name(_attempt_dict_name).clear()
    assert flaky.remote("k1") == 3


def test_retries_exhausted():
    modal.Dict.from_name(_attempt_dict_name).clear()

    @app.function(retries=modal.Retries(max_retries=1, initial_delay=0.05))
    def always_fails():
        raise ValueError("nope")

    with pytest.raises(ValueError):
        always_fails.remote()


def test_timeout_raises():
    with pytest.raises(modal.FunctionTimeoutError):
        sleeper.remote(30)


def test_timeout_fast_function_ok():
    assert sleeper.remote(0.01) == "done"


def test_timeout_then_retry_resumes():
    modal.Dict.from_name(_attempt_dict_name).clear()
    assert checkpointed.remote("ck") == "resumed-from-1"


def test_worker_crash_surfaces_error():
    from modal_examples_amd.exception import ExecutionError

    with pytest.raises(ExecutionError):
        crasher.remote()


def test_single_use_containers_fresh_pid():
    pids = {single_use.remote() for _ in range(3)}
    assert len(pids) == 3, "single_use_containers must not reuse workers"


@app.function(retries=modal.Retries(max_retries=2, initial_delay=0.1,
                                    backoff_coefficient=1.0))
def preemptible(key):
    """First attempt dies by SIGKILL (preemption analog); retry must land on
    a FRESH worker and succeed (kill-worker fault injection)."""
    import signal

    d = modal.Dict.from_name(_attempt_dict_name)
    if d.put_if_absent(key, os.getpid()):
        os.kill(os.getpid(), signal.SIGKILL)
    return os.getpid()


def test_worker_sigkill_midcall_retries_on_fresh_worker():
    d = modal.Dict.from_name(_attempt_dict_name, create_if_missing=True)
    key = f"preempt-{time.time()}"
    pid = preemptible.remote(key)
    assert isinstance(pid, int)
    assert pid != d.get(key), "retry must run in a new worker process"


def test_spawn_result_retention_gc():
    """Durable spawn results expire after the 7-day retention window: get()
    then raises OutputExpiredError; ancient tombstones are dropped
    (amazon_embeddings.py:17-18 durability contract)."""
    import time

    import pytest as _pytest

    import modal_examples_amd as modal
    from modal_examples_amd.exception import OutputExpiredError
    from modal_examples_amd.runtime import store

    store.put_result("fc-gc-fresh", True, 41)
    store.put_result("fc-gc-old", True, 42)
    store.put_result("fc-gc-ancient", True, 43)
    conn = store._DB.get()
    with conn:
        conn.execute("UPDATE results SET ts=? WHERE call_id='fc-gc-old'",
                     (time.time() - 8 * 86400,))
        conn.execute("UPDATE results SET ts=? WHERE call_id='fc-gc-ancient'",
                     (time.time() - 40 * 86400,))
    touched = store.gc_results()
    assert touched >= 2
    found, ok, val, _ = store.get_result("fc-gc-fresh")
    assert found and ok and val == 41
    fc = modal.FunctionCall.from_id("fc-gc-old")
    with _pytest.raises(OutputExpiredError):
        fc.get(timeout=1)
    # ancient: second sweep after tombstoning drops the row entirely
    with conn:
        conn.execute("UPDATE results SET ts=? WHERE call_id='fc-gc-ancient'",
                     (time.time() - 40 * 86400,))
    store.gc_results()
    found, *_ = store.get_result("fc-gc-ancient")
    assert not found
