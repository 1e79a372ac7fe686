"""Core Function verb semantics — the hello_world tier of the reference
(reference behavior spec: 01_getting_started/hello_world.py:34-70,
generators.py:13-21, 08_advanced/parallel_execution.py:41-48)."""
import time

import pytest

import modal_examples_amd as modal

app = modal.App("test-fn")


@app.function()
def square(x: int) -> int:
    return x * x


@app.function()
def fail_on_odd(x: int) -> int:
    if x % 2:
        raise ValueError(f"odd: {x}")
    return x


@app.function()
def gen_fn(n: int):
    for i in range(n):
        yield i * 10


@app.function()
def add(a: int, b: int) -> int:
    return a + b


@app.function()
def slow_echo(x):
    time.sleep(0.2)
    return x


def test_local():
    assert square.local(7) == 49


def test_call_direct():
    assert square(6) == 36


def test_remote():
    assert square.remote(9) == 81


def test_remote_kwargs():
    assert add.remote(2, b=3) == 5


def test_map_ordered():
    assert list(square.map(range(10))) == [x * x for x in range(10)]


def test_map_unordered():
    out = list(square.map(range(12), order_outputs=False))
    assert sorted(out) == sorted(x * x for x in range(12))


def test_map_multiple_iterators():
    assert list(add.map(range(4), range(4, 8))) == [4, 6, 8, 10]


def test_starmap():
    assert list(add.starmap([(1, 2), (3, 4)])) == [3, 7]


def test_map_exceptions_raise():
    with pytest.raises(ValueError):
        list(fail_on_odd.map(range(4)))


def test_map_return_exceptions():
    out = list(fail_on_odd.map(range(4), return_exceptions=True))
    assert out[0] == 0 and out[2] == 2
    assert isinstance(out[1], ValueError) and isinstance(out[3], ValueError)


def test_map_ignore_exceptions():
    out = list(fail_on_odd.map(range(6), ignore_exceptions=True))
    assert out == [0, 2, 4]


def test_for_each():
    fail_on_odd.for_each(range(0, 6, 2))


def test_spawn_and_get():
    fc = square.spawn(12)
    assert fc.get() == 144
    assert fc.object_id.startswith("fc-")


def test_spawn_gather():
    calls = [square.spawn(i) for i in range(5)]
    assert modal.functions.gather(*calls) == [0, 1, 4, 9, 16]


def test_spawn_exception_propagates():
    fc = fail_on_odd.spawn(3)
    with pytest.raises(ValueError, match="odd: 3"):
        fc.get()


def test_function_call_from_id():
    fc = square.spawn(5)
    assert fc.get() == 25
    time.sleep(0.2)  # allow durable store write
    fc2 = modal.FunctionCall.from_id(fc.object_id)
    assert fc2.get(timeout=20) == 25


def test_get_timeout_zero_raises():
    fc = slow_echo.spawn("hi")
    with pytest.raises(TimeoutError):
        fc.get(timeout=0)
    assert fc.get(timeout=30) == "hi"


def test_remote_gen():
    assert list(gen_fn.remote_gen(4)) == [0, 10, 20, 30]


def test_remote_exception_has_traceback():
    try:
        fail_on_odd.remote(1)
    except ValueError as e:
        assert "odd: 1" in str(e)
        assert getattr(e, "remote_traceback", "")
    else:
        pytest.fail("no exception")


def test_aio_remote():
    import asyncio

    async def go():
        return await square.remote.aio(4)

    assert asyncio.run(go()) == 16


def test_aio_map():
    import asyncio

    async def go():
        out = []
        async for r in square.map.aio(range(5)):
            out.append(r)
        return out

    assert asyncio.run(go()) == [0, 1, 4, 9, 16]


def test_function_from_name():
    assert modal.Function.from_name("test-fn", "square").remote(3) == 9
    import pytest as _pytest

    with _pytest.raises(modal.NotFoundError):
        modal.Function.from_name("test-fn", "nope")


def test_app_include():
    other = modal.App("test-fn-other")

    @other.function()
    def triple(x):
        return 3 * x

    merged = modal.App("test-fn-merged").include(other)
    assert merged.functions["triple"].remote(4) == 12


def test_sticky_key_routes_to_same_worker():
    """Calls carrying the same sticky_key are all served by one worker even
    when the pool is allowed to scale; the binding survives across calls."""
    import os as _os

    app2 = modal.App("test-sticky-pool")

    @app2.function(max_containers=4)
    def worker_pid(i: int = 0) -> int:
        return _os.getpid()

    same = {worker_pid._submit((i,), {}, sticky_key="sess-A").wait()
            for i in range(5)}
    assert len(same) == 1
    a = worker_pid._submit((0,), {}, sticky_key="k1").wait()
    b = worker_pid._submit((1,), {}, sticky_key="k1").wait()
    assert a == b


def test_function_from_name_handoff_from_worker():
    """A worker resolves a peer function by name and hands off to it: the
    store-backed stub enqueues, the client dispatcher executes, the durable
    result comes back to the worker (pipeline_orchestration pattern)."""
    app3 = modal.App("test-handoff-app")

    @app3.function()
    def stage_two(x: int) -> int:
        return x * 10

    @app3.function()
    def stage_one(x: int) -> int:
        fc = modal.Function.from_name("test-handoff-app", "stage_two").spawn(x + 1)
        return fc.get(timeout=30)

    assert stage_one.remote(4) == 50


def test_deployed_app_invoked_from_other_process(tmp_path):
    """app.deploy() in one process lets ANOTHER process call its functions by
    name: from_name returns the store-backed stub and the deployed process's
    dispatcher serves the call (deployed-app invocation pattern)."""
    import subprocess
    import sys
    import textwrap
    import time as _time
    from pathlib import Path

    mod = textwrap.dedent("""
        import time
        import modal_examples_amd as modal
        app = modal.App("test-deployed-demo")

        @app.function()
        def tripled(x: int) -> int:
            return x * 3

        if __name__ == "__main__":
            app.deploy()
            print("DEPLOYED", flush=True)
            time.sleep(60)
    """)
    path = tmp_path / "deployed_mod.py"
    path.write_text(mod)
    import os as _os

    repo = str(Path(__file__).resolve().parent.parent)
    env = dict(_os.environ)
    env["PYTHONPATH"] = repo + _os.pathsep + env.get("PYTHONPATH", "")
    server = subprocess.Popen([sys.executable, str(path)], cwd=repo, env=env,
                              stdout=subprocess.PIPE, text=True)
    try:
        assert "DEPLOYED" in server.stdout.readline()
        f = modal.Function.from_name("test-deployed-demo", "tripled")
        assert f.spawn(14).get(timeout=45) == 42
    finally:
        server.terminate()
        server.wait(timeout=10)


def test_function_and_cls_objects_are_picklable():
    """User code that references a Function/Cls from inside another function
    must serialize: runtime state (locks, pools, verbs) is dropped and
    rebuilt on unpickle (caught by the load_test example)."""
    import cloudpickle

    app4 = modal.App("test-pickle-fn")

    @app4.function()
    def helper(x: int) -> int:
        return x + 1

    @app4.cls()
    class Svc:
        @modal.method()
        def m(self) -> int:
            return 7

    blob = cloudpickle.dumps({"fn": helper, "cls": Svc, "obj": Svc()})
    back = cloudpickle.loads(blob)
    # the revived Function still works (rebuilds its pool in this process)
    assert back["fn"].remote(41) == 42
    assert back["obj"].m.remote() == 7


def test_is_local_true_in_client_false_in_worker():
    app5 = modal.App("test-is-local")

    @app5.function()
    def where() -> bool:
        return modal.is_local()

    assert modal.is_local() is True
    assert where.remote() is False
    assert where.local() is True  # .local runs in the client


def test_current_function_call_id_inside_worker():
    app6 = modal.App("test-call-id")

    @app6.function()
    def my_id() -> str:
        return modal.current_function_call_id()

    assert modal.current_function_call_id() is None
    call = my_id.spawn()
    assert call.get(timeout=30) == call.object_id


def test_allow_concurrent_inputs_legacy_kwarg():
    """@app.function(allow_concurrent_inputs=N): legacy spelling of
    @modal.concurrent — inputs overlap inside one worker."""
    import time as _time

    app7 = modal.App("test-legacy-concurrent")

    @app7.function(allow_concurrent_inputs=4)
    def slow(i: int) -> int:
        _time.sleep(0.5)
        return i

    # untimed warm-up call so worker process spawn/import stays out of the
    # timed region (loaded CI boxes take seconds to fork+import — was a flake)
    assert slow.remote(99) == 99
    t0 = _time.monotonic()
    out = list(slow.map(range(4)))
    dt = _time.monotonic() - t0
    assert sorted(out) == [0, 1, 2, 3]
    # serial would be >= 2.0 s
    assert dt < 1.7, f"inputs did not overlap: {dt:.2f}s"


def test_named_handoff_parallel_burst():
    """Dispatcher under load: many workers hand off by name simultaneously;
    every durable result resolves."""
    app8 = modal.App("test-handoff-burst")

    @app8.function()
    def double(x: int) -> int:
        return x * 2

    @app8.function()
    def relay(x: int) -> int:
        fc = modal.Function.from_name("test-handoff-burst", "double").spawn(x)
        return fc.get(timeout=60)

    out = sorted(relay.map(range(10)))
    assert out == [2 * x for x in range(10)]


def test_function_call_cancel_pending_and_running():
    """FunctionCall.cancel(): a queued input is dropped, an executing input's
    container is torn down; get() raises FunctionCancelledError
    (08_advanced/poll_delayed_result.py cancellation semantics)."""
    import time as _time

    from modal_examples_amd.exception import FunctionCancelledError

    appc = modal.App("test-cancel")

    @appc.function(max_containers=1)
    def sleeper(sec: float) -> str:
        _time.sleep(sec)
        return "done"

    # fill the single container, then queue a second call behind it
    running = sleeper.spawn(8.0)
    _time.sleep(1.0)  # let it dispatch
    queued = sleeper.spawn(8.0)
    queued.cancel()
    with pytest.raises(FunctionCancelledError):
        queued.get(timeout=5)
    running.cancel()
    with pytest.raises(FunctionCancelledError):
        running.get(timeout=5)


def test_function_call_cancel_after_done_keeps_result():
    appc2 = modal.App("test-cancel-done")

    @appc2.function()
    def quick() -> int:
        return 7

    call = quick.spawn()
    assert call.get(timeout=30) == 7
    call.cancel()  # no-op after completion
    assert call.get(timeout=5) == 7


def test_spawn_aio_and_gather_aio():
    """`.spawn.aio` + async gather (reference async patterns, 08_advanced)."""
    import asyncio

    appa = modal.App("test-spawn-aio")

    @appa.function()
    def plus1(x: int) -> int:
        return x + 1

    async def go():
        fcs = await asyncio.gather(*(plus1.spawn.aio(i) for i in range(4)))
        return await asyncio.gather(*(fc.get_aio(30) for fc in fcs))

    assert asyncio.run(go()) == [1, 2, 3, 4]
