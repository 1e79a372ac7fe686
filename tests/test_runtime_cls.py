"""Cls lifecycle: @enter/@exit/@method/parameter/batched/concurrent.

Reference behavior spec: 06_gpu_and_ml/stable_diffusion/text_to_image.py:92-137
(enter + method), flux.py:126-128 (parameter), dynamic_batching.py:29-57
(batched on function and method), sglang_snapshot.py:260 (concurrent)."""
import time

import pytest

import modal_examples_amd as modal

app = modal.App("test-cls")


@app.cls()
class Model:
    prefix: str = modal.parameter(default=">>")

    @modal.enter()
    def setup(self):
        self.state = "loaded"

    @modal.method()
    def run(self, x: str) -> str:
        return f"{self.prefix}{self.state}:{x}"

    @modal.method()
    def boom(self):
        raise RuntimeError("kaboom")

    @modal.exit()
    def teardown(self):
        pass


@app.cls()
class SnapModel:
    @modal.enter(snap=True)
    def load_weights(self):
        self.order = ["snap"]

    @modal.enter(snap=False)
    def finalize(self):
        self.order.append("wake")

    @modal.method()
    def get_order(self):
        return self.order


@app.cls()
class Batched:
    @modal.enter()
    def setup(self):
        self.calls = 0

    @modal.batched(max_batch_size=4, wait_ms=150)
    def mult(self, xs):
        self.calls += 1
        return [x * 2 for x in xs]


@app.function()
@modal.batched(max_batch_size=8, wait_ms=150)
def batch_sq(xs):
    return [x * x for x in xs]


@modal.concurrent(max_inputs=4)
@app.function()
def conc_sleep(x):
    time.sleep(0.6)
    return x


def test_enter_runs_before_method():
    m = Model()
    assert m.run.remote("a") == ">>loaded:a"


def test_parameter_binding():
    m = Model(prefix="##")
    assert m.run.remote("b") == "##loaded:b"


def test_parameter_pools_are_distinct():
    assert Model(prefix="A").run.remote("x") == "Aloaded:x"
    assert Model(prefix="B").run.remote("x") == "Bloaded:x"


def test_snap_enter_ordering():
    assert SnapModel().get_order.remote() == ["snap", "wake"]


def test_method_exception():
    with pytest.raises(RuntimeError, match="kaboom"):
        Model().boom.remote()


def test_method_local():
    m = Model(prefix="&")
    assert m.run.local("z") == "&loaded:z"


def test_cls_method_map():
    m = Model()
    assert list(m.run.map(["1", "2"])) == [">>loaded:1", ">>loaded:2"]


def test_batched_function_batches():
    import threading

    results = {}

    def call(i):
        results[i] = batch_sq.remote(i)

    threads = [threading.Thread(target=call, args=(i,)) for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert results == {i: i * i for i in range(6)}


def test_batched_method():
    b = Batched()
    out = list(b.mult.map(range(6)))
    assert out == [0, 2, 4, 6, 8, 10]


def test_concurrent_overlaps():
    t0 = time.monotonic()
    out = list(conc_sleep.map(range(4)))
    dt = time.monotonic() - t0
    assert out == [0, 1, 2, 3]
    # 4 concurrent 0.6s sleeps on one worker: ~0.6-1.5s; serial is >= 2.4s
    assert dt < 2.0, f"no concurrency: {dt:.2f}s"


def test_with_options():
    M2 = Model.with_options(max_containers=2)
    assert M2().run.remote("w") == ">>loaded:w"


def test_debug_exec_shell_into_running_worker():
    """`modal shell app.py::Cls` path: __debug_exec__ runs code inside the
    live worker with access to the service instance (`obj`)."""
    app2 = modal.App("test-shell-attach")

    @app2.cls()
    class Svc:
        @modal.enter()
        def boot(self):
            self.loaded = "model-v1"

        @modal.method()
        def f(self) -> int:
            return 1

    svc = Svc()
    assert svc.f.remote() == 1  # worker is now running with state loaded
    pool = svc.f.obj._get_pool()
    out = pool.submit("__debug_exec__", ("obj.loaded",), {},
                      sticky_key="__shell__").wait()
    assert out.strip() == "'model-v1'"
    # statements work too, and mutate the LIVE instance
    pool.submit("__debug_exec__", ("obj.loaded = 'patched'",), {},
                sticky_key="__shell__").wait()
    assert pool.submit("__debug_exec__", ("print(obj.loaded)",), {},
                       sticky_key="__shell__").wait().strip() == "patched"


def test_exit_hook_runs_on_scaledown():
    """@modal.exit fires when the worker is reaped (container shutdown)."""
    appx = modal.App("test-exit-hook")
    d = modal.Dict.from_name("exit-hook-proof", create_if_missing=True)
    d.clear()

    @appx.cls()
    class Svc:
        @modal.method()
        def ping(self) -> int:
            return 1

        @modal.exit()
        def bye(self):
            modal.Dict.from_name("exit-hook-proof")["ran"] = True

    s = Svc()
    assert s.ping.remote() == 1
    pool = s.ping.obj._get_pool()
    pool.reap_idle(force=True)
    deadline = time.monotonic() + 25
    while not d.get("ran") and time.monotonic() < deadline:
        time.sleep(0.05)
    assert d.get("ran") is True, "exit hook did not run at worker shutdown"
    modal.Dict.delete("exit-hook-proof")


def test_subsecond_scaledown_does_not_reap_booting_worker():
    """Liveness: scaledown_window shorter than @enter must not kill the
    booting container (reap-on-boot + respawn looped forever; idle time
    counts from READY, and queued inputs pin the pool)."""
    app = modal.App("test-slow-enter-scaledown")

    @app.cls(scaledown_window=0.2)
    class Slow:
        @modal.enter()
        def boot(self):
            time.sleep(1.5)  # >> scaledown_window
            self.ok = 41

        @modal.method()
        def get(self):
            return self.ok + 1

    assert Slow().get.remote() == 42
