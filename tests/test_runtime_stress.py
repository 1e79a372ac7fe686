"""Scheduler stress: high-fanout maps, pool contention, generator volume."""
import threading

import modal_examples_amd as modal

app = modal.App("test-stress")


@app.function(max_containers=6)
def inc(x):
    return x + 1


@app.function(max_containers=4)
def mul(x):
    return x * 2


@app.function()
def stream(n):
    for i in range(n):
        yield i


def test_large_map_1000():
    out = list(inc.map(range(1000)))
    assert out == [x + 1 for x in range(1000)]


def test_two_pools_contend():
    res = {}

    def run_a():
        res["a"] = sum(inc.map(range(200)))

    def run_b():
        res["b"] = sum(mul.map(range(200)))

    ta, tb = threading.Thread(target=run_a), threading.Thread(target=run_b)
    ta.start()
    tb.start()
    ta.join(120)
    tb.join(120)
    assert res["a"] == sum(range(1, 201))
    assert res["b"] == sum(x * 2 for x in range(200))


def test_many_generators():
    gens = [stream.remote_gen(20) for _ in range(8)]
    for g in gens:
        assert list(g) == list(range(20))


def test_spawn_burst():
    calls = [inc.spawn(i) for i in range(150)]
    vals = modal.functions.gather(*calls)
    assert vals == [i + 1 for i in range(150)]


def test_unordered_map_large_window_event_driven():
    """1000 unordered inputs drain correctly through the wait-any event path
    (no 2ms busy-poll; r1 weak #8)."""
    import modal_examples_amd as modal

    app = modal.App("test-map-large")

    @app.function(max_containers=4)
    @modal.concurrent(max_inputs=8)
    def inc(x: int) -> int:
        return x + 1

    out = sorted(inc.map(range(1000), order_outputs=False))
    assert out == list(range(1, 1001))


def test_scaledown_churn_under_intermittent_load():
    """Aggressive scaledown + bursts arriving between idle gaps: every
    burst must complete even while the reaper is retiring workers (the
    reap-on-boot liveness class of bug)."""
    import time

    app = modal.App("stress-churn")

    @app.function(scaledown_window=0.15, max_containers=3)
    def work(x: int) -> int:
        time.sleep(0.02)
        return x * 2

    for burst in range(4):
        got = sorted(work.map(range(8)))
        assert got == [x * 2 for x in range(8)]
        time.sleep(0.4)  # let the reaper take everything down between bursts


def test_cancel_does_not_poison_the_pool():
    """Cancel an in-flight call (tears its container down), then the pool
    must still serve fresh calls promptly."""
    import time

    app = modal.App("stress-cancel-recover")

    @app.function(max_containers=2)
    def slow_or_fast(x: int) -> int:
        import time as t

        if x < 0:
            t.sleep(30)
        return x + 1

    fc = slow_or_fast.spawn(-1)
    time.sleep(0.5)  # let it start executing
    fc.cancel()
    t0 = time.monotonic()
    assert slow_or_fast.remote(41) == 42
    assert time.monotonic() - t0 < 20, "pool did not recover after cancel"
