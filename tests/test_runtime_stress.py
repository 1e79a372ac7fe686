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
