"""Tracing + metrics subsystems."""
import modal_examples_amd as modal
from modal_examples_amd.observability import metrics, tracing


def test_span_nesting_and_file():
    tracing.clear()
    with tracing.span("outer", kind="test"):
        with tracing.span("inner"):
            pass
    spans = tracing.read_traces()
    names = [s["name"] for s in spans]
    assert "inner" in names and "outer" in names
    inner = next(s for s in spans if s["name"] == "inner")
    outer = next(s for s in spans if s["name"] == "outer")
    assert inner["parent_id"] == outer["span_id"]
    assert inner["trace_id"] == outer["trace_id"]


def test_span_records_error():
    tracing.clear()
    try:
        with tracing.span("boom"):
            raise ValueError("x")
    except ValueError:
        pass
    s = tracing.read_traces()[-1]
    assert "ValueError" in s["error"]


def test_runner_emits_call_spans_and_metrics():
    tracing.clear()
    metrics.reset()
    app = modal.App("obs-test")

    @app.function()
    def noop(x):
        return x

    assert noop.remote(1) == 1
    calls = [s for s in tracing.read_traces() if s["name"].startswith("call:")]
    assert calls, "runner did not emit call spans"
    text = metrics.render_prometheus()
    assert "calls_total" in text and "call_duration_s_p50" in text


def test_metrics_histogram_percentiles():
    metrics.reset()
    for v in range(100):
        metrics.observe("lat", v / 100, {"x": "1"})
    text = metrics.render_prometheus()
    assert 'lat_p50{x="1"} 0.5' in text
    assert 'lat_count{x="1"} 100' in text
