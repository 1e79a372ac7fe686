# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/otel_traces.py"]
# ---
# # Tracing: OTLP-style spans from functions
#
# The runner already emits a span per dispatched call; applications nest
# their own with `observability.tracing.span` and trace/span ids flow into
# log lines (the parseable_otel pattern).

import modal_examples_amd as modal
from modal_examples_amd.observability import tracing

app = modal.App("example-otel")


@app.function()
def pipeline_stage(x: int) -> int:
    with tracing.span("transform", input=x):
        y = x * 2
        with tracing.span("validate"):
            assert y % 2 == 0
        print(f"[trace={tracing.current_trace_id()}] transformed {x} -> {y}")
    return y


@app.local_entrypoint()
def main():
    tracing.clear()
    outs = list(pipeline_stage.map(range(4)))
    print("outputs:", outs)
    # spans from this process (runner call-spans) are in the local trace file
    spans = tracing.read_traces()
    calls = [s for s in spans if s["name"].startswith("call:")]
    print(f"runner emitted {len(calls)} call spans; "
          f"last: {calls[-1]['name']} {calls[-1]['duration_ms']}ms")
    assert len(calls) >= 4
