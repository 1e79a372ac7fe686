# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/10_integrations/pushgateway_metrics.py"]
# ---
# # Prometheus metrics from ephemeral workers
#
# Workers are ephemeral, so they PUSH metrics to a single-replica gateway
# (a Dict-backed aggregator here) instead of being scraped; a web endpoint
# exposes the aggregate in Prometheus text format.

import os
import time

import modal_examples_amd as modal

app = modal.App("example-pushgateway")

metrics = modal.Dict.from_name("push-metrics", create_if_missing=True)


def push_metric(name: str, value: float, labels: dict = None):
    key = (name, tuple(sorted((labels or {}).items())))
    metrics[key] = {"value": value, "ts": time.time()}


@app.function()
def do_work(i: int) -> int:
    t0 = time.time()
    total = sum(range(i * 1000))
    push_metric("job_duration_seconds", time.time() - t0,
                {"worker": str(os.getpid() % 100), "job": str(i)})
    push_metric("jobs_completed_total", 1, {"job": str(i)})
    return total


@app.function()
@modal.fastapi_endpoint(method="GET", label="metrics")
def metrics_endpoint() -> str:
    return render_prometheus()


def render_prometheus() -> str:
    lines = []
    for (name, labels), rec in metrics.items():
        lbl = ",".join(f'{k}="{v}"' for k, v in labels)
        lines.append(f"{name}{{{lbl}}} {rec['value']}")
    return "\n".join(sorted(lines))


@app.local_entrypoint()
def main():
    metrics.clear()
    list(do_work.map(range(5)))
    text = render_prometheus()
    print(text)
    assert "jobs_completed_total" in text
    metrics.clear()
