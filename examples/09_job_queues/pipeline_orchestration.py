# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/09_job_queues/pipeline_orchestration.py"]
# ---
# # Orchestrate a multi-step pipeline with Functions
#
# Every stage is a Function that hands off to its successor by name
# (`Function.from_name(...).spawn(...)`) — no external orchestrator, and each
# stage can carry its own resources (a GPU stage between CPU stages).  Stage
# outputs are content-addressed artifacts on a Volume, so reruns skip work
# already done; a shared Dict records the trace.
# Reference shape: 09_job_queues/pipeline_orchestration.py.

import hashlib
import json
import time

import modal_examples_amd as modal

APP_NAME = "example-pipeline"
app = modal.App(APP_NAME)

state = modal.Dict.from_name(f"{APP_NAME}-state", create_if_missing=True)
data = modal.Volume.from_name(f"{APP_NAME}-data", create_if_missing=True)


def _key(stage: str, payload) -> str:
    return f"{stage}-{hashlib.sha256(json.dumps(payload).encode()).hexdigest()[:12]}"


def _trace(run_id: str, stage: str, cached: bool):
    log = state.get(run_id) or []
    log.append({"stage": stage, "cached": cached, "t": time.time()})
    state.put(run_id, log)


def _artifact(key: str, compute):
    """Content-addressed stage output on the Volume; compute() only on miss."""
    path = data.path / f"{key}.json"
    if path.exists():
        return json.loads(path.read_text()), True
    out = compute()
    path.write_text(json.dumps(out))
    data.commit()
    return out, False


@app.function()
def make_range(run_id: str, n: int):
    out, cached = _artifact(_key("range", n), lambda: list(range(n)))
    _trace(run_id, "make_range", cached)
    modal.Function.from_name(APP_NAME, "square").spawn(run_id, n)


@app.function(gpu="mi355x")  # the "heavy" middle stage gets its own resources
def square(run_id: str, n: int):
    def compute():
        import torch

        device = "cuda" if torch.cuda.is_available() else "cpu"
        xs = json.loads((data.path / f"{_key('range', n)}.json").read_text())
        t = torch.tensor(xs, dtype=torch.float32, device=device)
        return (t * t).tolist()

    out, cached = _artifact(_key("square", n), compute)
    _trace(run_id, "square", cached)
    modal.Function.from_name(APP_NAME, "reduce_sum").spawn(run_id, n)


@app.function()
def reduce_sum(run_id: str, n: int):
    def compute():
        sq = json.loads((data.path / f"{_key('square', n)}.json").read_text())
        return sum(sq)

    out, cached = _artifact(_key("sum", n), compute)
    _trace(run_id, "reduce_sum", cached)
    state.put(f"{run_id}-result", out)


def _run_once(n: int, tag: str) -> list:
    import uuid

    run_id = f"run-{uuid.uuid4().hex[:8]}"
    state.put(run_id, [])
    make_range.spawn(run_id, n)
    deadline = time.time() + 60
    while state.get(f"{run_id}-result") is None:
        assert time.time() < deadline, "pipeline did not finish"
        time.sleep(0.1)
    result = state.get(f"{run_id}-result")
    trace = state.get(run_id)
    print(f"{tag}: result={result}  trace="
          + " → ".join(f"{t['stage']}{'(cached)' if t['cached'] else ''}"
                       for t in trace))
    return [result, trace]


@app.local_entrypoint()
def main(n: int = 10):
    # fresh artifacts for a deterministic demo
    for f in list(data.listdir("/")):
        data.remove_file(f)
    r1, t1 = _run_once(n, "first run")
    assert r1 == sum(i * i for i in range(n))
    assert not any(s["cached"] for s in t1)
    r2, t2 = _run_once(n, "rerun    ")
    assert r2 == r1
    assert all(s["cached"] for s in t2), "rerun should hit every stage cache"
    print("pipeline orchestration OK (handoff by name, content-keyed cache)")
