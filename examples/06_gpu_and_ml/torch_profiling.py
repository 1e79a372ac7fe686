# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/torch_profiling.py"]
# ---
# # Profiling any function
#
# A generic profiler Function: wrap any registered function in torch.profiler
# (CPU+GPU activities), save a Perfetto-compatible trace to a Volume, print
# the op table.  On MI355X, pair with rocprofv3 for per-kernel counters
# (`modal_examples_amd.observability.profiling.rocprof_stats_command`).

import modal_examples_amd as modal

app = modal.App("example-profiling")

traces = modal.Volume.from_name("profiler-traces", create_if_missing=True)


def matmul_workload():
    import torch

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    a = torch.randn(512, 512, device=dev)
    b = torch.randn(512, 512, device=dev)
    (a @ b).sum().item()


@app.function(gpu="mi355x")
def profile_workload() -> str:
    from modal_examples_amd.observability.profiling import profile_call

    path = profile_call(matmul_workload, trace_dir=str(traces.path / "demo"),
                        steps=2, warmup=1)
    traces.commit()
    return path


@app.function()
@modal.wsgi_app(label="trace-board")
def board():
    """TensorBoard-role dashboard over the trace Volume, reloaded per request
    (torch_profiling.py:294-316 contract): lists saved traces + any scalar
    events alongside them."""
    import json
    from pathlib import Path

    from modal_examples_amd.observability.board import (
        VolumeReloadMiddleware,
        make_board_wsgi,
    )

    scalars = make_board_wsgi(traces.path)

    def wsgi(environ, start_response):
        if environ.get("PATH_INFO", "/").rstrip("/").endswith("traces"):
            files = sorted(str(f.relative_to(traces.path))
                           for f in Path(traces.path).glob("**/*.json"))
            start_response("200 OK", [("Content-Type", "application/json")])
            return [json.dumps({"traces": files}).encode()]
        return scalars(environ, start_response)

    return VolumeReloadMiddleware(wsgi, traces)


@app.local_entrypoint()
def main():
    print("trace written to:", profile_workload.remote())

    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    async def check():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            r = (await c.get("/trace-board/traces")).json()
            assert r["traces"], r
            return r["traces"]

    print("board lists traces:", asyncio.run(check()))
