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


@app.local_entrypoint()
def main():
    print("trace written to:", profile_workload.remote())
