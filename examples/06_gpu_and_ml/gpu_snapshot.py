# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/gpu_snapshot.py"]
# ---
# # GPU memory snapshots
#
# The cold-start lifecycle: `@modal.enter(snap=True)` does the expensive
# weight load ONCE and captures a pinned-host snapshot; later instances
# restore via batched `hipMemcpyAsync` (the ~10× cold-boot pattern) and run
# only the cheap `@modal.enter(snap=False)` wake hook.

import time

import modal_examples_amd as modal

app = modal.App("example-gpu-snapshot")


@app.cls(gpu="mi355x", enable_memory_snapshot=True,
         experimental_options={"enable_gpu_snapshot": True})
class Model:
    @modal.enter(snap=True)
    def load(self):
        import torch

        t0 = time.time()
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        # stand-in for an expensive from_pretrained: 1 GB of weights
        n = 8 if self.device == "cuda" else 1
        self.weights = {
            f"w{i}": torch.randn(64, 1024, 512, device=self.device)
            for i in range(n)
        }
        if self.device == "cuda":
            from modal_examples_amd.gpu.snapshot import WeightSnapshot

            self.snap = WeightSnapshot.capture(self.weights)
        print(f"loaded+captured in {time.time() - t0:.2f}s")

    @modal.enter(snap=False)
    def wake(self):
        self.ready_at = time.time()

    @modal.method()
    def restore_benchmark(self) -> dict:
        if self.device != "cuda":
            return {"device": "cpu", "note": "restore path needs a GPU"}
        for w in self.weights.values():
            w.zero_()
        t0 = time.time()
        self.snap.restore(self.weights)
        dt = time.time() - t0
        gb = self.snap.total_bytes / 1e9
        return {"restored_gb": round(gb, 2), "seconds": round(dt, 3),
                "gb_per_s": round(gb / dt, 1)}


@app.local_entrypoint()
def main():
    m = Model()
    print(m.restore_benchmark.remote())
