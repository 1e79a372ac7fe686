# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/stable_diffusion/flux.py", "--prompt", "a lighthouse in a storm"]
# ---
# # Flux-class flow-matching MMDiT (the "canonical perf" diffusion example)
#
# Mirrors the reference's FLUX.1-schnell recipe (stable_diffusion/flux.py:
# 111-273) the MI355X way: where the reference spends up to 20 minutes in
# `torch.compile(max-autotune)` and caches inductor/triton artifacts on
# Volumes, this pipeline captures the whole flow step into a hipGraph in
# seconds — the `capture: bool` `modal.parameter` splits graph-on/graph-off
# into SEPARATE autoscaling pools exactly like the reference's
# `compile: bool = modal.parameter(...)` (flux.py:126-128).

import time

import modal_examples_amd as modal

app = modal.App("example-flux")


@app.cls(gpu="mi355x", scaledown_window=120, enable_memory_snapshot=True,
         experimental_options={"enable_gpu_snapshot": True})
class Flux:
    capture: bool = modal.parameter(default=True)  # hipGraph on/off pools

    @modal.enter(snap=True)
    def load(self):
        import torch

        from modal_examples_amd.models.flux import FluxPipeline, MMDiTConfig

        gpu = torch.cuda.is_available()
        cfg = MMDiTConfig.schnell() if gpu else MMDiTConfig.small()
        self.pipe = FluxPipeline(
            cfg, device="cuda" if gpu else "cpu",
            dtype=torch.bfloat16 if gpu else torch.float32,
            latent_size=128 if gpu else 8, use_graph=self.capture)
        print(f"flux-class MMDiT: {self.pipe.param_count()/1e9:.2f}B params, "
              f"hipGraph={'on' if self.capture and gpu else 'off'}")

    @modal.method()
    def generate(self, prompt: str, steps: int = 4, seed: int = 42) -> dict:
        t0 = time.time()
        imgs = self.pipe.generate([prompt], steps=steps, seed=seed)
        dt = time.time() - t0
        return {"shape": list(imgs.shape), "latency_s": round(dt, 3),
                "captured": bool(self.pipe.use_graph)}


@app.local_entrypoint()
def main(prompt: str = "a lighthouse in a storm", steps: int = 4):
    fast = Flux(capture=True)
    warm = fast.generate.remote(prompt, steps=steps)   # includes capture
    hot = fast.generate.remote(prompt, steps=steps)    # pure replay
    print(f"captured: warm {warm['latency_s']}s -> hot {hot['latency_s']}s, "
          f"image {hot['shape']}")
    eager = Flux(capture=False).generate.remote(prompt, steps=steps)
    print(f"eager pool (separate container): {eager['latency_s']}s")
    assert hot["shape"][1:] == [warm["shape"][1], warm["shape"][2], 3]
