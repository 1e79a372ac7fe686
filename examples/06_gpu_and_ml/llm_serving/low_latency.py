# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/low_latency.py"]
# ---
# # Latency-tuned LLM serving with a FAST_BOOT trade-off flag
#
# The vllm_low_latency / FAST_BOOT recipe (vllm_inference.py:93-101,
# vllm_low_latency.py:20-22): one knob trades boot time against steady-state
# speed.  Here `fast_boot=True` skips hipGraph capture (serves eagerly,
# boots in ~a second); `fast_boot=False` captures the full-width decode
# graph up front (slower boot, fastest tokens).  The `modal.parameter`
# makes each setting its own autoscaling pool; the entrypoint measures both
# boot-to-first-token and per-token latency for each.  `spec_tokens` turns
# on ngram speculative decoding (vllm_inference.py:195-202's
# speculative-config role): prompt-lookup drafts verified in ONE expanded
# paged-decode forward — greedy outputs stay token-identical.

import time

import modal_examples_amd as modal

app = modal.App("example-llm-low-latency")


@app.cls(gpu="mi355x", timeout=600, scaledown_window=60)
class LLM:
    fast_boot: bool = modal.parameter(default=True)
    spec_tokens: int = modal.parameter(default=0)  # ngram speculation depth

    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        t0 = time.time()
        gpu = torch.cuda.is_available()
        cfg = LlamaConfig.llama3_8b() if gpu else LlamaConfig.small()
        eng = LlamaEngine(cfg, device="cuda" if gpu else "cpu",
                          dtype=torch.bfloat16 if gpu else torch.float32,
                          use_graph=gpu and not self.fast_boot,
                          kv_blocks=None if gpu else 128,
                          spec_tokens=self.spec_tokens)
        if not self.fast_boot:
            eng.warmup()  # capture the decode graph now, not on request 1
        self.server = LLMServer(eng, model_name="low-latency")
        self.boot_s = time.time() - t0

    @modal.method()
    def timed_chat(self, prompt: str, max_tokens: int = 24) -> dict:
        t0 = time.monotonic()
        text = self.server.generate(prompt, max_tokens=max_tokens)
        dt = time.monotonic() - t0
        return {"boot_s": round(self.boot_s, 2),
                "ms_per_token": round(dt * 1000 / max_tokens, 2),
                "tokens": max_tokens, "text_head": text[:40]}


@app.local_entrypoint()
def main():
    fast = LLM(fast_boot=True).timed_chat.remote("hello")
    slow = LLM(fast_boot=False).timed_chat.remote("hello")
    print(f"fast_boot=True : boot {fast['boot_s']}s, "
          f"{fast['ms_per_token']} ms/token (eager)")
    print(f"fast_boot=False: boot {slow['boot_s']}s, "
          f"{slow['ms_per_token']} ms/token (hipGraph)")
    spec = LLM(fast_boot=True, spec_tokens=4).timed_chat.remote(
        "repeat repeat repeat repeat")
    print(f"spec_tokens=4  : boot {spec['boot_s']}s, "
          f"{spec['ms_per_token']} ms/token (ngram speculation)")
    assert fast["boot_s"] <= slow["boot_s"] + 5.0
