# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/moe_serving.py"]
# ---
# # Serving a mixture-of-experts model (deepseek_v4 / gpt_oss role)
#
# The reference's biggest serving examples run MoE models (DeepSeek-V4 on
# B200:8 at llm-serving/deepseek_v4.py:179-198, gpt-oss, Mixtral-class
# under vLLM).  Same shape here: a routed top-2 MoE Llama variant
# (`LlamaConfig(n_experts=...)`, `models/llama/model.py:MoEFFN`) rides the
# SAME continuous-batching engine — paged KV attention is expert-agnostic,
# each expert's SwiGLU is a dense hipBLASLt GEMM over its routed tokens —
# so prefix caching / speculation / fp8-KV all compose.

import modal_examples_amd as modal

app = modal.App("example-moe-serving")


@app.cls(gpu="mi355x", scaledown_window=60)
class MoELLM:
    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        gpu = torch.cuda.is_available()
        # 8-expert top-2 at llama-8B geometry on GPU; tiny config on CPU
        cfg = (LlamaConfig(n_experts=8, ffn_dim=4096) if gpu
               else LlamaConfig.moe_small())
        eng = LlamaEngine(cfg, device="cuda" if gpu else "cpu",
                          dtype=torch.bfloat16 if gpu else torch.float32,
                          use_graph=False,  # expert routing is data-dependent
                          kv_blocks=None if gpu else 128, prefix_cache=True)
        self.cfg = cfg
        self.engine = eng
        self.server = LLMServer(eng, model_name="moe-8x")

    @modal.method()
    def generate(self, prompt: str, max_tokens: int = 12) -> dict:
        text = self.server.generate(prompt, max_tokens=max_tokens)
        total = sum(p.numel() for p in self.engine.model.parameters())
        # active params per token: dense layers + top-k of the expert pool
        moe = self.engine.model.blocks[0].moe
        expert_p = (moe.gate_up[0].numel() + moe.down[0].numel())
        active = total - self.cfg.n_layers * (
            self.cfg.n_experts - moe.top_k) * expert_p
        return {"text_words": len(text.split()), "total_params": total,
                "active_params_per_tok": active}

    @modal.exit()
    def stop(self):
        self.server.shutdown()


@app.local_entrypoint()
def main():
    r = MoELLM().generate.remote("route me through the experts")
    assert r["text_words"] >= 1
    frac = r["active_params_per_tok"] / r["total_params"]
    print(f"MoE serving: {r['total_params'] / 1e6:.1f}M total params, "
          f"{frac:.0%} active per token (top-2 of the expert pool)")
