# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/llm_serving/openai_compatible_server.py"]
# ---
# # OpenAI-compatible LLM serving (the canonical serving example)
#
# A Llama-3-8B-class engine with continuous batching, paged KV cache and
# hipGraph-captured decode, exposed as an OpenAI-compatible server on a raw
# port (`@modal.web_server`).  The entrypoint is the health-check-then-request
# smoke test the reference runs against its deployments.

import modal_examples_amd as modal

app = modal.App("example-openai-server")

PORT = 8971


@app.cls(gpu="mi355x", timeout=600)
@modal.concurrent(max_inputs=64)
class LLMService:
    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        import os

        gpu = torch.cuda.is_available()
        cfg = LlamaConfig.llama3_8b() if gpu else LlamaConfig.small()
        # serving knobs (the vLLM/sglang flag roles): ngram speculation,
        # chunked prefill, prefix caching — all exact-output-preserving
        eng = LlamaEngine(cfg, device="cuda" if gpu else "cpu",
                          dtype=torch.bfloat16 if gpu else torch.float32,
                          use_graph=gpu,
                          kv_blocks=None if gpu else 128,
                          spec_tokens=int(os.environ.get("SPEC_TOKENS", "0")),
                          chunked_prefill=int(
                              os.environ.get("CHUNKED_PREFILL", "2048")),
                          prefix_cache=os.environ.get(
                              "PREFIX_CACHE", "1") == "1")
        self.server = LLMServer(eng, model_name="llama-3-8b-mi355x")

    @modal.web_server(port=PORT, startup_timeout=300)
    def serve(self):
        from modal_examples_amd.models.llama.server import serve_openai

        self.uvicorn = serve_openai(self.server, port=PORT, block=False)

    @modal.method()
    def chat(self, prompt: str, max_tokens: int = 32) -> str:
        return self.server.generate(prompt, max_tokens=max_tokens)

    @modal.exit()
    def stop(self):
        if hasattr(self, "uvicorn"):
            self.uvicorn.should_exit = True
        self.server.shutdown()


@app.local_entrypoint()
def main(prompt: str = "what is a wavefront"):
    svc = LLMService()
    out = svc.chat.remote(prompt, 16)
    print("completion:", out)
    assert isinstance(out, str) and out
