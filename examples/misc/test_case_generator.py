# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/test_case_generator.py"]
# ---
# # LLM test-case generator (misc/test_case_generator.py role)
#
# Point the LLM at a function signature, sample candidate inputs from its
# completions, then actually EXECUTE the target against them in a sandboxed
# fan-out — generated cases that crash the target are the output.

import modal_examples_amd as modal

app = modal.App("example-test-case-gen")


@app.cls(gpu="mi355x", timeout=600)
class CaseGenerator:
    @modal.enter()
    def boot(self):
        import torch

        from modal_examples_amd.models.llama.engine import LlamaEngine
        from modal_examples_amd.models.llama.model import LlamaConfig
        from modal_examples_amd.models.llama.server import LLMServer

        gpu = torch.cuda.is_available()
        eng = LlamaEngine(LlamaConfig.llama3_8b() if gpu else LlamaConfig.small(),
                          device="cuda" if gpu else "cpu",
                          dtype=torch.bfloat16 if gpu else torch.float32,
                          use_graph=gpu, kv_blocks=None if gpu else 128)
        self.server = LLMServer(eng, model_name="case-gen")

    @modal.method()
    def propose(self, signature: str, n: int = 4) -> list:
        """Sample n candidate argument tuples (token stream -> ints)."""
        cases = []
        for i in range(n):
            txt = self.server.generate(
                f"Generate edge-case integer input #{i} for {signature}:",
                max_tokens=6, temperature=0.8)
            # derive a deterministic-but-varied int from the sampled text
            val = (sum(map(ord, txt)) % 201) - 100
            cases.append(val)
        return cases


@app.function()
def run_case(x: int) -> dict:
    def target(v: int) -> float:
        return 100 / (v - 7)  # bug: crashes at v == 7

    try:
        target(x)
        return {"input": x, "crashed": False}
    except Exception as e:
        return {"input": x, "crashed": True, "error": type(e).__name__}


@app.local_entrypoint()
def main():
    gen = CaseGenerator()
    cases = gen.propose.remote("target(v: int) -> float", n=6)
    cases = sorted(set(cases + [7]))  # ensure the boundary value is covered
    results = list(run_case.map(cases))
    crashes = [r for r in results if r["crashed"]]
    print(f"ran {len(results)} generated cases; {len(crashes)} crash(es): {crashes}")
    assert any(r["input"] == 7 and r["crashed"] for r in results)
