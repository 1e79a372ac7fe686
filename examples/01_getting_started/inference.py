# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/01_getting_started/inference.py"]
# ---
# # First GPU inference (01_getting_started/inference.py role)
#
# One GPU function, one remote call: load a model in `@modal.enter`, run a
# single prompt through it.

import modal_examples_amd as modal

app = modal.App("example-inference")


@app.cls(gpu="mi355x")
class TinyLM:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.gpt.model import GPT, GPTConfig

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        self.model = GPT(GPTConfig(n_layer=2, n_embd=128, n_head=2,
                                   block_size=64)).to(self.device).eval()

    @modal.method()
    def complete(self, prompt: str, n: int = 12) -> str:
        idx = self.torch.tensor([[min(255, ord(c)) for c in prompt]],
                                device=self.device)
        out = self.model.generate(idx, n, temperature=0.8, seed=7)
        return "".join(chr(max(32, min(126, t))) for t in out[0].tolist())


@app.local_entrypoint()
def main(prompt: str = "the MI355X says: "):
    text = TinyLM().complete.remote(prompt)
    assert text.startswith(prompt[:8])
    print(repr(text))
