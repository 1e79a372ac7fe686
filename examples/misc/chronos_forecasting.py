# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/misc/chronos_forecasting.py"]
# ---
# # Time-series forecasting service (misc/chronos_forecasting.py role)
#
# A Chronos-class forecaster: quantize the series into tokens, run the GPT
# backbone (hipBLASLt GEMMs + gfx950 attention on GPU), sample future tokens,
# map back to values.  The self-test forecasts a clean seasonal signal and
# checks the prediction stays inside the series' range.

import modal_examples_amd as modal

app = modal.App("example-chronos")


@app.cls(gpu="mi355x", timeout=600)
class Forecaster:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.gpt.model import GPT, GPTConfig

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        self.bins = 256
        torch.manual_seed(0)
        cfg = GPTConfig(n_layer=4, n_embd=256, n_head=4, block_size=256,
                        vocab_size=self.bins)
        self.model = GPT(cfg).to(self.device).eval()

    @modal.method()
    def forecast(self, series, horizon: int = 8) -> dict:
        torch = self.torch
        x = torch.as_tensor(series, dtype=torch.float32)
        lo, hi = float(x.min()), float(x.max())
        scale = (hi - lo) or 1.0
        tokens = ((x - lo) / scale * (self.bins - 1)).round().long()
        idx = tokens[None, -128:].to(self.device)
        with torch.no_grad():
            out = self.model.generate(idx, horizon, temperature=0.7, seed=1)
        pred_tokens = out[0, -horizon:].float().cpu()
        preds = (pred_tokens / (self.bins - 1) * scale + lo).tolist()
        return {"forecast": [round(v, 3) for v in preds],
                "context_range": [round(lo, 3), round(hi, 3)]}


@app.local_entrypoint()
def main():
    import math

    series = [math.sin(i / 6.0) * 2 + 5 for i in range(96)]
    out = Forecaster().forecast.remote(series, horizon=8)
    lo, hi = out["context_range"]
    assert len(out["forecast"]) == 8
    assert all(lo - 0.01 <= v <= hi + 0.01 for v in out["forecast"]), out
    print("forecast:", out["forecast"])
