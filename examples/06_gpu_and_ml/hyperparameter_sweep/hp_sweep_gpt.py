# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/hyperparameter_sweep/hp_sweep_gpt.py", "--variants", "4", "--steps", "8"]
# ---
# # Hyperparameter sweep over nanoGPT variants
#
# Train several GPT variants in parallel with `starmap(order_outputs=False)`,
# checkpoint each to the shared Volume, pick the best by validation loss, and
# resume the winner from its checkpoint — then serve it from a parameterized
# class (one autoscale pool per variant).

import json

import modal_examples_amd as modal

app = modal.App("example-hp-sweep-gpt")

runs = modal.Volume.from_name("hp-sweep-runs", create_if_missing=True)


def make_batch(cfg, step, device):
    """Synthetic byte-level LM data (deterministic per step)."""
    import torch

    g = torch.Generator().manual_seed(step)
    x = torch.randint(0, cfg.vocab_size, (8, cfg.block_size), generator=g)
    y = torch.roll(x, -1, dims=1)
    return x.to(device), y.to(device)


@app.function(gpu="mi355x", timeout=1800)
def train_variant(n_layer: int, n_embd: int, steps: int) -> dict:
    import torch

    from modal_examples_amd.models.gpt.model import GPT, GPTConfig
    from modal_examples_amd.train.lora import FusedAdamW

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = GPTConfig(n_layer=n_layer, n_embd=n_embd,
                    n_head=max(1, n_embd // 64), block_size=128)
    torch.manual_seed(0)
    model = GPT(cfg).to(device)
    if device == "cuda":
        model = model.to(torch.bfloat16)
    for p in model.parameters():
        p.data = p.data.float() if device == "cpu" else p.data
    opt = FusedAdamW([p for p in model.parameters()], lr=3e-4)
    name = f"gpt_l{n_layer}_d{n_embd}"
    from modal_examples_amd.observability.board import log_scalar

    for step in range(steps):
        x, y = make_batch(cfg, step, device)
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        opt.zero_grad()
        log_scalar(runs.path / "logs" / name, "train/loss", step, float(loss))
    vx, vy = make_batch(cfg, 10_000, device)
    with torch.no_grad():
        _, val_loss = model(vx, vy)
    ckpt = runs.path / f"{name}.pt"
    torch.save({"cfg": vars(cfg), "state": model.state_dict(),
                "val_loss": float(val_loss), "steps": steps}, ckpt)
    runs.commit()
    out = {"name": name, "val_loss": round(float(val_loss), 4), "steps": steps}
    print("trained", json.dumps(out))
    return out


@app.function(gpu="mi355x", timeout=1800)
def resume_best(name: str, extra_steps: int) -> dict:
    import torch

    from modal_examples_amd.models.gpt.model import GPT, GPTConfig

    runs.reload()
    ck = torch.load(runs.path / f"{name}.pt", map_location="cpu",
                    weights_only=False)
    cfg = GPTConfig(**ck["cfg"])
    device = "cuda" if torch.cuda.is_available() else "cpu"
    model = GPT(cfg).to(device)
    model.load_state_dict(ck["state"])
    from modal_examples_amd.train.lora import FusedAdamW

    opt = FusedAdamW(list(model.parameters()), lr=3e-4)
    for step in range(ck["steps"], ck["steps"] + extra_steps):
        x, y = make_batch(cfg, step, device)
        _, loss = model(x, y)
        loss.backward()
        opt.step()
        opt.zero_grad()
    return {"name": name, "resumed_from": ck["steps"],
            "now_at": ck["steps"] + extra_steps}


@app.function()
@modal.wsgi_app(label="hp-board")
def board():
    """TensorBoard-on-Volume role: scalar dashboard over the sweep's logs,
    volume reloaded before every request (hp_sweep_gpt.py:396-414 contract)."""
    from modal_examples_amd.observability.board import (
        VolumeReloadMiddleware,
        make_board_wsgi,
    )

    return VolumeReloadMiddleware(make_board_wsgi(runs.path / "logs"), runs)


@app.cls(gpu="mi355x")
class GPTServer:
    variant: str = modal.parameter(default="gpt_l2_d128")

    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.gpt.model import GPT, GPTConfig

        runs.reload()
        ck = torch.load(runs.path / f"{self.variant}.pt", map_location="cpu",
                        weights_only=False)
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        self.model = GPT(GPTConfig(**ck["cfg"])).to(self.device)
        self.model.load_state_dict(ck["state"])

    @modal.method()
    def complete(self, prefix_bytes: list, n: int = 16) -> list:
        import torch

        idx = torch.tensor([prefix_bytes], device=self.device)
        out = self.model.generate(idx, n, temperature=0.8, seed=1)
        return out[0].tolist()


@app.local_entrypoint()
def main(variants: int = 4, steps: int = 8):
    grid = [(l, d) for l in (2, 3) for d in (128, 192)][:variants]
    results = list(train_variant.starmap(
        [(l, d, steps) for l, d in grid], order_outputs=False))
    best = min(results, key=lambda r: r["val_loss"])
    print("best variant:", best)
    print("resumed:", resume_best.remote(best["name"], steps))
    srv = GPTServer(variant=best["name"])
    completion = srv.complete.remote([1, 2, 3], 8)
    print("completion:", completion)

    # the TensorBoard-role dashboard serves the sweep's loss curves
    import asyncio

    import httpx

    from modal_examples_amd.web.ingress import build_ingress_app

    async def check_board():
        root = build_ingress_app(app)
        async with httpx.AsyncClient(transport=httpx.ASGITransport(app=root),
                                     base_url="http://t") as c:
            r = await c.get("/hp-board/data")
            data = r.json()
            assert any("train/loss" in tags for tags in data.values()), data.keys()
            html = (await c.get("/hp-board/")).text
            assert "train/loss" in html and "<svg" in html
            return len(data)

    n_runs = asyncio.run(check_board())
    print(f"board serves {n_runs} run(s) of loss curves")
