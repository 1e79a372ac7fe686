# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/world_model.py"]
# ---
# # Learning a world model (the world-models example role)
#
# Train a latent dynamics model ("world model") on synthetic environment
# rollouts, then evaluate it by DREAMING: rolling the model forward without
# the environment and measuring drift against reality.  The training loop
# runs the fused-AdamW kernel (K9); rollout collection fans out with
# `.starmap` across containers.

import modal_examples_amd as modal

app = modal.App("example-world-model")

ckpts = modal.Volume.from_name("world-model-ckpts", create_if_missing=True)

STATE, ACT = 8, 2


def env_step(s, a):
    """The 'real' environment: damped spring + action forcing (numpy)."""
    import numpy as np

    x, v, rest = s[..., 0], s[..., 1], s[..., 2:]
    x2 = x + 0.1 * v
    v2 = 0.98 * v - 0.1 * x + 0.2 * a[..., 0]
    rest2 = 0.95 * rest + 0.05 * np.roll(rest, 1, axis=-1)
    return np.concatenate([x2[..., None], v2[..., None], rest2], axis=-1)


@app.function()
def collect_rollouts(seed: int, episodes: int = 16, horizon: int = 32) -> list:
    """One worker collects a shard of (s, a, s') transitions."""
    import numpy as np

    rng = np.random.default_rng(seed)
    out = []
    for _ in range(episodes):
        s = rng.standard_normal((STATE,)).astype("float32")
        for _ in range(horizon):
            a = rng.uniform(-1, 1, ACT).astype("float32")
            s2 = env_step(s, a).astype("float32")
            out.append((s.tolist(), a.tolist(), s2.tolist()))
            s = s2
    return out


@app.function(gpu="mi355x", timeout=1200)
def train_world_model(shards: list, steps: int = 150) -> dict:
    import numpy as np
    import torch
    import torch.nn as nn

    from modal_examples_amd.train.lora import FusedAdamW

    device = "cuda" if torch.cuda.is_available() else "cpu"
    data = [t for shard in shards for t in shard]
    S = torch.tensor(np.array([d[0] for d in data], dtype="float32"), device=device)
    A = torch.tensor(np.array([d[1] for d in data], dtype="float32"), device=device)
    S2 = torch.tensor(np.array([d[2] for d in data], dtype="float32"), device=device)

    torch.manual_seed(0)
    model = nn.Sequential(
        nn.Linear(STATE + ACT, 128), nn.SiLU(),
        nn.Linear(128, 128), nn.SiLU(),
        nn.Linear(128, STATE),
    ).to(device)
    opt = FusedAdamW(list(model.parameters()), lr=2e-3)
    n = S.shape[0]
    first = last = None
    for step in range(steps):
        idx = torch.randint(0, n, (256,), device=device)
        pred = model(torch.cat([S[idx], A[idx]], -1))
        loss = ((pred - S2[idx]) ** 2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        if step == 0:
            first = float(loss)
        last = float(loss)
    torch.save(model.state_dict(), ckpts.path / "dynamics.pt")
    ckpts.commit()
    return {"first_loss": round(first, 4), "last_loss": round(last, 5)}


@app.function(gpu="mi355x")
def dream_eval(horizon: int = 16) -> dict:
    """Roll the LEARNED model forward ('dreaming') vs the real env."""
    import numpy as np
    import torch
    import torch.nn as nn

    device = "cuda" if torch.cuda.is_available() else "cpu"
    ckpts.reload()
    model = nn.Sequential(
        nn.Linear(STATE + ACT, 128), nn.SiLU(),
        nn.Linear(128, 128), nn.SiLU(),
        nn.Linear(128, STATE),
    ).to(device)
    model.load_state_dict(torch.load(ckpts.path / "dynamics.pt",
                                     map_location=device))
    rng = np.random.default_rng(99)
    s_real = rng.standard_normal((STATE,)).astype("float32")
    s_dream = torch.tensor(s_real, device=device)
    drift = []
    with torch.no_grad():
        for t in range(horizon):
            a = rng.uniform(-1, 1, ACT).astype("float32")
            s_real = env_step(s_real, a).astype("float32")
            s_dream = model(torch.cat([s_dream, torch.tensor(a, device=device)]))
            drift.append(float(np.abs(s_dream.cpu().numpy() - s_real).mean()))
    return {"drift_t1": round(drift[0], 4), "drift_final": round(drift[-1], 3)}


@app.local_entrypoint()
def main():
    shards = list(collect_rollouts.map(range(8)))
    stats = train_world_model.remote(shards)
    print("training:", stats)
    assert stats["last_loss"] < stats["first_loss"] * 0.2, stats
    dream = dream_eval.remote()
    print("dream drift:", dream)
    assert dream["drift_t1"] < 0.5, dream  # one-step prediction is tight
