# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/speech_to_text/whisper_finetune.py"]
# ---
# # Fine-tune Whisper with LoRA
#
# The ASR fine-tuning shape (reference:
# 06_gpu_and_ml/openai_whisper/fine_tune_asr.py + finetuning/):
# LoRA adapters on the decoder projections, teacher-forced cross-entropy on
# synthetic (mel, transcript) pairs, fused-AdamW optimizer, adapter saved to
# a Volume and reloaded for a before/after comparison.  Runs on CPU with the
# small test config; pass --large on an MI355X for the large-v3 shape.

import modal_examples_amd as modal

app = modal.App("example-whisper-finetune")

adapters = modal.Volume.from_name("whisper-adapters", create_if_missing=True)


def synthetic_batch(cfg, batch: int, seed: int):
    """Deterministic (mel, tokens): the 'transcript' is a function of the
    audio so there is real signal to learn."""
    import torch

    g = torch.Generator().manual_seed(seed)
    mel = torch.randn(batch, cfg.n_mels, 2 * cfg.n_audio_ctx, generator=g)
    # token at position t depends on the audio's energy in slice t
    S = min(16, cfg.n_text_ctx)
    chunks = mel.abs().mean(dim=1).reshape(batch, -1).chunk(S, dim=1)
    tokens = torch.stack([(c.mean(dim=1) * 997).long() % cfg.vocab_size
                          for c in chunks], dim=1)
    return mel, tokens


@app.function(gpu="mi355x", timeout=600)
def finetune(steps: int = 30, rank: int = 8, lr: float = 2e-3) -> dict:
    import torch
    import torch.nn.functional as F

    from modal_examples_amd.models.whisper.model import WhisperConfig, WhisperModel
    from modal_examples_amd.train.lora import (FusedAdamW, apply_lora,
                                               lora_parameters,
                                               lora_state_dict)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = WhisperConfig.small_test()
    torch.manual_seed(0)
    model = WhisperModel(cfg).to(device)
    model.requires_grad_(False)

    wrapped = apply_lora(model, rank=rank, targets=("q", "out", "qkv"))
    params = lora_parameters(model)
    opt = FusedAdamW(params, lr=lr)

    def loss_of(seed: int) -> "torch.Tensor":
        mel, tokens = synthetic_batch(cfg, batch=4, seed=seed)
        mel, tokens = mel.to(device), tokens.to(device)
        logits = model.forward_train(mel, tokens[:, :-1])
        return F.cross_entropy(logits.reshape(-1, cfg.vocab_size),
                               tokens[:, 1:].reshape(-1))

    with torch.no_grad():
        loss0 = float(loss_of(seed=999))
    losses = []
    for step in range(steps):
        loss = loss_of(seed=step % 8)  # small epoch so the model can fit it
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))

    with torch.no_grad():
        loss1 = float(loss_of(seed=999))

    sd = lora_state_dict(model)
    torch.save(sd, adapters.path / "whisper_lora.pt")
    adapters.commit()
    return {"wrapped_layers": len(wrapped), "n_lora_params": len(params),
            "loss_first": losses[0], "loss_last": losses[-1],
            "eval_before": loss0, "eval_after": loss1}


@app.function()
def evaluate_adapter() -> float:
    """Reload the committed adapter into a FRESH model; eval loss must match
    the fine-tuned model (adapter round-trip through the Volume)."""
    import torch
    import torch.nn.functional as F

    from modal_examples_amd.models.whisper.model import WhisperConfig, WhisperModel
    from modal_examples_amd.train.lora import apply_lora

    cfg = WhisperConfig.small_test()
    torch.manual_seed(0)
    model = WhisperModel(cfg)
    apply_lora(model, rank=8, targets=("q", "out", "qkv"))
    adapters.reload()
    sd = torch.load(adapters.path / "whisper_lora.pt", weights_only=True)
    missing = model.load_state_dict(sd, strict=False).unexpected_keys
    assert not missing, f"unexpected adapter keys: {missing}"
    mel, tokens = synthetic_batch(cfg, batch=4, seed=999)
    with torch.no_grad():
        logits = model.forward_train(mel, tokens[:, :-1])
        return float(F.cross_entropy(logits.reshape(-1, cfg.vocab_size),
                                     tokens[:, 1:].reshape(-1)))


@app.local_entrypoint()
def main(steps: int = 30):
    out = finetune.remote(steps=steps)
    print({k: round(v, 3) if isinstance(v, float) else v for k, v in out.items()})
    assert out["eval_after"] < out["eval_before"], "fine-tune did not improve loss"
    reloaded = evaluate_adapter.remote()
    print(f"reloaded-adapter eval loss: {reloaded:.3f} "
          f"(fine-tuned: {out['eval_after']:.3f})")
    assert abs(reloaded - out["eval_after"]) < 1e-3
    print("whisper LoRA fine-tune OK")
