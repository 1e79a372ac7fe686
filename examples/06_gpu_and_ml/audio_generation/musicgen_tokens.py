# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/audio_generation/musicgen_tokens.py"]
# ---
# # Music generation: autoregressive audio-token LM
# (reference: 06_gpu_and_ml/text-to-audio/generate_music.py — MusicGen, an
# AR transformer over EnCodec audio tokens)
#
# The musicgen shape — an autoregressive transformer over discrete audio
# tokens — end to end and hermetic: the "codec" is 8-bit mu-law at 4 kHz
# (256-token vocabulary), the corpus is synthesized scale melodies, the LM
# is the nanoGPT-class model, and generation decodes tokens back to a
# waveform written as a WAV artifact on a Volume.

import modal_examples_amd as modal

app = modal.App("example-musicgen")

tracks = modal.Volume.from_name("generated-audio", create_if_missing=True)

SR = 4000  # tokens per second of audio == sample rate (1 token / sample)
NOTE = int(0.2 * SR)  # samples per note
SCALE = [262.0, 294.0, 330.0, 349.0, 392.0, 440.0, 494.0, 523.0]  # C major


def mulaw_encode(x, mu: float = 255.0):
    import torch

    y = torch.sign(x) * torch.log1p(mu * x.abs()) / torch.log1p(torch.tensor(mu))
    return ((y + 1) / 2 * mu + 0.5).long().clamp(0, 255)


def mulaw_decode(tok, mu: float = 255.0):
    import torch

    y = tok.float() / mu * 2 - 1
    return torch.sign(y) * ((1 + mu) ** y.abs() - 1) / mu


def melody_tokens(n_notes: int, seed: int):
    """A random walk over the scale, synthesized and mu-law tokenized."""
    import torch

    g = torch.Generator().manual_seed(seed)
    t = torch.arange(NOTE) / SR
    idx, notes = 0, []
    for _ in range(n_notes):
        idx = max(0, min(len(SCALE) - 1,
                         idx + int(torch.randint(-2, 3, (1,), generator=g))))
        notes.append(0.6 * torch.sin(2 * torch.pi * SCALE[idx] * t))
    return mulaw_encode(torch.cat(notes))


@app.function(gpu="mi355x", timeout=900)
def train_and_generate(steps: int = 250, gen_seconds: float = 1.0) -> dict:
    import torch
    import torch.nn.functional as F

    from modal_examples_amd.models.gpt.model import GPT, GPTConfig

    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = GPTConfig(vocab_size=256, block_size=256, n_layer=3, n_head=4,
                    n_embd=128)
    torch.manual_seed(0)
    model = GPT(cfg).to(device)
    opt = torch.optim.AdamW(model.parameters(), lr=3e-3)

    corpus = torch.cat([melody_tokens(16, seed=s) for s in range(4)]).to(device)
    B, S = 16, cfg.block_size

    def batch(step: int):
        g = torch.Generator().manual_seed(step)
        starts = torch.randint(0, corpus.numel() - S - 1, (B,), generator=g)
        x = torch.stack([corpus[i:i + S] for i in starts])
        y = torch.stack([corpus[i + 1:i + S + 1] for i in starts])
        return x, y

    losses = []
    for step in range(steps):
        x, y = batch(step)
        _, loss = model(x, y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))

    # --- generate: prompt with one real note, continue autoregressively
    prompt = corpus[:NOTE].unsqueeze(0)
    model.eval()
    toks = model.generate(prompt, int(gen_seconds * SR), temperature=0.9,
                          seed=1234)[0]
    wave_f = mulaw_decode(toks.cpu())

    import wave

    path = tracks.path / "melody.wav"
    with wave.open(str(path), "wb") as w:
        w.setnchannels(1)
        w.setsampwidth(2)
        w.setframerate(SR)
        w.writeframes((wave_f * 32767).to(torch.int16).numpy().tobytes())
    tracks.commit()
    return {"loss_first": losses[0], "loss_last": losses[-1],
            "wav_samples": int(wave_f.numel()), "wav": str(path)}


@app.local_entrypoint()
def main(steps: int = 250):
    out = train_and_generate.remote(steps=steps)
    print({k: round(v, 3) if isinstance(v, float) else v for k, v in out.items()})
    assert out["loss_last"] < out["loss_first"] * 0.6, "audio LM did not learn"
    assert out["wav_samples"] >= SR  # prompt + 1 s of generated audio
    tracks.reload()
    assert "melody.wav" in tracks.listdir("/")
    print("audio-token LM generation OK →", out["wav"])
