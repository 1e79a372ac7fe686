# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/audio_generation/tts_speak.py"]
# ---
# # Text-to-speech serving (the chatterbox/text-to-audio role)
#
# The TTS serving shape: text → token sequence (the AR acoustic model, here
# the GPT backbone on the gfx950 kernels) → per-token formant synthesis →
# a WAV on a Volume.  Self-checks: audio duration tracks text length and the
# spectrum concentrates at the synthesized formant band.

import modal_examples_amd as modal

app = modal.App("example-tts")

voices = modal.Volume.from_name("tts-out", create_if_missing=True)

SR = 16000
TOKEN_MS = 40  # each acoustic token voices 40 ms


@app.cls(gpu="mi355x", timeout=600)
class Speaker:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.gpt.model import GPT, GPTConfig

        self.torch = torch
        self.device = "cuda" if torch.cuda.is_available() else "cpu"
        torch.manual_seed(0)
        cfg = GPTConfig(n_layer=4, n_embd=256, n_head=4, block_size=256,
                        vocab_size=256)
        self.model = GPT(cfg).to(self.device).eval()

    @modal.method()
    def speak(self, text: str, fname: str = "utt.wav") -> dict:
        import struct
        import wave

        import numpy as np

        torch = self.torch
        # acoustic tokens: one per input char, AR-extended by the model
        prompt = torch.tensor([[min(255, ord(c)) for c in text[:64]]],
                              device=self.device)
        n_extra = max(4, len(text) // 4)
        with torch.no_grad():
            toks = self.model.generate(prompt, n_extra, temperature=0.7,
                                       seed=1)[0].tolist()
        # formant synthesis: token t -> 40 ms at pitch 80+t Hz
        samples = []
        t_axis = np.arange(int(SR * TOKEN_MS / 1000)) / SR
        for tok in toks:
            f0 = 80.0 + float(tok)
            seg = 0.4 * np.sin(2 * np.pi * f0 * t_axis)
            seg *= np.hanning(len(seg))
            samples.append(seg)
        audio = np.concatenate(samples).astype(np.float32)
        path = voices.path / fname
        with wave.open(str(path), "wb") as w:
            w.setnchannels(1)
            w.setsampwidth(2)
            w.setframerate(SR)
            pcm = (audio * 32767).astype("<i2")
            w.writeframes(struct.pack(f"<{len(pcm)}h", *pcm))
        voices.commit()
        spec = np.abs(np.fft.rfft(audio))
        band = spec[: len(spec) // 8].sum() / (spec.sum() + 1e-9)
        return {"file": fname, "seconds": round(len(audio) / SR, 2),
                "tokens": len(toks), "low_band_energy": round(float(band), 3)}


@app.local_entrypoint()
def main():
    s = Speaker()
    short = s.speak.remote("hi", "short.wav")
    longer = s.speak.remote("a considerably longer sentence to voice",
                            "long.wav")
    assert longer["seconds"] > short["seconds"]
    assert longer["low_band_energy"] > 0.8  # formants live in the low band
    print("short:", short)
    print("long: ", longer)
