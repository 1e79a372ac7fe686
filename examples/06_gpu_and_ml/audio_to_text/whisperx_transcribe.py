# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/audio_to_text/whisperx_transcribe.py"]
# ---
# # WhisperX-style transcription: word timestamps + speaker turns
#
# The whisperx pipeline role (reference: audio-to-text/whisperx_transcribe.py):
# transcribe long audio, then ALIGN — word-level timestamps — and attribute
# speaker turns.  MI355X mapping: transcription runs the Whisper kernels
# (K5/K6); alignment is energy-weighted forced distribution of the words over
# each segment; "diarization" clusters segments by spectral signature (two
# synthetic speakers with distinct bands are separated cleanly).

import modal_examples_amd as modal

app = modal.App("example-whisperx")


@app.cls(gpu="mi355x", timeout=600)
class WhisperX:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.whisper.model import WhisperConfig
        from modal_examples_amd.models.whisper.pipeline import WhisperPipeline

        gpu = torch.cuda.is_available()
        cfg = WhisperConfig.large_v3() if gpu else WhisperConfig.small_test()
        self.pipe = WhisperPipeline(cfg, device="cuda" if gpu else "cpu",
                                    dtype=torch.bfloat16 if gpu else torch.float32)

    @modal.method()
    def transcribe_align(self, audio, sr: int = 16000) -> dict:
        """Transcript + per-word [start, end] + a speaker label per segment."""
        import numpy as np
        import torch

        a = np.asarray(audio, dtype=np.float32)
        seg_len = sr  # 1 s segments
        segs = [a[i:i + seg_len] for i in range(0, len(a), seg_len)
                if len(a[i:i + seg_len]) > sr // 10]
        texts = self.pipe.transcribe_text(
            [torch.as_tensor(s) for s in segs], max_tokens=6)

        # --- alignment: distribute each segment's words over its span,
        # weighted by short-window energy (the wav2vec-CTC role)
        words, spk_feats = [], []
        for si, (seg, text) in enumerate(zip(segs, texts)):
            t0 = si * seg_len / sr
            ws = text.split() or ["..."]
            win = sr // 50
            frames = seg[: len(seg) - len(seg) % win].reshape(-1, win)
            energy = (frames ** 2).mean(axis=1) + 1e-8
            cum = np.concatenate([[0.0], np.cumsum(energy) / energy.sum()])
            bounds = np.interp(np.linspace(0, 1, len(ws) + 1), cum,
                               np.arange(len(cum)) * win / sr)
            for wi, wrd in enumerate(ws):
                words.append({"word": wrd, "start": round(t0 + bounds[wi], 3),
                              "end": round(t0 + bounds[wi + 1], 3),
                              "segment": si})
            # spectral signature for diarization: low/high band energy ratio
            spec = np.abs(np.fft.rfft(seg))
            half = len(spec) // 2
            spk_feats.append(float(spec[:half].sum() / (spec.sum() + 1e-8)))

        # --- 2-speaker clustering on the band ratio (threshold at midpoint)
        f = np.asarray(spk_feats)
        thr = (f.min() + f.max()) / 2
        speakers = ["S1" if v >= thr else "S2" for v in f]
        return {"segments": [{"text": t, "speaker": speakers[i]}
                             for i, t in enumerate(texts)],
                "words": words}


@app.local_entrypoint()
def main():
    import numpy as np

    sr = 16000
    rng = np.random.default_rng(3)
    # speaker A: low-frequency tone bursts; speaker B: high-frequency noise
    t = np.linspace(0, 1, sr, endpoint=False)
    spk_a = (np.sin(2 * np.pi * 150 * t) * 0.6).astype(np.float32)
    spk_b = rng.standard_normal(sr).astype(np.float32) * 0.3
    audio = np.concatenate([spk_a, spk_b, spk_a, spk_b])

    wx = WhisperX()
    out = wx.transcribe_align.remote(audio)
    assert len(out["segments"]) == 4
    labels = [s["speaker"] for s in out["segments"]]
    assert labels[0] == labels[2] and labels[1] == labels[3] and labels[0] != labels[1], labels
    # word timestamps are monotone within each segment and inside its span
    for w in out["words"]:
        assert 0.0 <= w["start"] <= w["end"] <= 4.0, w
    print(f"speakers: {labels}")
    print(f"{len(out['words'])} aligned words; first: {out['words'][0]}")
