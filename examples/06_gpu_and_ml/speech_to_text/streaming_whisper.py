# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/speech_to_text/streaming_whisper.py"]
# ---
# # Streaming transcription of long audio
#
# The long-audio pattern: split on silence, `starmap` the segments across the
# pool, stream results back in order as they complete.

import modal_examples_amd as modal

app = modal.App("example-streaming-whisper")


def split_silences(audio, sr: int = 16000, min_segment_s: float = 0.5):
    """Energy-based silence splitter (the ffmpeg silencedetect analog)."""
    import numpy as np

    a = np.asarray(audio, dtype=np.float32)
    win = sr // 20
    frames = a[: len(a) - len(a) % win].reshape(-1, win)
    energy = (frames**2).mean(axis=1)
    quiet = energy < max(1e-6, float(np.median(energy)) * 0.2)
    segments, start = [], 0
    for i, q in enumerate(quiet):
        if q and (i * win - start) >= min_segment_s * sr:
            segments.append((start / sr, i * win / sr))
            start = i * win
    if (len(a) - start) / sr >= 0.05:
        segments.append((start / sr, len(a) / sr))
    return segments


@app.cls(gpu="mi355x")
class Transcriber:
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
    def transcribe_segment(self, start_s: float, end_s: float, audio) -> dict:
        import torch

        sr = 16000
        seg = torch.as_tensor(audio[int(start_s * sr):int(end_s * sr)],
                              dtype=torch.float32)
        text = self.pipe.transcribe_text([seg], max_tokens=8)[0]
        return {"start": round(start_s, 2), "end": round(end_s, 2), "text": text}


@app.local_entrypoint()
def main():
    import numpy as np

    rng = np.random.default_rng(0)
    sr = 16000
    # synthetic speech-like audio: bursts with silence gaps
    audio = np.concatenate([
        np.sin(np.linspace(0, 440 * 6.28, sr)) * 0.5,
        np.zeros(sr // 2),
        rng.standard_normal(sr) * 0.3,
        np.zeros(sr // 2),
        np.sin(np.linspace(0, 220 * 6.28, sr)) * 0.4,
    ]).astype(np.float32)
    segs = split_silences(audio)
    print(f"split into {len(segs)} segments")
    t = Transcriber()
    for res in t.transcribe_segment.starmap([(s, e, audio) for s, e in segs]):
        print(f"[{res['start']:6.2f}-{res['end']:6.2f}] {res['text'][:50]}")
