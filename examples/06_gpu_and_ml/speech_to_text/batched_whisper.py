# ---
# cmd: ["python", "-m", "modal_examples_amd", "run", "examples/06_gpu_and_ml/speech_to_text/batched_whisper.py"]
# ---
# # Batched Whisper transcription (the canonical fan-out + batching example)
#
# Two nested fan-outs: `.map` shards audio across container replicas, and
# inside each replica `@modal.batched` collects individual calls for up to
# `wait_ms` into ONE batched kernel launch set (the 2.8×-throughput pattern).

import modal_examples_amd as modal

app = modal.App("example-batched-whisper")


@app.cls(gpu="mi355x", scaledown_window=120)
@modal.concurrent(max_inputs=8)
class WhisperTranscriber:
    @modal.enter()
    def load(self):
        import torch

        from modal_examples_amd.models.whisper.model import WhisperConfig
        from modal_examples_amd.models.whisper.pipeline import WhisperPipeline

        gpu = torch.cuda.is_available()
        cfg = WhisperConfig.large_v3() if gpu else WhisperConfig.small_test()
        self.pipe = WhisperPipeline(
            cfg, device="cuda" if gpu else "cpu",
            dtype=torch.bfloat16 if gpu else torch.float32)

    @modal.batched(max_batch_size=16, wait_ms=1000)
    def transcribe(self, audios: list) -> list:
        import torch

        waves = [torch.as_tensor(a, dtype=torch.float32) for a in audios]
        print(f"transcribing batch of {len(waves)}")
        return self.pipe.transcribe_text(waves, max_tokens=12)


def synth_audio(i: int):
    """Synthetic 16 kHz clips (no dataset download in this environment)."""
    import math

    import numpy as np

    t = np.arange(16000, dtype=np.float32) / 16000
    return np.sin(2 * math.pi * (220 + 20 * i) * t)


@app.local_entrypoint()
def main(n: int = 8):
    model = WhisperTranscriber()
    texts = list(model.transcribe.map([synth_audio(i) for i in range(n)]))
    for i, t in enumerate(texts):
        print(f"clip {i}: {t[:60]}")
    assert len(texts) == n
