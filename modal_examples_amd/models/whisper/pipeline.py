"""Whisper transcription pipeline: log-mel frontend + batched greedy decode.

Mirrors the reference's batched_whisper flow (batched_whisper.py:127-138:
``@modal.batched(max_batch_size=64)`` feeding ``pipeline(audio, batch_size)``)
— here the batch feeds the gfx950 encoder attention in one launch set and the
decoder loop runs all sequences in lockstep over contiguous KV caches.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch

from .model import WhisperConfig, WhisperModel

SAMPLE_RATE = 16000
HOP = 160
N_FFT = 400


def log_mel_spectrogram(audio: torch.Tensor, n_mels: int, n_frames: int) -> torch.Tensor:
    """audio [B, samples] f32 → log-mel [B, n_mels, n_frames*2] (Whisper DSP:
    400-pt STFT, hop 160, mel filterbank, log10, max-normalized)."""
    device = audio.device
    window = torch.hann_window(N_FFT, device=device)
    stft = torch.stft(audio, N_FFT, HOP, window=window, return_complex=True)
    mag = stft.abs() ** 2  # [B, n_fft/2+1, T]
    # triangular mel filterbank
    n_freqs = N_FFT // 2 + 1
    mel_min, mel_max = 0.0, 2595.0 * math.log10(1 + (SAMPLE_RATE / 2) / 700.0)
    mel_pts = torch.linspace(mel_min, mel_max, n_mels + 2, device=device)
    hz_pts = 700.0 * (10 ** (mel_pts / 2595.0) - 1)
    bins = (hz_pts / (SAMPLE_RATE / 2) * (n_freqs - 1)).long()
    fb = torch.zeros(n_mels, n_freqs, device=device)
    for m in range(n_mels):
        lo, c, hi = bins[m], bins[m + 1], bins[m + 2]
        if c > lo:
            fb[m, lo:c] = (torch.arange(lo, c, device=device) - lo) / max(1, (c - lo))
        if hi > c:
            fb[m, c:hi] = (hi - torch.arange(c, hi, device=device)) / max(1, (hi - c))
    mel = fb @ mag
    logmel = torch.clamp(mel, min=1e-10).log10()
    logmel = torch.maximum(logmel, logmel.amax(dim=(1, 2), keepdim=True) - 8.0)
    logmel = (logmel + 4.0) / 4.0
    T = n_frames * 2
    if logmel.shape[-1] < T:
        logmel = torch.nn.functional.pad(logmel, (0, T - logmel.shape[-1]))
    return logmel[..., :T]


class WhisperPipeline:
    def __init__(self, cfg: Optional[WhisperConfig] = None, device: str = "cuda",
                 dtype=torch.bfloat16, seed: int = 0, init_weights: bool = True):
        self.cfg = cfg or WhisperConfig.large_v3()
        self.device = torch.device(device)
        self.dtype = dtype
        torch.manual_seed(seed)
        if init_weights:
            with torch.device(self.device):
                self.model = WhisperModel(self.cfg).to(self.device, dtype)
        else:
            # cold-restore: meta-build in target dtype (from_safetensors)
            prev = torch.get_default_dtype()
            try:
                torch.set_default_dtype(dtype)
                with torch.device("meta"):
                    self.model = WhisperModel(self.cfg)
            finally:
                torch.set_default_dtype(prev)
            self.model = self.model.to_empty(device=self.device)
        self.model.eval()
        self.sot, self.eot = 1, 2  # synthetic special tokens
        self.use_graph = self.device.type == "cuda"
        import threading

        self._graph_lock = threading.Lock()  # graph state is per-pipeline;
        # serialize transcribe under @modal.concurrent callers
        from ...gpu.graphs import GraphLRU

        self._graphs = GraphLRU(3)  # per-batch-size captured decode steps

    def _decode_state(self, B: int):
        """Persistent per-batch-size decode state + ONE captured step:
        consume cur @ pos -> logits -> cur := argmax, pos += 1 (all on
        device; the host loop is replay + one [B] read per token)."""
        st = self._graphs.get(B)
        if st is not None:
            return st
        cfg = self.cfg
        H, D = cfg.n_head, cfg.n_state // cfg.n_head
        dev = self.device
        caches = [{
            "persistent": True,
            "k": torch.zeros(B, H, cfg.n_text_ctx, D, device=dev, dtype=self.dtype),
            "v": torch.zeros(B, H, cfg.n_text_ctx, D, device=dev, dtype=self.dtype),
            "ck": torch.zeros(B, H, cfg.n_audio_ctx, D, device=dev, dtype=self.dtype),
            "cv": torch.zeros(B, H, cfg.n_audio_ctx, D, device=dev, dtype=self.dtype),
        } for _ in range(cfg.n_text_layer)]
        st = {
            "caches": caches,
            "cur": torch.zeros(B, dtype=torch.long, device=dev),
            "pos": torch.ones(1, dtype=torch.long, device=dev),
        }

        def one_step():
            logits = self.model.decode_step_dev(st["cur"], st["pos"],
                                                st["caches"], cfg.n_audio_ctx)
            st["cur"].copy_(logits.argmax(-1))
            st["pos"].add_(1)

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side), torch.no_grad():
            for _ in range(2):
                st["pos"].fill_(1)
                one_step()
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        st["pos"].fill_(1)
        with torch.cuda.graph(g), torch.no_grad():
            one_step()
        st["graph"] = g
        self._graphs.put(B, st)
        return st

    @torch.no_grad()
    def transcribe(self, audio_batch: List[torch.Tensor], max_tokens: int = 32
                   ) -> List[List[int]]:
        """Batch of mono 16 kHz waveforms → token id sequences (greedy)."""
        if self.use_graph:
            with self._graph_lock:
                return self._transcribe_inner(audio_batch, max_tokens)
        return self._transcribe_inner(audio_batch, max_tokens)

    def _transcribe_inner(self, audio_batch, max_tokens):
        B_req = len(audio_batch)
        cfg = self.cfg
        if self.use_graph:
            # the dynamic batcher produces VARIABLE batch sizes; pad to the
            # next power of two so graph captures stay bounded (<=7 keys)
            B = 1
            while B < B_req:
                B *= 2
            audio_batch = list(audio_batch) + [
                torch.zeros(1600) for _ in range(B - B_req)]
        else:
            B = B_req
        maxlen = cfg.n_audio_ctx * 2 * HOP
        padded = torch.zeros(B, maxlen, device=self.device)
        for i, a in enumerate(audio_batch):
            a = a.to(self.device).float()[:maxlen]
            padded[i, : a.numel()] = a
        mel = log_mel_spectrogram(padded, cfg.n_mels, cfg.n_audio_ctx).to(self.dtype)
        audio = self.model.encode(mel)

        H, D = cfg.n_head, cfg.n_state // cfg.n_head
        if self.use_graph:
            # hipGraph decode (K10): persistent caches + one captured step
            st = self._decode_state(B)
            caches = st["caches"]
        else:
            caches = [
                {
                    "k": torch.zeros(B, H, cfg.n_text_ctx, D, device=self.device, dtype=self.dtype),
                    "v": torch.zeros(B, H, cfg.n_text_ctx, D, device=self.device, dtype=self.dtype),
                }
                for _ in range(cfg.n_text_layer)
            ]
        tokens = torch.full((B, 1), self.sot, device=self.device, dtype=torch.long)
        logits = self.model.decode_prefill(tokens, audio, caches)
        outs = [[] for _ in range(B)]
        alive = torch.ones(B, dtype=torch.bool)
        cur = logits.argmax(-1)
        for i in range(B):
            outs[i].append(int(cur[i]))
        if self.use_graph:
            st["cur"].copy_(cur)
            st["pos"].fill_(1)
        for pos in range(1, min(max_tokens, cfg.n_text_ctx - 1)):
            if self.use_graph:
                st["graph"].replay()
                cur_h = st["cur"].cpu()  # ONE [B] read per token
            else:
                logits = self.model.decode_step(cur, pos, caches, cfg.n_audio_ctx)
                cur = logits.argmax(-1)
                cur_h = cur.cpu()
            for i in range(B):
                if alive[i]:
                    t = int(cur_h[i])
                    outs[i].append(t)
                    if t == self.eot:
                        alive[i] = False
            if not alive.any():
                break
        return outs[:B_req]

    def transcribe_text(self, audio_batch, max_tokens: int = 32) -> List[str]:
        return [" ".join(f"t{t}" for t in seq if t > 2)
                for seq in self.transcribe(audio_batch, max_tokens)]

    # ------------------------------------------------ cold boot

    def save_safetensors(self, path: str) -> int:
        """Bake the model for `from_safetensors` cold boot."""
        from ...gpu import fastload

        return fastload.save_file(dict(self.model.state_dict()), path)

    @classmethod
    def from_safetensors(cls, path: str, device: str = "cuda",
                         **kw) -> "WhisperPipeline":
        from ...gpu import fastload

        pipe = cls(device=device, init_weights=False, **kw)
        pipe.model.load_state_dict(fastload.load_file(path, device=device),
                                   assign=True)
        return pipe
