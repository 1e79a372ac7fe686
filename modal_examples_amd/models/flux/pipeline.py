"""Flux-class txt2img pipeline: rectified flow matching + hipGraph denoise.

Reference role: 06_gpu_and_ml/stable_diffusion/flux.py:111-273 — the
"canonical perf" diffusion example whose lever is torch.compile with cached
inductor/triton artifacts.  The MI355X replacement for that lever is ahead-of-
time hipGraph capture (seconds, not a 20-minute compile; nothing to cache on
a Volume except the MIOpen find-db the SDXL pipeline already ships).

The flow-matching Euler update x += (sigma_next - sigma)*v reuses the fused
device-sigma kernel (K4's cfg_euler_dev) with the flow schedule resident on
device, so the whole step replays as one graph.  Graph cache is LRU-bounded.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch

from ...gpu.graphs import GraphLRU
from ..sdxl.text import encode_prompts
from ..sdxl.vae import VAEDecoder, VAEDecoderSmall
from .mmdit import MMDiT, MMDiTConfig


def flow_sigmas(steps: int, shift: float = 1.0) -> torch.Tensor:
    """Rectified-flow schedule 1 -> 0 with optional timestep shift."""
    s = torch.linspace(1.0, 0.0, steps + 1)
    if shift != 1.0:
        s = shift * s / (1 + (shift - 1) * s)
    return s


class FluxPipeline:
    def __init__(self, cfg: Optional[MMDiTConfig] = None, device: str = "cuda",
                 dtype=torch.bfloat16, latent_size: int = 128,
                 use_graph: bool = True, seed: int = 0, graph_cache: int = 4,
                 init_weights: bool = True):
        self.cfg = cfg or MMDiTConfig.schnell()
        self.device = torch.device(device)
        self.dtype = dtype
        self.latent = latent_size
        self.use_graph = use_graph and self.device.type == "cuda"
        if self.device.type == "cuda":
            from ...gpu import kernel_cache

            kernel_cache.restore()
        torch.manual_seed(seed)
        vae_cls = VAEDecoder if self.cfg.hidden >= 1024 else VAEDecoderSmall
        if init_weights:
            with torch.device(self.device):
                self.model = MMDiT(self.cfg).to(self.device, dtype)
                self.vae = vae_cls().to(self.device, dtype)
        else:
            # cold-restore: meta-build in target dtype, weights assigned by
            # from_safetensors (gpu/fastload.py blob views)
            prev = torch.get_default_dtype()
            try:
                torch.set_default_dtype(dtype)
                with torch.device("meta"):
                    self.model = MMDiT(self.cfg)
                    self.vae = vae_cls()
            finally:
                torch.set_default_dtype(prev)
            self.model = self.model.to_empty(device=self.device)
            self.vae = self.vae.to_empty(device=self.device)
        self.model.eval()
        self.vae.eval()
        self._graphs = GraphLRU(graph_cache)
        import threading

        self._graph_lock = threading.Lock()  # shared graph buffers: one
        # generate at a time per pipeline (concurrent callers serialize)
        self.image_size = latent_size * 8

    def encode(self, prompts: List[str]):
        ctx, pooled = encode_prompts(
            prompts, self.cfg.ctx_dim, self.cfg.pooled_dim,
            seq_len=self.cfg.txt_len, device=self.device, dtype=self.dtype)
        return ctx, pooled

    @torch.no_grad()
    def _denoise_eager(self, x, ctx, pooled, sigmas):
        for i in range(len(sigmas) - 1):
            t = torch.full((x.shape[0],), float(sigmas[i]) * 1000.0,
                           device=self.device)
            v = self.model(x, t, ctx, pooled)
            x = x + (float(sigmas[i + 1]) - float(sigmas[i])) * v.float()
            x = x.to(self.dtype)
        return x

    def _get_graph(self, batch: int, steps: int):
        key = (batch, steps)
        st = self._graphs.get(key)
        if st is not None:
            return st
        from ...ops._build import get_ext

        ext = get_ext(required=True)
        dev = self.device
        sigmas = flow_sigmas(steps)
        st = {
            "x": torch.zeros(batch, self.cfg.latent_channels, self.latent,
                             self.latent, device=dev, dtype=self.dtype),
            "x_out": torch.zeros(batch, self.cfg.latent_channels, self.latent,
                                 self.latent, device=dev, dtype=self.dtype),
            "ctx": torch.zeros(batch, self.cfg.txt_len, self.cfg.ctx_dim,
                               device=dev, dtype=self.dtype),
            "pooled": torch.zeros(batch, self.cfg.pooled_dim, device=dev,
                                  dtype=self.dtype),
            "sigmas": sigmas.float().to(dev),
            "step": torch.zeros(1, dtype=torch.int64, device=dev),
        }

        def one_step():
            sig = st["sigmas"].index_select(0, st["step"].clamp(max=steps - 1))
            t = (sig * 1000.0).expand(batch)
            v = self.model(st["x"], t, st["ctx"], st["pooled"])
            ext.cfg_euler_dev(st["x"], v.contiguous(), None, st["x_out"],
                              st["sigmas"], st["step"], 0.0)
            st["x"].copy_(st["x_out"])
            st["step"].add_(1)

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(2):
                st["step"].zero_()
                one_step()
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        st["step"].zero_()
        with torch.cuda.graph(g), torch.no_grad():
            one_step()
        st["graph"] = g
        self._graphs.put(key, st)
        return st

    @torch.no_grad()
    def generate(self, prompts: List[str], steps: int = 4,
                 seed: Optional[int] = None, decode: bool = True):
        batch = len(prompts)
        gen = torch.Generator(device="cpu").manual_seed(seed if seed is not None else 42)
        x = torch.randn(batch, self.cfg.latent_channels, self.latent,
                        self.latent, generator=gen).to(self.device, self.dtype)
        ctx, pooled = self.encode(prompts)
        sigmas = flow_sigmas(steps)
        if self.use_graph:
            with self._graph_lock:
                st = self._get_graph(batch, steps)
                st["x"].copy_(x)
                st["ctx"].copy_(ctx)
                st["pooled"].copy_(pooled)
                st["step"].zero_()
                for _ in range(steps):
                    st["graph"].replay()
                x = st["x"].clone()
        else:
            x = self._denoise_eager(x, ctx, pooled, sigmas)
        if not decode:
            return x
        img = self.vae(x)
        img = ((img.float().clamp(-1, 1) + 1) * 127.5).round().to(torch.uint8)
        return img.permute(0, 2, 3, 1).contiguous()

    def param_count(self) -> int:
        return sum(p.numel() for p in self.model.parameters()) + sum(
            p.numel() for p in self.vae.parameters())

    # ------------------------------------------------ cold boot

    def save_safetensors(self, path: str) -> int:
        """Bake mmdit+vae into one safetensors-layout file (the compile-
        cache Volume role of flux.py:246-272 — here the bakeable artifact is
        the weights; graphs capture in seconds at boot)."""
        from ...gpu import fastload

        state = {f"model.{k}": v for k, v in self.model.state_dict().items()}
        state.update({f"vae.{k}": v for k, v in self.vae.state_dict().items()})
        return fastload.save_file(state, path)

    @classmethod
    def from_safetensors(cls, path: str, device: str = "cuda",
                         **kw) -> "FluxPipeline":
        from ...gpu import fastload

        pipe = cls(device=device, init_weights=False, **kw)
        sd = fastload.load_file(path, device=device)
        pipe.model.load_state_dict(
            {k[6:]: v for k, v in sd.items() if k.startswith("model.")},
            assign=True)
        pipe.vae.load_state_dict(
            {k[4:]: v for k, v in sd.items() if k.startswith("vae.")},
            assign=True)
        return pipe
