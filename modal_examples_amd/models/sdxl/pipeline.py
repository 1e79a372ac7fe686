"""SDXL txt2img pipeline: hipGraph-captured denoise loop + fused sampler.

The MI355X-native replacement for the reference's canonical serving pipeline
(06_gpu_and_ml/stable_diffusion/text_to_image.py:92-137: enter-hook weight
load, 4-step turbo sampling, batch 1-16).  Differences by design (SURVEY.md
§7 phase 3): no torch.compile/Triton — the denoise step (UNet fwd + fused
CFG/Euler update, sigma schedule + step counter ON DEVICE) is captured once
into a hipGraph and replayed per step; VAE decode follows eagerly.

Cold-start path: weights random-init directly on device here; with
`snapshot=True` the pipeline captures a pinned-host weight snapshot after
first init so later instances restore via hipMemcpyAsync (the reference's
enable_gpu_snapshot contract, gpu_snapshot.py:41-53).
"""
from __future__ import annotations

import math
import time
from typing import List, Optional

import torch

from ...gpu.graphs import GraphLRU
from ...ops import functional as OF
from .text import encode_prompts, fourier_time_ids
from .unet import UNetConfig, UNetXL
from .vae import VAEDecoder, VAEDecoderSmall


def euler_sigmas(steps: int, num_train: int = 1000, beta_start: float = 0.00085,
                 beta_end: float = 0.012) -> torch.Tensor:
    """EulerDiscrete (scaled-linear) sigma schedule, + trailing 0."""
    betas = torch.linspace(beta_start**0.5, beta_end**0.5, num_train) ** 2
    alphas_bar = torch.cumprod(1.0 - betas, dim=0)
    all_sigmas = ((1 - alphas_bar) / alphas_bar).sqrt()
    idx = torch.linspace(num_train - 1, 0, steps).round().long()
    sig = all_sigmas[idx]
    return torch.cat([sig, torch.zeros(1)]), idx.float()


class SDXLPipeline:
    def __init__(self, cfg: Optional[UNetConfig] = None, device: str = "cuda",
                 dtype=torch.bfloat16, latent_size: int = 128,
                 use_graph: bool = True, seed: int = 0,
                 init_weights: bool = True):
        self.cfg = cfg or UNetConfig.sdxl()
        self.device = torch.device(device)
        self.dtype = dtype
        self.latent = latent_size
        self.use_graph = use_graph and self.device.type == "cuda"
        if self.device.type == "cuda":
            # MIOpen conv algos: restore the repo-cached find-db so fresh
            # boxes skip tuning.  Benchmark mode (Find-Ex, always re-times —
            # minutes of cold start) is opt-in via MODAL_AMD_CONV_BENCHMARK=1;
            # default immediate mode reads the shipped db.
            import os as _os

            from ...gpu import kernel_cache

            kernel_cache.restore()
            torch.backends.cudnn.benchmark = (
                _os.environ.get("MODAL_AMD_CONV_BENCHMARK", "0") == "1")
        torch.manual_seed(seed)
        vae_cls = VAEDecoder if self.cfg.channels[0] >= 320 else VAEDecoderSmall
        if init_weights:
            with torch.device(self.device):
                self.unet = UNetXL(self.cfg).to(self.device, dtype)
                self.vae = vae_cls().to(self.device, dtype)
        else:
            # cold-restore path: build on meta (no init compute) directly in
            # the target dtype — to_empty materializes ONCE, no f32->bf16
            # cast pass; load_state_dict fills it (bench_cold.py)
            prev = torch.get_default_dtype()
            try:
                torch.set_default_dtype(dtype)
                with torch.device("meta"):
                    self.unet = UNetXL(self.cfg)
                    self.vae = vae_cls()
            finally:
                torch.set_default_dtype(prev)
            self.unet = self.unet.to_empty(device=self.device)
            self.vae = self.vae.to_empty(device=self.device)
        self.unet.eval()
        self.vae.eval()
        self._graphs = GraphLRU(4)  # (batch, steps, cfg_on) -> graph state, LRU-bounded
        import threading

        self._graph_lock = threading.Lock()  # graph buffers are shared state
        self.image_size = latent_size * 8

    # -------------------------------------------------- conditioning

    def encode(self, prompts: List[str]):
        ctx, pooled = encode_prompts(
            prompts, self.cfg.ctx_dim, self.cfg.pooled_dim,
            device=self.device, dtype=self.dtype,
        )
        tid = fourier_time_ids(len(prompts), self.cfg.fourier_dim,
                               self.image_size, self.image_size,
                               device=self.device, dtype=self.dtype)
        add = torch.cat([pooled, tid], dim=-1)
        return ctx, add

    # -------------------------------------------------- eager denoise (CPU + fallback)

    @torch.no_grad()
    def _denoise_eager(self, x, ctx, add, sigmas, timesteps, guidance):
        cfg_on = guidance > 1.0
        if cfg_on:
            uc_ctx = torch.zeros_like(ctx)
            uc_add = add  # micro-conditioning shared
        for i in range(len(timesteps)):
            sig = float(sigmas[i])
            c_in = 1.0 / math.sqrt(sig * sig + 1.0)
            x_in = (x * c_in).to(self.dtype)
            t = torch.full((x.shape[0],), float(timesteps[i]), device=self.device)
            eps_c = self.unet(x_in, t, ctx, add)
            eps_u = self.unet(x_in, t, uc_ctx, uc_add) if cfg_on else None
            dsig = float(sigmas[i + 1] - sigmas[i])
            x = OF.cfg_euler(x, eps_c, eps_u, guidance if cfg_on else 0.0, dsig)
        return x

    # -------------------------------------------------- hipGraph denoise

    def _get_graph(self, batch: int, steps: int, guidance: float):
        key = (batch, steps, guidance > 1.0)
        st = self._graphs.get(key)
        if st is not None:
            return st
        from ...ops._build import get_ext

        ext = get_ext(required=True)
        dev = self.device
        sigmas, timesteps = euler_sigmas(steps)
        st = {
            "x": torch.zeros(batch, 4, self.latent, self.latent, device=dev, dtype=self.dtype),
            "x_in": torch.zeros(batch, 4, self.latent, self.latent, device=dev, dtype=self.dtype),
            "x_out": torch.zeros(batch, 4, self.latent, self.latent, device=dev, dtype=self.dtype),
            "ctx": torch.zeros(batch, 77, self.cfg.ctx_dim, device=dev, dtype=self.dtype),
            "add": torch.zeros(batch, self.cfg.addition_dim, device=dev, dtype=self.dtype),
            "sigmas": sigmas.float().to(dev),
            "t_all": timesteps.to(dev),
            "step": torch.zeros(1, dtype=torch.int64, device=dev),
            "guidance": guidance,
        }

        def one_step():
            ext.scale_in_dev(st["x"], st["x_in"], st["sigmas"], st["step"])
            t = st["t_all"].index_select(0, st["step"].clamp(max=steps - 1)[0:1]).expand(batch)
            eps = self.unet(st["x_in"], t, st["ctx"], st["add"])
            ext.cfg_euler_dev(st["x"], eps.contiguous(), None, st["x_out"],
                              st["sigmas"], st["step"], 0.0)
            st["x"].copy_(st["x_out"])
            st["step"].add_(1)

        # warmup on a side stream (allocator state), then capture
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

    # -------------------------------------------------- public API

    @torch.no_grad()
    def generate(self, prompts: List[str], steps: int = 4, guidance: float = 0.0,
                 seed: Optional[int] = None, decode: bool = True):
        """Returns uint8 images [B, H, W, 3] (or latents if decode=False)."""
        batch = len(prompts)
        gen = torch.Generator(device="cpu").manual_seed(seed if seed is not None else 42)
        sigmas, timesteps = euler_sigmas(steps)
        x = (torch.randn(batch, 4, self.latent, self.latent, generator=gen)
             * float(sigmas[0])).to(self.device, self.dtype)
        ctx, add = self.encode(prompts)

        if self.use_graph and guidance <= 1.0:
            with self._graph_lock:
                st = self._get_graph(batch, steps, guidance)
                st["x"].copy_(x)
                st["ctx"].copy_(ctx)
                st["add"].copy_(add)
                st["step"].zero_()
                for _ in range(steps):
                    st["graph"].replay()
                x = st["x"].clone()
        else:
            x = self._denoise_eager(x, ctx, add, sigmas, timesteps, guidance)

        if not decode:
            return x
        img = self.vae(x)
        img = ((img.float().clamp(-1, 1) + 1) * 127.5).round().to(torch.uint8)
        return img.permute(0, 2, 3, 1).contiguous()

    def param_count(self) -> int:
        return sum(p.numel() for p in self.unet.parameters()) + sum(
            p.numel() for p in self.vae.parameters()
        )

    # ------------------------------------------------ cold boot

    def save_safetensors(self, path: str) -> int:
        """Bake unet+vae into ONE safetensors-layout file ('unet.'/'vae.'
        key prefixes) for `from_safetensors` (gpu_snapshot.py role)."""
        from ...gpu import fastload

        state = {f"unet.{k}": v for k, v in self.unet.state_dict().items()}
        state.update({f"vae.{k}": v for k, v in self.vae.state_dict().items()})
        return fastload.save_file(state, path)

    @classmethod
    def from_safetensors(cls, path: str, device: str = "cuda",
                         **kw) -> "SDXLPipeline":
        """Cold-boot from baked weights: meta-init pipeline (no init
        compute, params in target dtype), weights streamed via the preadv
        pinned-staging loader and assigned as device-blob views.  Measured:
        p50 2.29 s fresh-process boot (profiles/final_cold_sdxl.txt)."""
        from ...gpu import fastload

        pipe = cls(device=device, init_weights=False, **kw)
        sd = fastload.load_file(path, device=device)
        pipe.unet.load_state_dict(
            {k[5:]: v for k, v in sd.items() if k.startswith("unet.")},
            assign=True)
        pipe.vae.load_state_dict(
            {k[4:]: v for k, v in sd.items() if k.startswith("vae.")},
            assign=True)
        return pipe
