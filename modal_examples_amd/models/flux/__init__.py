"""Flux-class MMDiT (second diffusion architecture, flux.py:111-273)."""
from .mmdit import MMDiT, MMDiTConfig
from .pipeline import FluxPipeline, flow_sigmas
