"""Protein binder design package (reference: 06_gpu_and_ml/binder-design —
an installable typed package driven through Modal functions)."""
from .config import DesignConfig
from .scoring import score_binder
from .sequences import mutate, random_binder

__all__ = ["DesignConfig", "random_binder", "mutate", "score_binder"]
