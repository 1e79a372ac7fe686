"""Flux MMDiT on hardware: hipGraph capture, LRU eviction mid-serving."""
import pytest
import torch

from modal_examples_amd.models.flux import FluxPipeline, MMDiTConfig

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@requires_gpu
def test_flux_graph_generate_and_lru_eviction():
    p = FluxPipeline(MMDiTConfig.small(), latent_size=16, use_graph=True,
                     graph_cache=1)
    a = p.generate(["x"], steps=2, seed=3)
    assert a.shape == (1, 128, 128, 3)
    b = p.generate(["x", "y"], steps=2)  # new batch key -> evicts batch-1 graph
    assert b.shape == (2, 128, 128, 3)
    assert p._graphs.evictions >= 1
    # re-capture after eviction still serves correctly; bit-identity is not
    # guaranteed across captures (workspace/algorithm re-selection) but the
    # image must be essentially the same
    a2 = p.generate(["x"], steps=2, seed=3)
    diff = (a.int() - a2.int()).abs().float()
    assert diff.mean().item() < 2.0, diff.mean().item()


@requires_gpu
def test_flux_graph_matches_eager():
    pg = FluxPipeline(MMDiTConfig.small(), latent_size=16, use_graph=True, seed=5)
    pe = FluxPipeline(MMDiTConfig.small(), latent_size=16, use_graph=False, seed=5)
    lg = pg.generate(["same prompt"], steps=2, seed=9, decode=False)
    le = pe.generate(["same prompt"], steps=2, seed=9, decode=False)
    assert (lg.float() - le.float()).abs().max().item() < 0.1
