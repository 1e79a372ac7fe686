"""Flux-class MMDiT: CPU shape/structure tests + graph-LRU bound."""
import torch

from modal_examples_amd.gpu.graphs import GraphLRU
from modal_examples_amd.models.flux import FluxPipeline, MMDiT, MMDiTConfig, flow_sigmas


def test_mmdit_forward_shapes():
    cfg = MMDiTConfig.small()
    m = MMDiT(cfg).eval()
    x = torch.randn(2, 4, 8, 8)
    t = torch.tensor([500.0, 100.0])
    ctx = torch.randn(2, cfg.txt_len, cfg.ctx_dim)
    pooled = torch.randn(2, cfg.pooled_dim)
    with torch.no_grad():
        v = m(x, t, ctx, pooled)
    assert v.shape == x.shape
    assert torch.isfinite(v).all()


def test_flow_sigmas_schedule():
    s = flow_sigmas(4)
    assert s.shape == (5,) and s[0] == 1.0 and s[-1] == 0.0
    assert (s[:-1] > s[1:]).all()
    sh = flow_sigmas(4, shift=3.0)
    assert sh[0] == 1.0 and abs(float(sh[-1])) < 1e-6


def test_flux_pipeline_cpu_generates():
    p = FluxPipeline(MMDiTConfig.small(), device="cpu", dtype=torch.float32,
                     latent_size=8, use_graph=False)
    img = p.generate(["a red square", "blue circle"], steps=2)
    assert img.shape == (2, 64, 64, 3) and img.dtype == torch.uint8
    # determinism per seed
    a = p.generate(["x"], steps=2, seed=1, decode=False)
    b = p.generate(["x"], steps=2, seed=1, decode=False)
    assert torch.equal(a, b)


def test_graph_lru_evicts_oldest():
    lru = GraphLRU(2)
    lru.put((1,), {"graph": "g1"})
    lru.put((2,), {"graph": "g2"})
    assert lru.get((1,)) is not None  # touch -> (2,) is now oldest
    lru.put((3,), {"graph": "g3"})
    assert len(lru) == 2 and lru.evictions == 1
    assert lru.get((2,)) is None and (1,) in lru and (3,) in lru
