"""GPU end-to-end: SDXL-class pipeline on MI355X — hipGraph path vs eager
path numerics, and the runtime's GPU worker dispatch."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="no GPU")


@requires_gpu
def test_small_pipeline_graph_matches_eager():
    from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
    from modal_examples_amd.models.sdxl.unet import UNetConfig

    pipe_g = SDXLPipeline(UNetConfig.small(), device="cuda", latent_size=32,
                          use_graph=True, seed=7)
    # one step: later steps amplify bf16 rounding chaotically through the
    # random-weight net, so single-step agreement is the meaningful check
    lat_g = pipe_g.generate(["x"], steps=1, decode=False)
    pipe_g.use_graph = False
    lat_e = pipe_g.generate(["x"], steps=1, decode=False)
    scale = lat_e.float().abs().mean().item()
    err = (lat_g.float() - lat_e.float()).abs().max().item()
    assert err < 0.05 * max(scale, 1.0), f"graph vs eager diverged: {err} (scale {scale})"
    # replay stability (conv algos picked by benchmark mode may accumulate
    # atomically, so bit-exactness is not guaranteed — only closeness)
    pipe_g.use_graph = True
    lat_g2 = pipe_g.generate(["x"], steps=1, decode=False)
    err2 = (lat_g.float() - lat_g2.float()).abs().max().item()
    assert err2 < 0.05 * max(scale, 1.0), f"replay unstable: {err2}"


@requires_gpu
def test_small_pipeline_decode():
    from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
    from modal_examples_amd.models.sdxl.unet import UNetConfig

    pipe = SDXLPipeline(UNetConfig.small(), device="cuda", latent_size=32)
    img = pipe.generate(["a", "b"], steps=2)
    assert img.shape == (2, 256, 256, 3) and img.dtype == torch.uint8


@requires_gpu
def test_runtime_gpu_worker_sees_device(gpu_env):
    import modal_examples_amd as modal

    app = modal.App("gpu-test")

    @app.function(gpu="mi355x")
    def probe():
        import torch

        return {
            "visible": os.environ.get("HIP_VISIBLE_DEVICES"),
            "available": torch.cuda.is_available(),
            "count": torch.cuda.device_count(),
        }

    out = probe.remote()
    assert out["available"] and out["count"] == 1
    assert out["visible"] is not None


@requires_gpu
def test_snapshot_cold_start_speedup():
    """Snapshot restore must beat fresh random-init + no it's about IO: here we
    check restore correctness + that restore of ~1GB achieves >5 GB/s."""
    import time

    from modal_examples_amd.gpu.snapshot import WeightSnapshot

    t = {f"w{i}": torch.randn(64, 1024, 1024, device="cuda", dtype=torch.bfloat16)
         for i in range(4)}  # 4 × 128 MB = 512 MB
    snap = WeightSnapshot.capture(t)
    ref0 = t["w0"].clone()
    for x in t.values():
        x.zero_()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    snap.restore(t)
    dt = time.perf_counter() - t0
    assert torch.equal(t["w0"], ref0)
    gbps = snap.total_bytes / dt / 1e9
    assert gbps > 5.0, f"restore only {gbps:.1f} GB/s"
    snap.close()
