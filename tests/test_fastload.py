"""Fast weight loader: safetensors-layout roundtrip + library interop."""
import pytest
import torch

from modal_examples_amd.gpu import fastload


def _state():
    g = torch.Generator().manual_seed(3)
    return {
        "w": torch.randn(64, 32, generator=g).to(torch.bfloat16),
        "b": torch.randn(64, generator=g),
        "idx": torch.arange(10, dtype=torch.int64),
        "flag": torch.tensor([True, False]),
    }


def test_roundtrip_cpu(tmp_path):
    p = str(tmp_path / "w.safetensors")
    state = _state()
    payload = fastload.save_file(state, p)
    assert payload == sum(t.numel() * t.element_size() for t in state.values())
    out = fastload.load_file(p)
    assert set(out) == set(state)
    for k in state:
        assert out[k].dtype == state[k].dtype and out[k].shape == state[k].shape
        assert torch.equal(out[k], state[k]), k


def test_safetensors_library_reads_our_layout(tmp_path):
    """The layout IS safetensors: the reference library must read it back."""
    st = pytest.importorskip("safetensors.torch")
    p = str(tmp_path / "w.safetensors")
    state = _state()
    fastload.save_file(state, p)
    theirs = st.load_file(p)
    for k in state:
        assert torch.equal(theirs[k], state[k]), k


def test_we_read_safetensors_library_files(tmp_path):
    st = pytest.importorskip("safetensors.torch")
    p = str(tmp_path / "lib.safetensors")
    state = {k: v.contiguous() for k, v in _state().items()}
    st.save_file(state, p)
    ours = fastload.load_file(p)
    for k in state:
        assert torch.equal(ours[k], state[k]), k


def test_module_roundtrip_assign(tmp_path):
    p = str(tmp_path / "m.safetensors")
    m = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Linear(32, 4))
    m = m.to(torch.bfloat16)
    fastload.save_file(dict(m.state_dict()), p)
    m2 = torch.nn.Sequential(torch.nn.Linear(16, 32), torch.nn.Linear(32, 4))
    m2 = m2.to(torch.bfloat16)
    m2.load_state_dict(fastload.load_file(p), assign=True)
    x = torch.randn(2, 16, dtype=torch.bfloat16)
    assert torch.equal(m(x), m2(x))


@pytest.mark.gpu
def test_load_to_gpu_matches(tmp_path):
    p = str(tmp_path / "g.safetensors")
    state = {"a": torch.randn(3000, 257).to(torch.bfloat16),
             "b": torch.randn(5).float()}
    fastload.save_file(state, p)
    # small staging forces the double-buffer wraparound path
    out = fastload.load_file(p, device="cuda:0", staging_mb=1)
    for k in state:
        assert out[k].is_cuda
        assert torch.equal(out[k].cpu(), state[k]), k


def test_engine_meta_init_restore_matches():
    """LlamaEngine(init_weights=False) + fastload assign == source model."""
    import tempfile

    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig, LlamaModel

    cfg = LlamaConfig.small()
    src = LlamaModel(cfg).to(torch.bfloat16)
    p = tempfile.mkdtemp() + "/w.safetensors"
    fastload.save_file(dict(src.state_dict()), p)
    eng = LlamaEngine(cfg, device="cpu", dtype=torch.bfloat16,
                      use_graph=False, init_weights=False)
    eng.model.load_state_dict(fastload.load_file(p), assign=True)
    toks = torch.randint(0, cfg.vocab_size, (2, 12),
                         generator=torch.Generator().manual_seed(0))
    assert torch.equal(src.prefill(toks), eng.model.prefill(toks))


def test_llama_engine_safetensors_roundtrip(tmp_path):
    """save_safetensors -> from_safetensors gives an identical engine."""
    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig

    cfg = LlamaConfig.small()
    src = LlamaEngine(cfg, device="cpu", dtype=torch.bfloat16, use_graph=False)
    p = str(tmp_path / "eng.safetensors")
    src.save_safetensors(p)
    eng = LlamaEngine.from_safetensors(p, cfg=cfg, device="cpu",
                                       dtype=torch.bfloat16, use_graph=False)
    toks = torch.randint(0, cfg.vocab_size, (2, 9),
                         generator=torch.Generator().manual_seed(1))
    assert torch.equal(src.model.prefill(toks), eng.model.prefill(toks))


def test_llama_from_safetensors_build_error_propagates(tmp_path):
    from modal_examples_amd.models.llama.engine import LlamaEngine
    from modal_examples_amd.models.llama.model import LlamaConfig, LlamaModel

    p = str(tmp_path / "w.safetensors")
    fastload.save_file(dict(LlamaModel(LlamaConfig.small())
                            .to(torch.bfloat16).state_dict()), p)
    with pytest.raises(TypeError):
        LlamaEngine.from_safetensors(p, cfg=LlamaConfig.small(), device="cpu",
                                     not_a_kwarg=1)


def test_sdxl_pipeline_safetensors_roundtrip(tmp_path):
    from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
    from modal_examples_amd.models.sdxl.unet import UNetConfig

    cfg = UNetConfig.small()
    src = SDXLPipeline(cfg, device="cpu", dtype=torch.bfloat16, seed=0)
    p = str(tmp_path / "pipe.safetensors")
    src.save_safetensors(p)
    pipe = SDXLPipeline.from_safetensors(p, device="cpu", cfg=cfg,
                                         dtype=torch.bfloat16)
    for (ka, va), (kb, vb) in zip(src.unet.state_dict().items(),
                                  pipe.unet.state_dict().items()):
        assert ka == kb and torch.equal(va, vb), ka
    for (ka, va), (kb, vb) in zip(src.vae.state_dict().items(),
                                  pipe.vae.state_dict().items()):
        assert ka == kb and torch.equal(va, vb), ka


def test_fp8_and_scalar_roundtrip(tmp_path):
    """fp8 e4m3 payloads (the KV-cache dtype) and 0-dim tensors survive."""
    p = str(tmp_path / "odd.safetensors")
    state = {
        "kv": torch.randn(4, 8).to(torch.float8_e4m3fn),
        "scalar": torch.tensor(3.5),
        "empty": torch.zeros(0, 7, dtype=torch.bfloat16),
    }
    fastload.save_file(state, p)
    out = fastload.load_file(p)
    assert torch.equal(out["kv"].view(torch.uint8), state["kv"].view(torch.uint8))
    assert out["scalar"].item() == 3.5 and out["scalar"].shape == ()
    assert out["empty"].shape == (0, 7) and out["empty"].dtype == torch.bfloat16


def test_flux_and_whisper_safetensors_roundtrip(tmp_path):
    """Every flagship pipeline cold-boots from one baked file."""
    from modal_examples_amd.models.flux.mmdit import MMDiTConfig
    from modal_examples_amd.models.flux.pipeline import FluxPipeline
    from modal_examples_amd.models.whisper.model import WhisperConfig
    from modal_examples_amd.models.whisper.pipeline import WhisperPipeline

    fcfg = MMDiTConfig.small() if hasattr(MMDiTConfig, "small") else None
    if fcfg is not None:
        src = FluxPipeline(fcfg, device="cpu", use_graph=False, latent_size=8)
        p = str(tmp_path / "flux.safetensors")
        src.save_safetensors(p)
        pipe = FluxPipeline.from_safetensors(
            p, device="cpu", cfg=fcfg, use_graph=False, latent_size=8)
        for (ka, va), (kb, vb) in zip(src.model.state_dict().items(),
                                      pipe.model.state_dict().items()):
            assert ka == kb and torch.equal(va, vb), ka

    wcfg = WhisperConfig.small_test()
    if wcfg is not None:
        src = WhisperPipeline(wcfg, device="cpu")
        p = str(tmp_path / "wh.safetensors")
        src.save_safetensors(p)
        pipe = WhisperPipeline.from_safetensors(p, device="cpu", cfg=wcfg)
        for (ka, va), (kb, vb) in zip(src.model.state_dict().items(),
                                      pipe.model.state_dict().items()):
            assert ka == kb and torch.equal(va, vb), ka
    assert fcfg is not None or wcfg is not None, "no small configs found"


def test_unicode_and_long_keys(tmp_path):
    """Header JSON handles unicode and very long parameter names."""
    p = str(tmp_path / "u.safetensors")
    state = {
        "ünïcødé.wéïght": torch.randn(3, 3),
        ("blocks." + "x" * 300 + ".weight"): torch.randn(2),
    }
    fastload.save_file(state, p)
    out = fastload.load_file(p)
    for k in state:
        assert torch.equal(out[k], state[k])


def test_corrupt_header_raises_cleanly(tmp_path):
    p = str(tmp_path / "bad.safetensors")
    with open(p, "wb") as f:
        f.write((1 << 20).to_bytes(8, "little"))  # header len > file size
        f.write(b"{not json")
    import json

    with pytest.raises((json.JSONDecodeError, ValueError, OSError)):
        fastload.load_file(p)


def test_sdxl_restored_pipeline_generates_identically(tmp_path):
    """A from_safetensors-restored pipeline produces the SAME image as the
    source pipeline (same seed), end to end on CPU."""
    from modal_examples_amd.models.sdxl.pipeline import SDXLPipeline
    from modal_examples_amd.models.sdxl.unet import UNetConfig

    cfg = UNetConfig.small()
    src = SDXLPipeline(cfg, device="cpu", dtype=torch.bfloat16, seed=0,
                       latent_size=16)
    p = str(tmp_path / "pipe.safetensors")
    src.save_safetensors(p)
    dup = SDXLPipeline.from_safetensors(p, device="cpu", cfg=cfg,
                                        dtype=torch.bfloat16, latent_size=16)
    a = src.generate(["restore check"], steps=1, seed=5)
    b = dup.generate(["restore check"], steps=1, seed=5)
    assert torch.equal(a, b)
