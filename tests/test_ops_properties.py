"""Property-based invariants of the fp32 reference ops (ops/reference.py).

These references are the GROUND TRUTH the GPU kernels are tested against
(tests/test_kernels_gpu.py), so their own invariants get property coverage:
hypothesis searches shapes/values for violations of mathematical facts that
must hold regardless of implementation."""
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from modal_examples_amd.ops import reference as ref

dims = st.integers(min_value=1, max_value=5)
seqs = st.integers(min_value=1, max_value=17)
small = st.floats(min_value=-3, max_value=3, allow_nan=False)


@settings(max_examples=25, deadline=None)
@given(b=dims, h=dims, s=seqs, d=st.sampled_from([4, 8, 16]), seed=st.integers(0, 10**6))
def test_attention_rows_are_convex_combinations(b, h, s, d, seed):
    """Each output row of softmax(QK^T)V lies inside the convex hull of the V
    rows: max over values bounds every output coordinate."""
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(b, h, s, d, generator=g)
    k = torch.randn(b, h, s, d, generator=g)
    v = torch.randn(b, h, s, d, generator=g)
    o = ref.attention_ref(q, k, v, causal=False)
    assert o.shape == v.shape
    assert (o <= v.max() + 1e-5).all() and (o >= v.min() - 1e-5).all()


@settings(max_examples=25, deadline=None)
@given(s=st.integers(2, 12), seed=st.integers(0, 10**6))
def test_attention_causal_prefix_invariance(s, seed):
    """Causal attention output at position t must not change when future
    tokens are edited."""
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(1, 2, s, 8, generator=g)
    k = torch.randn(1, 2, s, 8, generator=g)
    v = torch.randn(1, 2, s, 8, generator=g)
    o1 = ref.attention_ref(q, k, v, causal=True)
    k2, v2 = k.clone(), v.clone()
    k2[:, :, -1] += 5.0
    v2[:, :, -1] -= 7.0
    o2 = ref.attention_ref(q, k2, v2, causal=True)
    assert torch.allclose(o1[:, :, : s - 1], o2[:, :, : s - 1], atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(rows=st.integers(1, 9), cols=st.sampled_from([8, 32, 129]),
       seed=st.integers(0, 10**6))
def test_rmsnorm_scale_invariance(rows, cols, seed):
    """rmsnorm(c*x) == rmsnorm(x) for any positive scalar c."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(rows, cols, generator=g) + 0.1
    gamma = torch.randn(cols, generator=g)
    a = ref.rmsnorm_ref(x, gamma)
    b = ref.rmsnorm_ref(3.7 * x, gamma)
    assert torch.allclose(a, b, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(rows=st.integers(1, 9), cols=st.sampled_from([8, 32, 64]),
       shift=small, seed=st.integers(0, 10**6))
def test_layernorm_shift_invariance(rows, cols, shift, seed):
    """layernorm(x + c) == layernorm(x)."""
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(rows, cols, generator=g)
    gamma = torch.ones(cols)
    beta = torch.zeros(cols)
    a = ref.layernorm_ref(x, gamma, beta)
    b = ref.layernorm_ref(x + shift, gamma, beta)
    assert torch.allclose(a, b, atol=1e-4)
    # normalized rows have ~zero mean and unit variance
    assert a.mean(-1).abs().max() < 1e-4


@settings(max_examples=25, deadline=None)
@given(b=dims, h=st.sampled_from([1, 2, 4]), s=seqs,
       d=st.sampled_from([4, 8, 16]), seed=st.integers(0, 10**6))
def test_rope_preserves_pair_norms(b, h, s, d, seed):
    """RoPE is a rotation: the norm of every (even, odd) channel pair is
    preserved exactly."""
    from modal_examples_amd.ops.functional import rope_tables

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(b, h, s, d, generator=g)
    cos, sin = rope_tables(s + 3, d)
    y = ref.rope_ref(x, cos, sin)
    half = d // 2
    nx = x[..., :half] ** 2 + x[..., half:] ** 2
    ny = y[..., :half] ** 2 + y[..., half:] ** 2
    assert torch.allclose(nx, ny, atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(1, 200), seed=st.integers(0, 10**6))
def test_silu_mul_matches_definition(n, seed):
    g = torch.Generator().manual_seed(seed)
    a = torch.randn(n, generator=g)
    b = torch.randn(n, generator=g)
    want = a * torch.sigmoid(a) * b
    assert torch.allclose(ref.silu_mul_ref(a, b).float(), want, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(seed=st.integers(0, 10**6), lr=st.floats(1e-5, 1e-2),
       wd=st.floats(0.0, 0.1))
def test_adamw_matches_torch_optimizer(seed, lr, wd):
    """One reference AdamW step == torch.optim.AdamW on the same state."""
    g = torch.Generator().manual_seed(seed)
    p0 = torch.randn(33, generator=g)
    grad = torch.randn(33, generator=g)

    p_t = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p_t], lr=lr, betas=(0.9, 0.999), eps=1e-8,
                            weight_decay=wd)
    p_t.grad = grad.clone()
    opt.step()

    p_r = p0.clone()
    m = torch.zeros_like(p_r)
    v = torch.zeros_like(p_r)
    ref.adamw_ref(p_r, grad, m, v, lr, 0.9, 0.999, 1e-8, wd, step=1)
    assert torch.allclose(p_r, p_t.detach(), atol=1e-6)


@settings(max_examples=20, deadline=None)
@given(b=dims, hq=st.sampled_from([2, 4]), seed=st.integers(0, 10**6))
def test_paged_decode_gqa_matches_dense_attention(b, hq, seed):
    """Paged GQA decode (1 query token vs cached keys) must equal dense
    attention with the kv heads broadcast to the query heads."""
    g = torch.Generator().manual_seed(seed)
    hkv, d, s = 2, 8, 11
    q = torch.randn(b, hq, d, generator=g)
    k = torch.randn(b, hkv, s, d, generator=g)
    v = torch.randn(b, hkv, s, d, generator=g)
    lens = torch.full((b,), s, dtype=torch.int32)
    o = ref.paged_decode_ref(q, k, v, None, lens, 0, 1.0 / d ** 0.5)
    rep = hq // hkv
    kd = k.repeat_interleave(rep, dim=1)
    vd = v.repeat_interleave(rep, dim=1)
    want = ref.attention_ref(q.unsqueeze(2), kd, vd, causal=False)[:, :, 0]
    assert torch.allclose(o, want, atol=1e-4)
