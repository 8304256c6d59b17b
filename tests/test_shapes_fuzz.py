"""Property-based shape fuzzing of the conv/upfirdn dispatch (CPU eager
path) — the dispatch has many eligibility branches (slab/s2/up2/GEMM/
generic); every branch must agree with torch reference semantics."""

import pytest
import torch
import torch.nn.functional as F

from hypothesis import given, settings, strategies as st

from gansformer_amd.ops.conv2d_grad import conv2d_gradfix, conv2d_up2
from gansformer_amd.ops.upfirdn2d import setup_filter, upfirdn2d, _eager_upfirdn2d


@settings(max_examples=25, deadline=None)
@given(
    b=st.integers(1, 3), i=st.integers(1, 9), o=st.integers(1, 9),
    h=st.integers(4, 14), w=st.integers(4, 14),
    k=st.sampled_from([1, 3]), stride=st.sampled_from([1, 2]),
)
def test_conv2d_gradfix_fuzz(b, i, o, h, w, k, stride):
    torch.manual_seed(0)
    pad = k // 2
    if (h + 2 * pad - k) // stride + 1 <= 0:
        return
    x = torch.randn(b, i, h, w, dtype=torch.float64, requires_grad=True)
    wt = torch.randn(o, i, k, k, dtype=torch.float64) * 0.3
    y = conv2d_gradfix(x, wt, stride=stride, padding=pad)
    ref = F.conv2d(x, wt, stride=stride, padding=pad)
    assert torch.allclose(y, ref, atol=1e-10)
    # backward agreement
    dy = torch.randn_like(y)
    g1 = torch.autograd.grad(y, x, dy, retain_graph=True)[0]
    g2 = torch.autograd.grad(ref, x, dy)[0]
    assert torch.allclose(g1, g2, atol=1e-10)


@settings(max_examples=25, deadline=None)
@given(
    b=st.integers(1, 2), c=st.integers(1, 5),
    h=st.integers(3, 12), w=st.integers(3, 12),
    up=st.sampled_from([1, 2]), down=st.sampled_from([1, 2]),
    p0=st.integers(0, 3), p1=st.integers(0, 3),
)
def test_upfirdn2d_fuzz(b, c, h, w, up, down, p0, p1):
    torch.manual_seed(1)
    f = setup_filter([1, 3, 3, 1])
    if (h * up + p0 + p1 - 4) // down + 1 <= 0:
        return
    if (w * up + p0 + p1 - 4) // down + 1 <= 0:
        return
    x = torch.randn(b, c, h, w)
    y = upfirdn2d(x, f, up=up, down=down, padding=(p0, p1, p0, p1), gain=1.5)
    ref = _eager_upfirdn2d(x, f, (up, up), (down, down), (p0, p1, p0, p1), 1.5)
    assert torch.allclose(y, ref, atol=1e-5)


@settings(max_examples=15, deadline=None)
@given(
    b=st.integers(1, 2), i=st.integers(1, 6), o=st.integers(1, 6),
    h=st.integers(2, 8), w=st.integers(2, 8),
)
def test_conv2d_up2_fuzz(b, i, o, h, w):
    torch.manual_seed(2)
    from gansformer_amd.ops.conv2d_grad import _zero_stuff2
    x = torch.randn(b, i, h, w, dtype=torch.float64)
    wt = torch.randn(o, i, 3, 3, dtype=torch.float64) * 0.3
    y = conv2d_up2(x, wt)
    ref = F.conv2d(_zero_stuff2(x), wt, padding=1)
    assert torch.allclose(y, ref, atol=1e-10)


@settings(max_examples=20, deadline=None)
@given(
    b=st.integers(1, 3), c=st.integers(1, 6), n=st.integers(8, 64),
)
def test_modnorm_fuzz(b, c, n):
    from gansformer_amd.ops.modnorm import modnorm
    torch.manual_seed(3)
    x = torch.randn(b, c, n, dtype=torch.float64)
    g = torch.randn(b, c, n, dtype=torch.float64) * 0.3
    bt = torch.randn(b, c, n, dtype=torch.float64) * 0.2
    y = modnorm(x, g, bt)
    m = x.mean(-1, keepdim=True)
    v = x.var(-1, keepdim=True, unbiased=False)
    ref = (x - m) * (v + 1e-8).rsqrt() * (1 + g) + bt
    assert torch.allclose(y, ref, atol=1e-9)


@settings(max_examples=20, deadline=None)
@given(
    b=st.integers(1, 3), c=st.integers(1, 5), h=st.integers(2, 9),
    use_noise=st.booleans(), use_bias=st.booleans(),
)
def test_mod_bias_act_fuzz(b, c, h, use_noise, use_bias):
    import math as _m
    from gansformer_amd.ops.fused_act import mod_bias_act
    torch.manual_seed(4)
    x = torch.randn(b, c, h, h, dtype=torch.float64)
    d = torch.rand(b, c, dtype=torch.float64) + 0.5
    n = torch.randn(b, 1, h, h, dtype=torch.float64) if use_noise else None
    sig = torch.tensor([0.4], dtype=torch.float64)
    bias = (torch.randn(c, dtype=torch.float64) * 0.1) if use_bias else None
    y = mod_bias_act(x, d, noise=n, sigma=sig, b=bias, act="lrelu",
                     clamp=8.0)
    v = x * d.reshape(b, c, 1, 1)
    if use_noise:
        v = v + n * sig
    if use_bias:
        v = v + bias.reshape(1, c, 1, 1)
    ref = torch.clamp(torch.nn.functional.leaky_relu(v, 0.2)
                      * _m.sqrt(2.0), -8.0, 8.0)
    assert torch.allclose(y, ref, atol=1e-10)


@settings(max_examples=20, deadline=None)
@given(
    b=st.integers(1, 3), nq=st.integers(1, 70), nk=st.integers(1, 70),
    d=st.sampled_from([8, 16, 33]), e=st.sampled_from([8, 24]),
)
def test_bipartite_attention_fuzz(b, nq, nk, d, e):
    """Attention op vs plain softmax composition over random shapes
    (fwd + grads, fp64 so the CPU eager path is exact)."""
    from gansformer_amd.ops.bipartite import bipartite_attention
    torch.manual_seed(0)
    q = torch.randn(b, nq, d, dtype=torch.float64, requires_grad=True)
    k = torch.randn(b, nk, d, dtype=torch.float64, requires_grad=True)
    v = torch.randn(b, nk, e, dtype=torch.float64, requires_grad=True)
    out = bipartite_attention(q, k, v)
    a = torch.softmax((q @ k.transpose(1, 2)) * d ** -0.5, dim=-1)
    ref = a @ v
    assert torch.allclose(out, ref, atol=1e-10)
    g = torch.randn_like(out)
    got = torch.autograd.grad(out, (q, k, v), g, retain_graph=True)
    want = torch.autograd.grad(ref, (q, k, v), g)
    for gg, ww in zip(got, want):
        assert torch.allclose(gg, ww, atol=1e-9)


@settings(max_examples=15, deadline=None)
@given(
    b=st.integers(2, 6), c=st.integers(2, 8), hw=st.sampled_from([3, 5]),
    gs=st.integers(1, 6), f=st.sampled_from([1, 2]),
)
def test_mbstd_fuzz(b, c, hw, gs, f):
    from gansformer_amd.ops import minibatch_stddev
    if c % f:
        return
    torch.manual_seed(1)
    x = torch.randn(b, c, hw, hw, dtype=torch.float64, requires_grad=True)
    y = minibatch_stddev(x, group_size=gs, num_channels=f)
    assert y.shape == (b, c + f, hw, hw)
    assert torch.allclose(y[:, :c], x)
    y.square().sum().backward()
    assert torch.isfinite(x.grad).all()
