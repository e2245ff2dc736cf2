"""T2 fuzz tier: hypothesis-driven shape/value fuzz of the
CPU backend against numpy. Catches stride/edge bugs the fixed-shape golden
tests miss; the CPU backend is the oracle for the GPU tier."""
import numpy as np
from hypothesis import given, settings, strategies as st

import nezha_amd as nz

ctx = nz.Ctx(nz.CPU)

dims = st.integers(min_value=1, max_value=48)


@settings(max_examples=25, deadline=None)
@given(m=dims, n=dims, k=dims, ta=st.booleans(), tb=st.booleans(),
       seed=st.integers(0, 2**31))
def test_fuzz_gemm(m, n, k, ta, tb, seed):
    rng = np.random.default_rng(seed)
    A = rng.normal(size=(k, m) if ta else (m, k)).astype(np.float32)
    B = rng.normal(size=(n, k) if tb else (k, n)).astype(np.float32)
    ref = (A.T if ta else A) @ (B.T if tb else B)
    c = ctx.tensor((m, n))
    nz.gemm(ctx.from_numpy(A), ctx.from_numpy(B), c, ta, tb)
    np.testing.assert_allclose(c.numpy(), ref, rtol=1e-4, atol=1e-4)


@settings(max_examples=25, deadline=None)
@given(r=dims, n=st.integers(2, 64), scale=st.floats(0.01, 4.0),
       seed=st.integers(0, 2**31))
def test_fuzz_softmax(r, n, scale, seed):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(r, n)).astype(np.float32) * 3
    y = ctx.tensor((r, n))
    nz.softmax_fwd(ctx.from_numpy(X), y, scale)
    e = np.exp(scale * X - (scale * X).max(axis=1, keepdims=True))
    ref = e / e.sum(axis=1, keepdims=True)
    np.testing.assert_allclose(y.numpy(), ref, rtol=1e-4, atol=1e-6)


@settings(max_examples=25, deadline=None)
@given(r=dims, n=st.integers(2, 64), seed=st.integers(0, 2**31))
def test_fuzz_layernorm(r, n, seed):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(r, n)).astype(np.float32)
    G = rng.normal(size=n).astype(np.float32)
    B = rng.normal(size=n).astype(np.float32)
    y = ctx.tensor((r, n))
    mean, rstd = ctx.tensor((r,)), ctx.tensor((r,))
    nz.layernorm_fwd(ctx.from_numpy(X), ctx.from_numpy(G), ctx.from_numpy(B),
                     y, mean, rstd)
    mu = X.mean(axis=1, keepdims=True)
    var = X.var(axis=1, keepdims=True)
    ref = (X - mu) / np.sqrt(var + 1e-5) * G + B
    np.testing.assert_allclose(y.numpy(), ref, rtol=1e-3, atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(n=st.integers(1, 4000), lo=st.floats(-3, 0), hi=st.floats(0.1, 3),
       seed=st.integers(0, 2**31))
def test_fuzz_rng_bounds(n, lo, hi, seed):
    t = ctx.tensor((n,)).rand_uniform(seed, lo, hi)
    a = t.numpy()
    assert a.min() >= lo and a.max() < hi


@settings(max_examples=20, deadline=None)
@given(r=dims, n=dims, seed=st.integers(0, 2**31))
def test_fuzz_colsum_addout(r, n, seed):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(r, n)).astype(np.float32)
    Y = rng.normal(size=(r, n)).astype(np.float32)
    d = ctx.tensor((n,))
    nz.colsum(ctx.from_numpy(X), d, 0.0)
    np.testing.assert_allclose(d.numpy(), X.sum(axis=0), rtol=1e-4, atol=1e-4)
    o = ctx.tensor((r, n))
    nz.add_out(o, ctx.from_numpy(X), ctx.from_numpy(Y))
    np.testing.assert_allclose(o.numpy(), X + Y, rtol=1e-5, atol=1e-6)


@settings(max_examples=10, deadline=None)
@given(b=st.integers(1, 3), s=st.integers(2, 12), h=st.integers(1, 3),
       dh=st.sampled_from([4, 8]), seed=st.integers(0, 2**31))
def test_fuzz_attn_packed(b, s, h, dh, seed):
    """Packed attention fwd (CPU composed path) vs a numpy reference."""
    rng = np.random.default_rng(seed)
    hid = h * dh
    QKV = rng.normal(size=(b * s, 3 * hid)).astype(np.float32)
    scale = 1.0 / np.sqrt(dh)
    probs = ctx.tensor((b * h * s, s))
    out = ctx.tensor((b * s, hid))
    nz.attn_fwd_packed(ctx.from_numpy(QKV), probs, out, b, s, h, dh, scale)
    qkv = QKV.reshape(b, s, 3, h, dh).transpose(2, 0, 3, 1, 4)
    q, k, v = qkv[0], qkv[1], qkv[2]
    sc = np.einsum("bhsd,bhtd->bhst", q, k) * scale
    e = np.exp(sc - sc.max(-1, keepdims=True))
    p = e / e.sum(-1, keepdims=True)
    o_ref = np.einsum("bhst,bhtd->bhsd", p, v).transpose(0, 2, 1, 3)
    np.testing.assert_allclose(out.numpy(), o_ref.reshape(b * s, hid),
                               rtol=1e-3, atol=1e-4)
