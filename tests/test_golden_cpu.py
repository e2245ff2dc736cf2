"""T2: CPU f32 backend vs numpy/PyTorch golden values.

The CPU backend validated here is the oracle the HIP kernels are compared
against in the GPU tier (test_gpu_kernels.py).
"""
import numpy as np
import pytest
import torch

import nezha_amd as nz

RNG = np.random.default_rng(1234)


def randf(*shape):
    return RNG.normal(size=shape).astype(np.float32)


@pytest.mark.parametrize("transA,transB", [(False, False), (False, True),
                                           (True, False), (True, True)])
def test_gemm_trans(cpu_ctx, transA, transB):
    M, N, K = 33, 17, 25
    A = randf(*(K, M) if transA else (M, K))
    B = randf(*(N, K) if transB else (K, N))
    ref = (A.T if transA else A) @ (B.T if transB else B)
    a, b = cpu_ctx.from_numpy(A), cpu_ctx.from_numpy(B)
    c = cpu_ctx.tensor((M, N))
    nz.gemm(a, b, c, transA, transB)
    np.testing.assert_allclose(c.numpy(), ref, rtol=1e-5, atol=1e-5)


def test_gemm_alpha_beta_bias_gelu(cpu_ctx):
    M, N, K = 12, 20, 16
    A, B, C0, bias = randf(M, K), randf(K, N), randf(M, N), randf(N)
    a, b = cpu_ctx.from_numpy(A), cpu_ctx.from_numpy(B)
    c = cpu_ctx.from_numpy(C0)
    bt = cpu_ctx.from_numpy(bias)
    pre = cpu_ctx.tensor((M, N))
    nz.gemm(a, b, c, alpha=0.5, beta=2.0, bias=bt, gelu=True, save_pre=pre)
    z_ref = 0.5 * (A @ B) + 2.0 * C0 + bias
    y_ref = torch.nn.functional.gelu(torch.from_numpy(z_ref), approximate="tanh").numpy()
    np.testing.assert_allclose(pre.numpy(), z_ref, rtol=1e-5, atol=1e-5)
    np.testing.assert_allclose(c.numpy(), y_ref, rtol=1e-5, atol=1e-5)


def test_gemm_batched(cpu_ctx):
    G, M, N, K = 3, 8, 10, 6
    A, B = randf(G, M, K), randf(G, K, N)
    ref = np.einsum("gmk,gkn->gmn", A, B)
    a, b = cpu_ctx.from_numpy(A), cpu_ctx.from_numpy(B)
    c = cpu_ctx.tensor((G, M, N))
    nz.gemm_batched(a, b, c, False, False, 1.0, G, M, N, K, M * K, K * N, M * N)
    np.testing.assert_allclose(c.numpy(), ref, rtol=1e-5, atol=1e-5)


def test_bias_gelu_fwd_bwd(cpu_ctx):
    M, N = 24, 32
    X, bias, dY = randf(M, N), randf(N), randf(M, N)
    xt = torch.from_numpy(X).requires_grad_(True)
    bt = torch.from_numpy(bias).requires_grad_(True)
    yt = torch.nn.functional.gelu(xt + bt, approximate="tanh")
    yt.backward(torch.from_numpy(dY))

    x, b = cpu_ctx.from_numpy(X), cpu_ctx.from_numpy(bias)
    y, z = cpu_ctx.tensor((M, N)), cpu_ctx.tensor((M, N))
    nz.bias_gelu_fwd(x, b, y, z)
    np.testing.assert_allclose(y.numpy(), yt.detach().numpy(), rtol=1e-5, atol=1e-5)

    dy = cpu_ctx.from_numpy(dY)
    dz, db = cpu_ctx.tensor((M, N)), cpu_ctx.tensor((N,))
    nz.bias_gelu_bwd(dy, z, dz, db)
    np.testing.assert_allclose(dz.numpy(), xt.grad.numpy(), rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(db.numpy(), bt.grad.numpy(), rtol=1e-4, atol=1e-4)


def test_layernorm_fwd_bwd(cpu_ctx):
    R, N = 20, 48
    X, G, Be, dY = randf(R, N), randf(N), randf(N), randf(R, N)
    xt = torch.from_numpy(X).requires_grad_(True)
    gt = torch.from_numpy(G).requires_grad_(True)
    bt = torch.from_numpy(Be).requires_grad_(True)
    yt = torch.nn.functional.layer_norm(xt, (N,), gt, bt, eps=1e-5)
    yt.backward(torch.from_numpy(dY))

    x, g, b = cpu_ctx.from_numpy(X), cpu_ctx.from_numpy(G), cpu_ctx.from_numpy(Be)
    y = cpu_ctx.tensor((R, N))
    mean, rstd = cpu_ctx.tensor((R,)), cpu_ctx.tensor((R,))
    nz.layernorm_fwd(x, g, b, y, mean, rstd)
    np.testing.assert_allclose(y.numpy(), yt.detach().numpy(), rtol=1e-4, atol=1e-5)

    dy = cpu_ctx.from_numpy(dY)
    dx, dg, db = cpu_ctx.tensor((R, N)), cpu_ctx.tensor((N,)), cpu_ctx.tensor((N,))
    nz.layernorm_bwd(dy, x, g, mean, rstd, dx, dg, db)
    np.testing.assert_allclose(dx.numpy(), xt.grad.numpy(), rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(dg.numpy(), gt.grad.numpy(), rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(db.numpy(), bt.grad.numpy(), rtol=1e-4, atol=1e-4)


def test_softmax_fwd_bwd(cpu_ctx):
    R, N, scale = 14, 40, 0.125
    X, dY = randf(R, N), randf(R, N)
    xt = torch.from_numpy(X).requires_grad_(True)
    yt = torch.softmax(scale * xt, dim=-1)
    yt.backward(torch.from_numpy(dY))

    x = cpu_ctx.from_numpy(X)
    y = cpu_ctx.tensor((R, N))
    nz.softmax_fwd(x, y, scale)
    np.testing.assert_allclose(y.numpy(), yt.detach().numpy(), rtol=1e-5, atol=1e-6)

    dy = cpu_ctx.from_numpy(dY)
    dx = cpu_ctx.tensor((R, N))
    nz.softmax_bwd(dy, y, dx, scale)
    np.testing.assert_allclose(dx.numpy(), xt.grad.numpy(), rtol=1e-4, atol=1e-5)


def test_adam_step(cpu_ctx):
    n = 1000
    P, Gr = randf(n), randf(n)
    lr, b1, b2, eps = 1e-3, 0.9, 0.999, 1e-8
    # torch reference
    pt = torch.from_numpy(P.copy()).requires_grad_(True)
    opt = torch.optim.Adam([pt], lr=lr, betas=(b1, b2), eps=eps)
    pt.grad = torch.from_numpy(Gr.copy())
    opt.step()

    p = cpu_ctx.from_numpy(P.copy())
    g = cpu_ctx.from_numpy(Gr.copy())
    master = cpu_ctx.from_numpy(P.copy())
    m = cpu_ctx.tensor((n,)).fill(0)
    v = cpu_ctx.tensor((n,)).fill(0)
    nz.adam_step(p, g, master, m, v, lr, b1, b2, eps, step=1)
    np.testing.assert_allclose(p.numpy(), pt.detach().numpy(), rtol=1e-5, atol=1e-7)


def test_sgd_step(cpu_ctx):
    n = 100
    P, Gr = randf(n), randf(n)
    p = cpu_ctx.from_numpy(P.copy())
    g = cpu_ctx.from_numpy(Gr.copy())
    master = cpu_ctx.from_numpy(P.copy())
    nz.sgd_step(p, g, master, None, lr=0.1)
    np.testing.assert_allclose(p.numpy(), P - 0.1 * Gr, rtol=1e-6, atol=1e-7)


def test_mse_loss(cpu_ctx):
    Y, T = randf(8, 16), randf(8, 16)
    y, t = cpu_ctx.from_numpy(Y), cpu_ctx.from_numpy(T)
    loss = cpu_ctx.tensor((1,))
    dy = cpu_ctx.tensor((8, 16))
    nz.mse_loss(y, t, loss, dy)
    assert abs(loss.numpy()[0] - np.mean((Y - T) ** 2)) < 1e-6
    np.testing.assert_allclose(dy.numpy(), 2 * (Y - T) / Y.size, rtol=1e-5, atol=1e-8)


def test_ce_loss(cpu_ctx):
    R, N = 12, 10
    L = randf(R, N)
    labels = RNG.integers(0, N, size=R).astype(np.float32)
    lt = torch.from_numpy(L).requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lt, torch.from_numpy(labels).long())
    ref.backward()
    lo = cpu_ctx.from_numpy(L)
    lab = cpu_ctx.from_numpy(labels)
    loss = cpu_ctx.tensor((1,))
    dl = cpu_ctx.tensor((R, N))
    nz.ce_loss(lo, lab, loss, dl)
    assert abs(loss.numpy()[0] - ref.item()) < 1e-5
    np.testing.assert_allclose(dl.numpy(), lt.grad.numpy(), rtol=1e-4, atol=1e-6)


def test_colsum_transpose_reduce(cpu_ctx):
    X = randf(9, 21)
    x = cpu_ctx.from_numpy(X)
    cs = cpu_ctx.tensor((21,))
    nz.colsum(x, cs)
    np.testing.assert_allclose(cs.numpy(), X.sum(0), rtol=1e-5, atol=1e-5)
    tr = cpu_ctx.tensor((21, 9))
    nz.transpose2d(x, tr)
    np.testing.assert_array_equal(tr.numpy(), X.T)
    s = cpu_ctx.tensor((1,))
    nz.reduce_sum(x, s)
    assert abs(s.numpy()[0] - X.sum()) < 1e-4
    nz.l2norm_sq(x, s)
    assert abs(s.numpy()[0] - (X ** 2).sum()) < 1e-3


def test_attn_split_merge(cpu_ctx):
    B, S, H, D = 2, 6, 4, 8
    QKV = randf(B, S, 3 * H * D)
    qkv = cpu_ctx.from_numpy(QKV.reshape(B * S, 3 * H * D))
    q = cpu_ctx.tensor((B, H, S, D))
    k = cpu_ctx.tensor((B, H, S, D))
    v = cpu_ctx.tensor((B, H, S, D))
    nz.attn_split_qkv(qkv, q, k, v, B, S, H, D)
    ref = QKV.reshape(B, S, 3, H, D).transpose(2, 0, 3, 1, 4)  # [3,B,H,S,D]
    np.testing.assert_array_equal(q.numpy(), ref[0])
    np.testing.assert_array_equal(k.numpy(), ref[1])
    np.testing.assert_array_equal(v.numpy(), ref[2])
    out = cpu_ctx.tensor((B * S, H * D))
    nz.attn_merge(q, out, B, S, H, D)
    merged_ref = ref[0].transpose(0, 2, 1, 3).reshape(B * S, H * D)
    np.testing.assert_array_equal(out.numpy(), merged_ref)


def test_dropout(cpu_ctx):
    n = 10000
    X = randf(n)
    x = cpu_ctx.from_numpy(X)
    y, mask = cpu_ctx.tensor((n,)), cpu_ctx.tensor((n,))
    nz.dropout_fwd(x, y, mask, 0.3, seed=11)
    m = mask.numpy()
    assert abs(m.mean() - 0.7) < 0.03
    np.testing.assert_allclose(y.numpy(), X * m / 0.7, rtol=1e-6, atol=1e-7)
    dy = cpu_ctx.from_numpy(X)
    dx = cpu_ctx.tensor((n,))
    nz.dropout_bwd(dy, mask, dx, 0.3)
    np.testing.assert_allclose(dx.numpy(), X * m / 0.7, rtol=1e-6, atol=1e-7)
    # p=0 is identity with all-ones mask
    nz.dropout_fwd(x, y, mask, 0.0, seed=11)
    np.testing.assert_allclose(y.numpy(), X, rtol=1e-6)


def test_attn_packed_cpu_fallback():
    """ops::attn_fwd_packed / attn_bwd_packed CPU fallback vs a manual
    torch composition of the same math (f32)."""
    import torch
    B, S, H, D = 2, 16, 2, 8
    scale = 1.0 / np.sqrt(D)
    rng = np.random.default_rng(3)
    QKV = rng.normal(size=(B * S, 3 * H * D)).astype(np.float32)
    DM = rng.normal(size=(B * S, H * D)).astype(np.float32)
    ctx = nz.Ctx(nz.CPU)
    qkv = ctx.from_numpy(QKV)
    probs = ctx.tensor((B * H * S, S))
    out = ctx.tensor((B * S, H * D))
    nz.attn_fwd_packed(qkv, probs, out, B, S, H, D, scale)
    # torch reference
    qkv_t = torch.from_numpy(QKV).reshape(B, S, 3, H, D).permute(2, 0, 3, 1, 4)
    q, k, v = qkv_t[0], qkv_t[1], qkv_t[2]  # [B,H,S,D]
    q = q.clone().requires_grad_(True)
    k = k.clone().requires_grad_(True)
    v = v.clone().requires_grad_(True)
    p_ref = torch.softmax(scale * (q @ k.transpose(-1, -2)), dim=-1)
    o_ref = p_ref @ v
    merged_ref = o_ref.permute(0, 2, 1, 3).reshape(B * S, H * D)
    np.testing.assert_allclose(out.numpy(), merged_ref.detach().numpy(),
                               rtol=1e-3, atol=1e-4)
    dm_t = torch.from_numpy(DM)
    merged_ref.backward(dm_t)
    dqkv = ctx.tensor((B * S, 3 * H * D))
    dm = ctx.from_numpy(DM)
    nz.attn_bwd_packed(qkv, probs, dm, dqkv, B, S, H, D, scale)
    dq_ref = torch.stack([q.grad, k.grad, v.grad])  # [3,B,H,S,D]
    dqkv_ref = dq_ref.permute(1, 3, 0, 2, 4).reshape(B * S, 3 * H * D)
    np.testing.assert_allclose(dqkv.numpy(), dqkv_ref.numpy(), rtol=1e-3,
                               atol=1e-4)


def test_adam_weight_decay(cpu_ctx):
    """wd is classic Adam+L2 (g += wd*p), matching torch.optim.Adam's
    weight_decay (NOT AdamW's decoupled form)."""
    n = 512
    P, Gr = randf(n), randf(n)
    lr, b1, b2, eps, wd = 1e-3, 0.9, 0.999, 1e-8, 0.05
    pt = torch.from_numpy(P.copy()).requires_grad_(True)
    opt = torch.optim.Adam([pt], lr=lr, betas=(b1, b2), eps=eps,
                           weight_decay=wd)
    for i in range(3):
        pt.grad = torch.from_numpy(Gr.copy())
        opt.step()

    p = cpu_ctx.from_numpy(P.copy())
    g = cpu_ctx.from_numpy(Gr.copy())
    master = cpu_ctx.from_numpy(P.copy())
    m = cpu_ctx.tensor((n,)).fill(0)
    v = cpu_ctx.tensor((n,)).fill(0)
    for i in range(3):
        nz.adam_step(p, g, master, m, v, lr, b1, b2, eps, step=i + 1,
                     weight_decay=wd)
    np.testing.assert_allclose(p.numpy(), pt.detach().numpy(), rtol=1e-5,
                               atol=1e-6)


def test_sgd_momentum(cpu_ctx):
    """SGD momentum buffer matches torch.optim.SGD (v = mu*v + g; p -= lr*v)."""
    n = 256
    P, Gr = randf(n), randf(n)
    lr, mu = 0.1, 0.9
    pt = torch.from_numpy(P.copy()).requires_grad_(True)
    opt = torch.optim.SGD([pt], lr=lr, momentum=mu)
    for _ in range(3):
        pt.grad = torch.from_numpy(Gr.copy())
        opt.step()
    p = cpu_ctx.from_numpy(P.copy())
    g = cpu_ctx.from_numpy(Gr.copy())
    master = cpu_ctx.from_numpy(P.copy())
    mom = cpu_ctx.tensor((n,)).fill(0)
    for _ in range(3):
        nz.sgd_step(p, g, master, mom, lr=lr, mu=mu)
    np.testing.assert_allclose(p.numpy(), pt.detach().numpy(), rtol=1e-6,
                               atol=1e-7)
