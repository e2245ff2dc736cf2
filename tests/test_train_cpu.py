"""T1/T2: config C10 end-to-end (2-layer 128-dim f32 MLP trains on CPU,
no GPU — BASELINE.json:7) plus an end-to-end numeric replication of one
training step against a pure-numpy/torch reference reconstructed from the
checkpoint (validates tape + GEMM + bias/GELU + MSE + Adam together)."""
import os

import numpy as np
import pytest
import torch

import nezha_amd as nz


def splitmix64(x):
    mask = np.uint64(0xFFFFFFFFFFFFFFFF)
    x = np.uint64(x)
    x = (x + np.uint64(0x9E3779B97F4A7C15)) & mask
    x = ((x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)) & mask
    x = ((x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)) & mask
    return x ^ (x >> np.uint64(31))


def rng_normal_np(seed, n):
    """Replicates nz::rng_normal (src/core/common.h) for indices 0..n-1."""
    idx = np.arange(n, dtype=np.uint64)
    h1 = splitmix64(np.uint64(seed) ^ splitmix64(idx * np.uint64(2)))
    h2 = splitmix64(np.uint64(seed) ^ splitmix64(idx * np.uint64(2) + np.uint64(1)))
    u1 = ((h1 >> np.uint64(40)).astype(np.float32) + 0.5) / 16777216.0
    u2 = (h2 >> np.uint64(40)).astype(np.float32) / 16777216.0
    return (np.sqrt(-2 * np.log(u1)) * np.cos(2 * np.float32(np.pi) * u2)).astype(
        np.float32)


def test_c10_loss_decreases(cpu_ctx):
    tr = nz.Trainer.mlp(cpu_ctx, layers=2, dim=128, batch=64, lr=1e-3, seed=7)
    losses = []
    for _ in range(30):
        tr.step()
        losses.append(tr.loss())
    assert losses[-1] < 0.75 * losses[0], f"loss did not decrease: {losses}"
    assert np.isfinite(losses).all()


def test_sgd_training_runs(cpu_ctx):
    tr = nz.Trainer.mlp(cpu_ctx, layers=2, dim=64, batch=32, optim="sgd",
                        lr=1e-2, seed=9)
    tr.step()
    l0 = tr.loss()
    for _ in range(20):
        tr.step()
    assert tr.loss() < l0


def test_one_step_matches_reference(cpu_ctx, tmp_path):
    """Save ckpt -> numpy-reconstruct one fwd/bwd/Adam step with torch ->
    compare against the engine's own post-step checkpoint."""
    layers, dim, batch, lr, seed = 2, 32, 16, 1e-3, 55
    tr = nz.Trainer.mlp(cpu_ctx, layers=layers, dim=dim, batch=batch, lr=lr,
                        seed=seed)
    p0 = str(tmp_path / "step0.npz")
    p1 = str(tmp_path / "step1.npz")
    tr.save(p0)
    tr.step()
    tr.save(p1)

    ck0, ck1 = np.load(p0), np.load(p1)
    data_seed = seed + 12345
    x = torch.from_numpy(rng_normal_np(data_seed, batch * dim).reshape(batch, dim))
    target = torch.from_numpy(
        rng_normal_np(data_seed + 1, batch * dim).reshape(batch, dim))

    params = []
    h = x
    for i in range(layers):
        w = torch.from_numpy(ck0[f"param/mlp.l{i}.w"].copy()).requires_grad_(True)
        b = torch.from_numpy(ck0[f"param/mlp.l{i}.b"].copy()).requires_grad_(True)
        params.append((w, b))
        h = torch.nn.functional.gelu(h @ w + b, approximate="tanh")
    loss = torch.nn.functional.mse_loss(h, target)
    opt = torch.optim.Adam([p for wb in params for p in wb], lr=lr,
                           betas=(0.9, 0.999), eps=1e-8)
    loss.backward()
    opt.step()

    for i in range(layers):
        np.testing.assert_allclose(ck1[f"param/mlp.l{i}.w"],
                                   params[i][0].detach().numpy(), rtol=2e-4,
                                   atol=2e-6)
        np.testing.assert_allclose(ck1[f"param/mlp.l{i}.b"],
                                   params[i][1].detach().numpy(), rtol=2e-4,
                                   atol=2e-6)
    # loss value itself
    assert abs(tr.loss() - loss.item()) / loss.item() < 1e-4


def test_bert_cpu_smoke(cpu_ctx):
    """Tiny BERT encoder trains on CPU f32 (covers MHA/LN/residual tape)."""
    tr = nz.Trainer.bert(cpu_ctx, layers=2, hidden=32, ffn=64, heads=4, seq=8,
                         batch=4, lr=1e-3, seed=3)
    losses = []
    for _ in range(10):
        tr.step()
        losses.append(tr.loss())
    assert np.isfinite(losses).all()
    assert losses[-1] < losses[0]


def test_bert_one_step_matches_torch(cpu_ctx, tmp_path):
    """Full BERT encoder layer fwd/bwd/Adam vs a torch reconstruction."""
    L, H, F, heads, S, B, lr, seed = 1, 16, 32, 2, 4, 2, 1e-3, 77
    tr = nz.Trainer.bert(cpu_ctx, layers=L, hidden=H, ffn=F, heads=heads,
                         seq=S, batch=B, lr=lr, seed=seed)
    p0, p1 = str(tmp_path / "b0.npz"), str(tmp_path / "b1.npz")
    tr.save(p0)
    tr.step()
    engine_loss = tr.loss()
    tr.save(p1)
    ck0, ck1 = np.load(p0), np.load(p1)

    data_seed = seed + 12345
    x = torch.from_numpy(rng_normal_np(data_seed, B * S * H).reshape(B * S, H))
    tgt = torch.from_numpy(rng_normal_np(data_seed + 1, B * S * H).reshape(B * S, H))

    def P(name):
        return torch.from_numpy(ck0[name].copy()).requires_grad_(True)

    names = []
    tensors = {}
    for key in ck0.files:
        if key.startswith("param/"):
            nm = key[len("param/"):]
            names.append(nm)
            tensors[nm] = P(key)

    pre = "bert.l0"
    wqkv, bqkv = tensors[f"{pre}.mha.qkv.w"], tensors[f"{pre}.mha.qkv.b"]
    wo, bo = tensors[f"{pre}.mha.out.w"], tensors[f"{pre}.mha.out.b"]
    g1, be1 = tensors[f"{pre}.ln1.gamma"], tensors[f"{pre}.ln1.beta"]
    w1, b1 = tensors[f"{pre}.ffn1.w"], tensors[f"{pre}.ffn1.b"]
    w2, b2 = tensors[f"{pre}.ffn2.w"], tensors[f"{pre}.ffn2.b"]
    g2, be2 = tensors[f"{pre}.ln2.gamma"], tensors[f"{pre}.ln2.beta"]

    dh = H // heads
    qkv = (x @ wqkv + bqkv).reshape(B, S, 3, heads, dh).permute(2, 0, 3, 1, 4)
    q, k, v = qkv[0], qkv[1], qkv[2]
    att = torch.softmax((q @ k.transpose(-1, -2)) / np.sqrt(dh), dim=-1)
    ctx_t = (att @ v).permute(0, 2, 1, 3).reshape(B * S, H)
    attn_out = ctx_t @ wo + bo
    h1 = torch.nn.functional.layer_norm(x + attn_out, (H,), g1, be1, eps=1e-5)
    f = torch.nn.functional.gelu(h1 @ w1 + b1, approximate="tanh") @ w2 + b2
    out = torch.nn.functional.layer_norm(h1 + f, (H,), g2, be2, eps=1e-5)
    loss = torch.nn.functional.mse_loss(out, tgt)

    opt = torch.optim.Adam([tensors[nm] for nm in names], lr=lr)
    loss.backward()
    opt.step()

    assert abs(engine_loss - loss.item()) / loss.item() < 1e-3
    for nm in names:
        np.testing.assert_allclose(ck1[f"param/{nm}"],
                                   tensors[nm].detach().numpy(), rtol=5e-3,
                                   atol=3e-4, err_msg=nm)


def test_streaming_data_mode(tmp_path):
    """NEZHA_DATA=stream regenerates a fresh deterministic batch per step;
    two runs are identical, and losses differ step to step (new data)."""
    import subprocess, sys, os, json
    script = r"""
import os, sys, json
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.CPU)
tr = nz.Trainer.mlp(ctx, layers=2, dim=64, batch=32, lr=1e-3, seed=5)
losses = []
for _ in range(5):
    tr.step()
    losses.append(tr.loss())
print(json.dumps(losses))
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = []
    for _ in range(2):
        env = dict(os.environ, NEZHA_DATA="stream")
        r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                           capture_output=True, timeout=120)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
        outs.append(json.loads(r.stdout.decode().strip().splitlines()[-1]))
    assert outs[0] == outs[1]  # deterministic
    assert len(set(round(x, 6) for x in outs[0])) > 2  # data actually varies


def test_metrics_jsonl_valid(tmp_path):
    """NEZHA_METRICS per-step records must be valid JSONL (an unescaped
    embedded config string once produced malformed lines)."""
    import subprocess, sys
    mpath = tmp_path / "m.jsonl"
    script = """
import os, sys
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.CPU)
tr = nz.Trainer.mlp(ctx, layers=2, dim=64, batch=32, lr=1e-3, seed=5)
for _ in range(3):
    tr.step()
    tr.loss()
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, NEZHA_METRICS=str(mpath))
    r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                       capture_output=True, timeout=120)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
    lines = mpath.read_text().strip().splitlines()
    assert len(lines) >= 3
    import json as _json
    for ln in lines:
        rec = _json.loads(ln)
        assert "loss" in rec and "step" in rec


def test_bert_exotic_configs_cpu():
    """Non-contract geometries (dh!=64, odd seq, ragged ffn) train on the
    composed CPU path — mirrors the GPU fuzz in profiles/fuzz_models2.log."""
    ctx = nz.Ctx(nz.CPU)
    for kw in [dict(hidden=128, ffn=256, heads=4, seq=24),     # dh=32
               dict(hidden=96, ffn=100, heads=2, seq=10)]:     # dh=48, ragged
        tr = nz.Trainer.bert(ctx, layers=1, batch=2, lr=1e-3, seed=5, **kw)
        tr.step()
        tr.step()
        assert np.isfinite(tr.loss())
