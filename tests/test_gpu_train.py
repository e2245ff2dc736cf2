"""T5/T6: on-GPU integration: training runs, checkpointing,
RCCL self-test, CPU-vs-GPU end-to-end agreement."""
import os

import numpy as np
import pytest

import nezha_amd as nz

pytestmark = pytest.mark.gpu


def test_mlp_training_loss_decreases(hip_ctx):
    tr = nz.Trainer.mlp(hip_ctx, layers=4, dim=1024, batch=2048, lr=1e-3,
                        seed=17)
    tr.step()
    l0 = tr.loss()
    for _ in range(29):
        tr.step()
    l1 = tr.loss()
    assert np.isfinite(l0) and np.isfinite(l1)
    assert l1 < 0.9 * l0, f"{l0} -> {l1}"


def test_flagship_shape_steps(hip_ctx):
    """C11: the full 4x4096 MLP at batch 8192 steps and stays finite."""
    tr = nz.Trainer.mlp(hip_ctx, layers=4, dim=4096, batch=8192, lr=1e-4,
                        seed=1234)
    for _ in range(3):
        tr.step()
    assert np.isfinite(tr.loss())


def test_bert_training(hip_ctx):
    tr = nz.Trainer.bert(hip_ctx, layers=2, hidden=256, ffn=1024, heads=4,
                         seq=128, batch=8, lr=1e-4, seed=5)
    tr.step()
    l0 = tr.loss()
    for _ in range(19):
        tr.step()
    assert np.isfinite(l0) and tr.loss() < l0


def test_bert_large_config(hip_ctx):
    """BERT-large geometry (24x1024, 16 heads): exercises the N=1024
    wave-LayerNorm path and the fused attention kernels at a different
    head count than the flagship config."""
    tr = nz.Trainer.bert(hip_ctx, layers=24, hidden=1024, ffn=4096, heads=16,
                         seq=128, batch=32, lr=1e-4, seed=9)
    tr.step()
    l0 = tr.loss()
    for _ in range(4):
        tr.step()
    assert np.isfinite(l0) and np.isfinite(tr.loss())


def test_cpu_gpu_agreement():
    """Same model/seed/data on CPU f32 and GPU bf16: losses agree to bf16
    tolerance after the first step (validates the whole GPU path E2E)."""
    cpu = nz.Ctx(nz.CPU)
    gpu = nz.Ctx(nz.HIP, 0)
    kw = dict(layers=2, dim=256, batch=128, lr=1e-3, seed=33)
    a = nz.Trainer.mlp(cpu, **kw)
    b = nz.Trainer.mlp(gpu, **kw)
    a.step()
    b.step()
    la, lb = a.loss(), b.loss()
    assert abs(la - lb) / la < 0.05, f"cpu {la} vs gpu {lb}"


def test_gpu_ckpt_roundtrip(hip_ctx, tmp_path):
    tr = nz.Trainer.mlp(hip_ctx, layers=2, dim=512, batch=256, lr=1e-3, seed=3)
    for _ in range(2):
        tr.step()
    p = str(tmp_path / "gpu.npz")
    tr.save(p)
    ck = np.load(p)
    assert ck["param/mlp.l0.w"].dtype == np.uint16  # bf16 bits on GPU
    # load into a fresh trainer, step both, losses must match exactly
    t2 = nz.Trainer.mlp(hip_ctx, layers=2, dim=512, batch=256, lr=1e-3, seed=3)
    t2.load(p)
    tr.step()
    t2.step()
    assert tr.loss() == t2.loss()


def test_rccl_selftest(hip_ctx):
    """C5 plumbing: single-rank RCCL init + allreduce + broadcast."""
    uid = nz.Comm.get_uid()
    comm = nz.Comm.rccl(hip_ctx, 1, 0, uid)
    assert comm.nranks == 1 and comm.rank == 0
    assert comm.allreduce_scalar(3.5) == 3.5
    t = hip_ctx.tensor((1024,), nz.BF16).fill(2.0)
    comm.allreduce_tensor(hip_ctx, t)
    assert np.all(t.numpy() == 2.0)
    comm.broadcast_tensor(hip_ctx, t, 0)
    comm.barrier()


def test_ddp_single_rank_equivalence(hip_ctx, tmp_path):
    """DDP with world=1 must be identical to no DDP (hooks, buckets, sync)."""
    kw = dict(layers=2, dim=512, batch=256, lr=1e-3, seed=8)
    a = nz.Trainer.mlp(hip_ctx, **kw)
    for _ in range(3):
        a.step()
    pa = str(tmp_path / "a.npz")
    a.save(pa)

    b = nz.Trainer.mlp(hip_ctx, **kw)
    uid = nz.Comm.get_uid()
    comm = nz.Comm.rccl(hip_ctx, 1, 0, uid)
    b.attach_ddp(comm)
    for _ in range(3):
        b.step()
    pb = str(tmp_path / "b.npz")
    b.save(pb)
    ca, cb = np.load(pa), np.load(pb)
    for k in ca.files:
        np.testing.assert_array_equal(ca[k], cb[k], err_msg=k)


def test_graph_capture_matches_eager(tmp_path):
    """NEZHA_GRAPH=1 replays the captured step; losses must match the eager
    path exactly (same kernels, same order, bias corrections via device
    buffer). Run in subprocesses since the env is read at trainer ctor."""
    import subprocess, sys, os, json
    script = r"""
import os, sys, json
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
tr = nz.Trainer.mlp(ctx, layers=2, dim=512, batch=256, lr=1e-3, seed=44)
losses = []
for _ in range(6):
    tr.step()
    losses.append(tr.loss())
print(json.dumps(losses))
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = []
    for graph in ("0", "1"):
        env = dict(os.environ, NEZHA_GRAPH=graph)
        r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                           capture_output=True, timeout=300)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
        outs.append(json.loads(r.stdout.decode().strip().splitlines()[-1]))
    assert outs[0] == outs[1], f"eager {outs[0]} vs graph {outs[1]}"


def test_bert_cpu_gpu_agreement():
    """BERT with the fused attention path (GPU) vs the composed CPU oracle:
    first-step losses agree to bf16 tolerance."""
    cpu = nz.Ctx(nz.CPU)
    gpu = nz.Ctx(nz.HIP, 0)
    kw = dict(layers=2, hidden=256, ffn=512, heads=4, seq=128, batch=4,
              lr=1e-3, seed=66)
    a = nz.Trainer.bert(cpu, **kw)
    b = nz.Trainer.bert(gpu, **kw)
    a.step()
    b.step()
    la, lb = a.loss(), b.loss()
    assert abs(la - lb) / la < 0.05, f"cpu {la} vs gpu {lb}"


def test_gpu_resume_bitwise(hip_ctx, tmp_path):
    """Resume on GPU is bitwise: snapshot at step 2, run 3 more on both the
    original and a resumed trainer, compare full checkpoints exactly."""
    kw = dict(layers=2, dim=512, batch=256, lr=1e-3, seed=12)
    a = nz.Trainer.mlp(hip_ctx, **kw)
    for _ in range(2):
        a.step()
    snap = str(tmp_path / "snap.npz")
    a.save(snap)
    for _ in range(3):
        a.step()
    fa = str(tmp_path / "fa.npz")
    a.save(fa)

    b = nz.Trainer.mlp(hip_ctx, **kw)
    b.load(snap)
    for _ in range(3):
        b.step()
    fb = str(tmp_path / "fb.npz")
    b.save(fb)
    ca, cb = np.load(fa), np.load(fb)
    for k in ca.files:
        np.testing.assert_array_equal(ca[k], cb[k], err_msg=k)


def test_ddp_8rank_one_device():
    """8 simulated DP ranks (threads, own streams) on one GPU via the
    device-staging FakeComm — real event/comm-stream/chaining path."""
    import subprocess, os
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    binp = os.path.join(repo, "build", "test_ddp_fake_gpu")
    r = subprocess.run([binp], capture_output=True, timeout=600)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


def test_cross_device_resume(hip_ctx, tmp_path):
    """CPU-saved (f32) checkpoint resumes on GPU (bf16) and vice versa
    (VERDICT r01 missing #5): params are converted, exact values restored
    from the f32 master, training continues finite and comparable."""
    kw = dict(layers=2, dim=256, batch=128, lr=1e-3, seed=21)
    cpu = nz.Ctx(nz.CPU)
    a = nz.Trainer.mlp(cpu, **kw)
    for _ in range(2):
        a.step()
    p = str(tmp_path / "cpu.npz")
    a.save(p)
    ck = np.load(p)
    assert ck["param/mlp.l0.w"].dtype == np.float32

    b = nz.Trainer.mlp(hip_ctx, **kw)
    b.load(p)
    # bf16 params must equal round(master) of the CPU state
    q = str(tmp_path / "gpu_after_load.npz")
    b.save(q)
    cg = np.load(q)
    np.testing.assert_array_equal(cg["master/mlp.l0.w"],
                                  ck["master/mlp.l0.w"])
    np.testing.assert_array_equal(
        cg["param/mlp.l0.w"],
        nz.f32_to_bf16(ck["master/mlp.l0.w"]))
    a.step()
    b.step()
    la, lb = a.loss(), b.loss()
    assert np.isfinite(lb) and abs(la - lb) / la < 0.05, f"{la} vs {lb}"

    # and back: GPU (bf16) checkpoint onto CPU (f32)
    r = str(tmp_path / "gpu.npz")
    b.save(r)
    c = nz.Trainer.mlp(cpu, **kw)
    c.load(r)
    c.step()
    assert np.isfinite(c.loss())


def test_graph_metrics_coexist(tmp_path):
    """NEZHA_GRAPH=1 + NEZHA_METRICS both set must not abort capture
    (regression: metrics loss readback used to sync inside capture)."""
    import subprocess, sys, os
    script = r"""
import os, sys
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
tr = nz.Trainer.mlp(ctx, layers=2, dim=512, batch=256, lr=1e-3, seed=44)
for _ in range(6):
    tr.step()
assert tr.loss() > 0
print("GRAPH_METRICS_OK")
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, NEZHA_GRAPH="1",
               NEZHA_METRICS=str(tmp_path / "m.jsonl"))
    r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                       capture_output=True, timeout=300)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
    assert b"GRAPH_METRICS_OK" in r.stdout


def test_graph_odd_bucket_matches_eager():
    """Graph replay with a bucket whose numel % 4 != 0 exercises the scalar
    adam_k bc_dev path (regression: bias correction used to freeze at the
    capture-time step). dim=125, bucket_bytes=32000 -> bucket numels
    15875 and 15625, both odd."""
    import subprocess, sys, os, json
    script = r"""
import os, sys, json
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
tr = nz.Trainer.mlp(ctx, layers=2, dim=125, batch=64, lr=1e-2, seed=7,
                    bucket_bytes=32000)
losses = []
for _ in range(8):
    tr.step()
    losses.append(tr.loss())
print(json.dumps(losses))
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = []
    for graph in ("0", "1"):
        env = dict(os.environ, NEZHA_GRAPH=graph)
        r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                           capture_output=True, timeout=300)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
        outs.append(json.loads(r.stdout.decode().strip().splitlines()[-1]))
    assert outs[0] == outs[1], f"eager {outs[0]} vs graph {outs[1]}"


def test_graph_ddp_world1_matches_eager():
    """NEZHA_GRAPH=1 with DDP attached (RCCL world=1): the captured step
    includes the bucketed all-reduce; losses must match the eager DDP path
    exactly. On a 1-GPU box world=1 is the
    deepest RCCL-in-graph check available."""
    import subprocess, sys, os, json
    script = r"""
import os, sys, json
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
tr = nz.Trainer.mlp(ctx, layers=2, dim=512, batch=256, lr=1e-3, seed=44,
                    bucket_bytes=200000)
uid = nz.Comm.get_uid()
comm = nz.Comm.rccl(ctx, 1, 0, uid)
tr.attach_ddp(comm)
losses = []
for _ in range(6):
    tr.step()
    losses.append(tr.loss())
print(json.dumps(losses))
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = []
    for graph in ("0", "1"):
        env = dict(os.environ, NEZHA_GRAPH=graph)
        r = subprocess.run([sys.executable, "-c", script], env=env, cwd=repo,
                           capture_output=True, timeout=300)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
        # RCCL prints a version banner to stdout — take the JSON array line
        lines = [l for l in r.stdout.decode().splitlines()
                 if l.startswith("[")]
        outs.append(json.loads(lines[-1]))
    assert outs[0] == outs[1], f"eager {outs[0]} vs graph {outs[1]}"


def test_attn_fused_s512_matches_composed(hip_ctx):
    """S=512 tiled fused attention forward (VERDICT r01 missing #3) vs the
    composed batched-GEMM + softmax reference built from the same
    primitives, same inputs. Checks both the merged output and the
    normalized probs the composed backward consumes."""
    B, S, H, dh = 2, 512, 4, 64
    hidden = H * dh
    scale = 1.0 / (dh ** 0.5)
    qkv = hip_ctx.tensor((B * S, 3 * hidden), nz.BF16).rand_normal(7)
    probs = hip_ctx.tensor((B * H * S, S), nz.BF16)
    out = hip_ctx.tensor((B * S, hidden), nz.BF16)
    nz.attn_fwd_packed(qkv, probs, out, B, S, H, dh, scale)

    # composed reference from the same primitive kernels
    q = hip_ctx.tensor((B, H, S, dh), nz.BF16)
    k = hip_ctx.tensor((B, H, S, dh), nz.BF16)
    v = hip_ctx.tensor((B, H, S, dh), nz.BF16)
    nz.attn_split_qkv(qkv, q, k, v, B, S, H, dh)
    scores = hip_ctx.tensor((B * H * S, S), nz.BF16)
    nz.gemm_batched(q, k, scores, False, True, 1.0, B * H, S, S, dh,
                    S * dh, S * dh, S * S)
    probs_ref = hip_ctx.tensor((B * H * S, S), nz.BF16)
    nz.softmax_fwd(scores, probs_ref, scale)
    ctxt = hip_ctx.tensor((B, H, S, dh), nz.BF16)
    nz.gemm_batched(probs_ref, v, ctxt, False, False, 1.0, B * H, S, dh, S,
                    S * S, S * dh, S * dh)
    out_ref = hip_ctx.tensor((B * S, hidden), nz.BF16)
    nz.attn_merge(ctxt, out_ref, B, S, H, dh)

    o, orf = out.numpy(), out_ref.numpy()
    scale_o = np.abs(orf).max() + 1e-9
    assert np.abs(o - orf).max() / scale_o < 0.03, \
        f"out err {np.abs(o - orf).max() / scale_o}"
    p, prf = probs.numpy(), probs_ref.numpy()
    assert np.abs(p - prf).max() < 0.02, f"probs err {np.abs(p - prf).max()}"


def test_bert_s512_trains(hip_ctx):
    """BERT with S=512 routes the forward through the tiled fused kernel
    and the backward through the composed path; loss must decrease."""
    tr = nz.Trainer.bert(hip_ctx, layers=2, hidden=256, ffn=1024, heads=4,
                         seq=512, batch=4, lr=1e-4, seed=13)
    tr.step()
    l0 = tr.loss()
    for _ in range(9):
        tr.step()
    assert np.isfinite(l0) and tr.loss() < l0, f"{l0} -> {tr.loss()}"


def test_bert_noncontract_seq(hip_ctx):
    """seq=64 routes attention through the composed fallback on GPU
    (fused kernels are gated to the S=128 contract shape)."""
    tr = nz.Trainer.bert(hip_ctx, layers=2, hidden=256, ffn=1024, heads=4,
                         seq=64, batch=8, lr=1e-4, seed=11)
    tr.step()
    l0 = tr.loss()
    for _ in range(9):
        tr.step()
    assert np.isfinite(l0) and tr.loss() < l0


def test_attn_bwd_s512_matches_composed(tmp_path):
    """S=512 fused attention BACKWARD (two-kernel: q-block dQ + rowsum
    scratch, kv-block dK/dV with recomputed dP^T) vs the composed
    batched-GEMM + softmax_bwd fallback (NEZHA_ATTN512_FUSED=0), same
    inputs, full dqkv compared."""
    import subprocess, sys, os
    script = r"""
import os, sys
sys.path.insert(0, os.getcwd())
import numpy as np, nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
B, S, H, dh = 2, 512, 4, 64
hidden = H * dh
scale = 1.0 / dh ** 0.5
qkv = ctx.tensor((B * S, 3 * hidden), nz.BF16).rand_normal(17)
probs = ctx.tensor((B * H * S, S), nz.BF16)
out = ctx.tensor((B * S, hidden), nz.BF16)
nz.attn_fwd_packed(qkv, probs, out, B, S, H, dh, scale)
dmerged = ctx.tensor((B * S, hidden), nz.BF16).rand_normal(19)
dqkv = ctx.tensor((B * S, 3 * hidden), nz.BF16)
nz.attn_bwd_packed(qkv, probs, dmerged, dqkv, B, S, H, dh, scale)
np.save(sys.argv[1], dqkv.numpy())
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = {}
    for fused in ("1", "0"):
        pth = str(tmp_path / f"dqkv_{fused}.npy")
        env = dict(os.environ, NEZHA_ATTN512_FUSED=fused)
        r = subprocess.run([sys.executable, "-c", script, pth], env=env,
                           cwd=repo, capture_output=True, timeout=600)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
        outs[fused] = np.load(pth)
    got, ref = outs["1"], outs["0"]
    scale_g = np.abs(ref).max() + 1e-9
    err = np.abs(got - ref).max() / scale_g
    assert err < 0.04, f"dqkv rel err {err}"


def test_dgelu_fusion_matches_unfused(tmp_path):
    """Folding gelu' into the downstream dgrad GEMM epilogue
    (NEZHA_FUSE_DGELU, default on) must train to the same losses as the
    separate bias_gelu_bwd pass — both models, few steps, loose bf16
    rounding tolerance (the fused path multiplies before the bf16
    round of dx, the composed path after)."""
    import subprocess, sys, os, json
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for model in ("mlp", "bert"):
        losses = {}
        for fuse in ("1", "0"):
            env = dict(os.environ, NEZHA_FUSE_DGELU=fuse)
            r = subprocess.run(
                [sys.executable, os.path.join(repo, "bench.py"),
                 "--model", model, "--steps", "4", "--warmup", "1",
                 "--batch", "1024" if model == "mlp" else "64"],
                env=env, cwd=repo, capture_output=True, timeout=600)
            assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()
            line = [l for l in r.stdout.decode().splitlines()
                    if l.startswith("{")][-1]
            losses[fuse] = json.loads(line)["loss"]
        assert abs(losses["1"] - losses["0"]) < 5e-3, losses
