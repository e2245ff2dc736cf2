"""§5.3 failure detection: an injected rank death must fail the job fast
(no hang), and training resumes from checkpoint."""
import os
import subprocess
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, sys
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
rank = int(os.environ["R"])
ctx = nz.Ctx(nz.CPU)
tr = nz.Trainer.mlp(ctx, layers=2, dim=32, batch=8, lr=1e-3, seed=5)
comm = nz.Comm.tcp(2, rank, "127.0.0.1", int(os.environ["P"]))
tr.attach_ddp(comm)
for i in range(6):
    tr.step()
    if rank == 0:
        tr.save(os.environ["CKPT"])  # rank0 writes; rank1 hits barrier
    else:
        tr.save(os.environ["CKPT"])
print("rank", rank, "finished", tr.loss())
"""


def test_rank_death_fails_fast(tmp_path):
    env0 = dict(os.environ, R="0", P="29821", CKPT=str(tmp_path / "c.npz"))
    env1 = dict(env0, R="1", NEZHA_FAULT_STEP="3", NEZHA_FAULT_RANK="1")
    p0 = subprocess.Popen([sys.executable, "-c", WORKER], env=env0, cwd=REPO,
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    p1 = subprocess.Popen([sys.executable, "-c", WORKER], env=env1, cwd=REPO,
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    o1, _ = p1.communicate(timeout=60)
    assert p1.returncode == 42, o1.decode()  # injected death
    # rank 0 must terminate with an error (broken pipe), not hang
    o0, _ = p0.communicate(timeout=60)
    assert p0.returncode != 0, "rank 0 should fail when its peer dies"


def test_restart_from_checkpoint(tmp_path):
    """Single-rank crash at step 3 -> restart from the step-2 checkpoint and
    finish; final state equals an uninterrupted run."""
    import nezha_amd as nz
    ckpt = str(tmp_path / "snap.npz")
    kw = dict(layers=2, dim=32, batch=8, lr=1e-3, seed=5)

    # uninterrupted reference: 6 steps
    ctx = nz.Ctx(nz.CPU)
    ref = nz.Trainer.mlp(ctx, **kw)
    for _ in range(6):
        ref.step()
    ref_path = str(tmp_path / "ref.npz")
    ref.save(ref_path)

    # "crashing" run: 2 steps + checkpoint (simulates periodic snapshots)
    a = nz.Trainer.mlp(ctx, **kw)
    for _ in range(2):
        a.step()
    a.save(ckpt)
    del a  # crash

    # restart: load, run remaining 4 steps
    b = nz.Trainer.mlp(ctx, **kw)
    b.load(ckpt)
    assert b.step_count == 2
    for _ in range(4):
        b.step()
    out = str(tmp_path / "resumed.npz")
    b.save(out)

    cr, cb = np.load(ref_path), np.load(out)
    for k in cr.files:
        np.testing.assert_array_equal(cr[k], cb[k], err_msg=k)


def test_flags_precedence():
    """CLI --steps beats NEZHA_STEPS env; env applies when flag absent."""
    import json
    binp = os.path.join(REPO, "build", "nezha-train")
    base = [binp, "--model=mlp", "--gpus=1", "--warmup=0", "--dim=64",
            "--layers=1", "--batch=16"]
    env = dict(os.environ, NEZHA_STEPS="3")
    r = subprocess.run(base + ["--steps=2"], env=env, capture_output=True,
                       timeout=120)
    assert json.loads(r.stdout.decode().strip().splitlines()[-1])["steps"] == 2
    r = subprocess.run(base, env=env, capture_output=True, timeout=120)
    assert json.loads(r.stdout.decode().strip().splitlines()[-1])["steps"] == 3
