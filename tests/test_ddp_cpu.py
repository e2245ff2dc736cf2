"""T4: DDP logic without a cluster.

 - FakeComm 8-rank bucketing/averaging logic is covered by the C++ test
   binary (tests/cpp/test_ddp_fake.cc), invoked here via subprocess.
 - TcpComm true multi-process CPU data parallelism: 2 worker processes,
   rank-sharded synthetic data, verify the two ranks end bitwise-identical
   and that DDP-with-identical-data reproduces single-rank training.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_worker(script, env=None):
    e = dict(os.environ)
    if env:
        e.update(env)
    return subprocess.Popen([sys.executable, "-c", script], env=e, cwd=REPO,
                            stdout=subprocess.PIPE, stderr=subprocess.STDOUT)


WORKER = r"""
import os, sys
sys.path.insert(0, os.getcwd())
import numpy as np
import nezha_amd as nz

rank = int(os.environ["R"]); n = int(os.environ["N"]); port = int(os.environ["P"])
same_data = os.environ.get("SAME_DATA", "0") == "1"
ctx = nz.Ctx(nz.CPU)
tr = nz.Trainer.mlp(ctx, layers=2, dim=32, batch=8, lr=1e-3, seed=5,
                    bucket_bytes=4096)
comm = nz.Comm.tcp(n, rank, "127.0.0.1", port)
tr.attach_ddp(comm)
if not same_data:
    tr.reseed_data(1000 + rank)
for _ in range(4):
    tr.step()
tr.save(os.environ["OUT"])  # rank 0 writes, others hit the barrier inside
tr.save_local(os.environ["OUT"] + f".rank{rank}")  # per-rank, ungated
comm.barrier()
"""


@pytest.mark.parametrize("same_data", [True, False])
def test_tcp_ddp_two_ranks(tmp_path, same_data):
    out = str(tmp_path / "ddp.npz")
    port = 29801 + (1 if same_data else 0)
    env = {"N": "2", "P": str(port), "OUT": out,
           "SAME_DATA": "1" if same_data else "0"}
    ps = [run_worker(WORKER, {**env, "R": str(r)}) for r in range(2)]
    for p in ps:
        o, _ = p.communicate(timeout=120)
        assert p.returncode == 0, o.decode()
    c0 = np.load(out)
    assert len(c0.files) > 4
    # BITWISE cross-rank agreement: after broadcast + deterministic
    # all-reduce, both ranks' full optimizer state must be byte-identical
    # (params, master, moments, step) — this catches any rank divergence,
    # sharded data or not.
    r0 = np.load(out + ".rank0")
    r1 = np.load(out + ".rank1")
    assert set(r0.files) == set(r1.files)
    for k in r0.files:
        if k == "meta/data_seed":
            continue  # per-rank data shards legitimately differ in seed
        assert r0[k].tobytes() == r1[k].tobytes(), f"rank divergence in {k}"

    if same_data:
        # DP over identical shards == single-rank training (grad avg of two
        # identical grads). Verifies broadcast + allreduce + 1/n scaling.
        import nezha_amd as nz
        ctx = nz.Ctx(nz.CPU)
        tr = nz.Trainer.mlp(ctx, layers=2, dim=32, batch=8, lr=1e-3, seed=5,
                            bucket_bytes=4096)
        for _ in range(4):
            tr.step()
        single = str(tmp_path / "single.npz")
        tr.save(single)
        cs = np.load(single)
        for k in cs.files:
            np.testing.assert_allclose(c0[k], cs[k], rtol=1e-6, atol=1e-7,
                                       err_msg=k)


def test_fake_comm_ddp_cpp():
    """8 simulated ranks in threads (FakeComm) — C++ test binary."""
    binp = os.path.join(REPO, "build", "test_ddp_fake")
    if not os.path.exists(binp):
        subprocess.run(["make", "tests", "-j4"], cwd=REPO, check=True,
                       capture_output=True)
    r = subprocess.run([binp], capture_output=True, timeout=300)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


def test_modules_cpp():
    binp = os.path.join(REPO, "build", "test_modules")
    if not os.path.exists(binp):
        subprocess.run(["make", "tests", "-j4"], cwd=REPO, check=True,
                       capture_output=True)
    r = subprocess.run([binp], capture_output=True, timeout=300)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


def test_core_cpp():
    binp = os.path.join(REPO, "build", "test_core")
    if not os.path.exists(binp):
        subprocess.run(["make", "tests", "-j4"], cwd=REPO, check=True,
                       capture_output=True)
    r = subprocess.run([binp], capture_output=True, timeout=300)
    assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


def test_tcp_comm_scalar(tmp_path):
    script = r"""
import os, sys
sys.path.insert(0, os.getcwd())
import nezha_amd as nz
rank = int(os.environ["R"]); n = int(os.environ["N"])
comm = nz.Comm.tcp(n, rank, "127.0.0.1", int(os.environ["P"]))
s = comm.allreduce_scalar(float(rank + 1))
assert s == sum(range(1, n + 1)), s
m = comm.allreduce_scalar(float(rank + 1), "max")
assert m == n, m
comm.barrier()
print("OK")
"""
    ps = [run_worker(script, {"R": str(r), "N": "3", "P": "29811"})
          for r in range(3)]
    for p in ps:
        o, _ = p.communicate(timeout=60)
        assert p.returncode == 0 and b"OK" in o, o.decode()


def test_tcp_ddp_f32_allreduce(tmp_path):
    """--allreduce-dtype=f32 fallback: 2-rank CPU DP with the
    f32 upcast path trains and stays rank-synchronized."""
    out = str(tmp_path / "f32.npz")
    env = {"N": "2", "P": "29831", "OUT": out, "SAME_DATA": "1",
           "NEZHA_ALLREDUCE_DTYPE": "f32"}
    ps = [run_worker(WORKER, {**env, "R": str(r)}) for r in range(2)]
    for p in ps:
        o, _ = p.communicate(timeout=120)
        assert p.returncode == 0, o.decode()
    c = np.load(out)
    # same-data DP == single-rank training (f32 round trip is exact for
    # values that came from bf16 grads)
    import nezha_amd as nz
    ctx = nz.Ctx(nz.CPU)
    tr = nz.Trainer.mlp(ctx, layers=2, dim=32, batch=8, lr=1e-3, seed=5,
                        bucket_bytes=4096)
    for _ in range(4):
        tr.step()
    single = str(tmp_path / "single.npz")
    tr.save(single)
    cs = np.load(single)
    for k in cs.files:
        np.testing.assert_allclose(c[k], cs[k], rtol=1e-6, atol=1e-7, err_msg=k)
