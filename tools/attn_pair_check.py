"""A/B the two-kernel attention backward pair against the monolithic
S=128 kernel (NEZHA_ATTN128_BWD=pair routing in src/core/ops.cc). Run on
GPU. Spawns a subprocess per routing so the env latch is clean."""
import os
import subprocess
import sys
import tempfile

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CODE = r"""
import os, sys
sys.path.insert(0, %r)
import numpy as np, nezha_amd as nz
ctx = nz.Ctx(nz.HIP, 0)
B, S, H, dh = 4, 128, 12, 64
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
""" % (ROOT,)

outs = {}
for mode in ("mono", "pair"):
    env = dict(os.environ)
    env.pop("NEZHA_ATTN128_BWD", None)
    if mode == "pair":
        env["NEZHA_ATTN128_BWD"] = "pair"
    pth = tempfile.mktemp(suffix=".npy")
    r = subprocess.run([sys.executable, "-c", CODE, pth], env=env,
                       capture_output=True, timeout=600)
    if r.returncode != 0:
        print(r.stdout.decode() + r.stderr.decode())
        sys.exit(1)
    outs[mode] = np.load(pth)
    os.unlink(pth)

a, b = outs["mono"], outs["pair"]
scale = np.abs(a).max() + 1e-9
err = np.abs(a - b).max() / scale
print(f"dqkv: pair-vs-mono rel err {err:.5f}", flush=True)
ok = err < 0.04
print("ATTN128-PAIR-OK" if ok else "ATTN128-PAIR-FAIL", flush=True)
sys.exit(0 if ok else 1)
