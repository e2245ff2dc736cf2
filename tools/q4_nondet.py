"""Localize the pp64q4 nondeterminism: run the failing shape repeatedly,
collect mismatch coordinates, and histogram them by tile / quadrant /
slot parity so the racing unit is identifiable. NEZHA_PERS_DIAG=9 runs
the kernel in full-drain debug mode (every phase end waits vmcnt(0))."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import nezha_amd as nz

M, N, K = 32768, 2304, 768
ctx = nz.Ctx(nz.HIP, 0)
A = ctx.tensor((M, K), nz.BF16).rand_normal(1)
B = ctx.tensor((N, K), nz.BF16).rand_normal(2)
ref = None
bad = 0
coords = []
REPS = 8
for rep in range(REPS):
    C = ctx.tensor((M, N), nz.BF16)
    nz.gemm(A, B, C, False, True)
    raw = C.raw()
    if ref is None:
        ref = raw
        continue
    if not np.array_equal(ref, raw):
        bad += 1
        mism = np.nonzero((ref != raw).reshape(M, N))
        coords.append(mism)
print(f"{bad}/{REPS-1} reps mismatched", flush=True)
for mism in coords[:3]:
    r, c = mism
    print(f"  {len(r)} cells; rows {r.min()}-{r.max()} cols {c.min()}-{c.max()}")
    tm = np.unique(r // 128)
    tn = np.unique(c // 128)
    print(f"  m-tiles: {tm[:12]}{'...' if len(tm) > 12 else ''} ({len(tm)})")
    print(f"  n-tiles: {tn[:12]}{'...' if len(tn) > 12 else ''} ({len(tn)})")
    # within-tile structure: group half (row%128 < 64 -> G0 rows), b-half
    rh = np.bincount((r % 128) // 64, minlength=2)
    ch = np.bincount((c % 128) // 64, minlength=2)
    print(f"  row-half counts {rh.tolist()} col-half counts {ch.tolist()}")
    kpar = np.unique((r // 128) % 3)
    print(f"  m-tile %3 {kpar.tolist()}  m-tile %2 {np.unique((r//128)%2).tolist()}")
sys.exit(1 if bad else 0)
