"""Race screen + oracle check for the NT GEMM variants (guide two-lane
discipline): multi-run bitwise determinism at several shapes plus a
numpy-f32-on-bf16-inputs reference check. Select the kernel with
NEZHA_GEMM_VARIANT / NEZHA_GEMM_MIN_WG before running."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import nezha_amd as nz

ctx = nz.Ctx(nz.HIP, 0)
# shapes chosen to hit every dispatch tier: tier-A (512^3), plain nt_k
# (2048x1024), 8-phase (4096^3, 8192-), PERSISTENT short-K (32768x2304x768)
# and the split-K 8-phase wgrad path (TN 768x2304xK32768)
for (M, N, K, tA) in [(512, 512, 512, False), (2048, 1024, 768, False),
                      (4096, 4096, 4096, False), (8192, 4096, 4096, False),
                      (32768, 2304, 768, False), (768, 2304, 32768, True)]:
    A = ctx.tensor((K, M) if tA else (M, K), nz.BF16).rand_normal(1)
    B = ctx.tensor((K, N) if tA else (N, K), nz.BF16).rand_normal(2)
    outs = []
    for rep in range(3):
        C = ctx.tensor((M, N), nz.BF16)
        nz.gemm(A, B, C, tA, not tA)
        outs.append(C.raw())
    for rep in range(1, 3):
        assert np.array_equal(outs[0], outs[rep]), f"NONDET {M}x{N}x{K} rep{rep}"
    if M * N <= 2048 * 1024 and K <= 1024:  # oracle on small+cheap shapes
        a_np = A.numpy().astype(np.float32)
        b_np = B.numpy().astype(np.float32)
        ref = (a_np.T @ b_np) if tA else (a_np @ b_np.T)
        got = nz.bf16_to_f32(outs[0])
        scale = np.abs(ref).max()
        err = np.abs(got - ref).max() / scale
        assert err < 0.02, f"WRONG {M}x{N}x{K}: rel err {err}"
        print(f"{M}x{N}x{K}: deterministic + oracle ok (rel err {err:.4f})", flush=True)
    else:
        print(f"{M}x{N}x{K}: deterministic over 3 runs", flush=True)

# Cross-kernel oracle for shapes too big for numpy: the dispatch-selected
# kernel vs the proven tier-A/128^2 route (NEZHA_GEMM_VARIANT=0). Both
# accumulate f32 over the same K (different order), so rel err ~ bf16 eps.
# Covers the persistent ping-pong (short-K), ping-pong (deep-K) and the
# NATIVE NN/TN transpose-in-stage routes (r2) incl. native split-K TN.
for (M, N, K, tA, tB) in [(32768, 1024, 768, False, True),
                          (16384, 2304, 768, False, True),
                          (8192, 4096, 4096, False, True),
                          (8192, 4096, 4096, False, False),   # native NN
                          (4096, 4096, 32768, True, False),   # native TN
                          (768, 2304, 32768, True, False)]:   # native TN splitk
    A = ctx.tensor((K, M) if tA else (M, K), nz.BF16).rand_normal(11)
    B = ctx.tensor((N, K) if tB else (K, N), nz.BF16).rand_normal(12)
    C = ctx.tensor((M, N), nz.BF16)
    nz.gemm(A, B, C, tA, tB)
    # determinism of the dispatched route
    C2 = ctx.tensor((M, N), nz.BF16)
    nz.gemm(A, B, C2, tA, tB)
    assert np.array_equal(C.raw(), C2.raw()), f"NONDET {M}x{N}x{K} {tA}{tB}"
    got = C.numpy()
    import subprocess, sys as _sys
    # reference kernel in a subprocess (variant env is latched per process)
    import tempfile, os as _os
    ref_env = dict(os.environ, NEZHA_GEMM_VARIANT="0")
    code = (
        "import os,sys; sys.path.insert(0, %r)\n"
        "import numpy as np, nezha_amd as nz\n"
        "ctx = nz.Ctx(nz.HIP, 0)\n"
        "A = ctx.tensor((%d, %d), nz.BF16).rand_normal(11)\n"
        "B = ctx.tensor((%d, %d), nz.BF16).rand_normal(12)\n"
        "C = ctx.tensor((%d, %d), nz.BF16)\n"
        "nz.gemm(A, B, C, %r, %r)\n"
        "np.save(sys.argv[1], C.numpy())\n"
    ) % (os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
         K if tA else M, M if tA else K,
         N if tB else K, K if tB else N, M, N, tA, tB)
    with tempfile.NamedTemporaryFile(suffix=".npy", delete=False) as f:
        refp = f.name
    subprocess.run([_sys.executable, "-c", code, refp], env=ref_env,
                   check=True)
    ref = np.load(refp)
    _os.unlink(refp)
    scale = np.abs(ref).max() + 1e-9
    err = np.abs(got - ref).max() / scale
    assert err < 0.02, f"CROSS-KERNEL MISMATCH {M}x{N}x{K}: rel err {err}"
    print(f"{M}x{N}x{K}: cross-kernel oracle ok (rel err {err:.4f})",
          flush=True)

print("RACE-SCREEN-OK variant=%s min_wg=%s" % (os.environ.get("NEZHA_GEMM_VARIANT", "2"),
                                               os.environ.get("NEZHA_GEMM_MIN_WG", "224")))
