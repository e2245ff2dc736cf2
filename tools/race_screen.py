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
for (M, N, K) in [(512, 512, 512), (2048, 1024, 768), (4096, 4096, 4096),
                  (8192, 4096, 4096)]:
    A = ctx.tensor((M, K), nz.BF16).rand_normal(1)
    B = ctx.tensor((N, K), nz.BF16).rand_normal(2)
    outs = []
    for rep in range(3):
        C = ctx.tensor((M, N), nz.BF16)
        nz.gemm(A, B, C, False, True)
        outs.append(C.raw())
    for rep in range(1, 3):
        assert np.array_equal(outs[0], outs[rep]), f"NONDET {M}x{N}x{K} rep{rep}"
    if M * N <= 2048 * 1024:  # oracle check on the smaller shapes
        ref = A.numpy().astype(np.float32) @ B.numpy().astype(np.float32).T
        got = nz.bf16_to_f32(outs[0])
        scale = np.abs(ref).max()
        err = np.abs(got - ref).max() / scale
        assert err < 0.02, f"WRONG {M}x{N}x{K}: rel err {err}"
        print(f"{M}x{N}x{K}: deterministic + oracle ok (rel err {err:.4f})", flush=True)
    else:
        print(f"{M}x{N}x{K}: deterministic over 3 runs", flush=True)
print("RACE-SCREEN-OK variant=%s min_wg=%s" % (os.environ.get("NEZHA_GEMM_VARIANT", "2"),
                                               os.environ.get("NEZHA_GEMM_MIN_WG", "224")))
