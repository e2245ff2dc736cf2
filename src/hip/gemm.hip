// K1: hand-written CDNA4 bf16 GEMM (MFMA v_mfma_f32_16x16x32_bf16, f32
// accumulate) with fused bias(+GELU) epilogue and strided-batch mode.
//
// Kernel ladder in this file (docs/perf has the measured TF/s per rung):
//   gemm_bf16_k           generic trans-variant tiled kernel (TT + ragged)
//   gemm_bf16_nt_k        2-buffer glds NT, 128^2 tile — batched/small/
//                         split-K slices
//   gemm_bf16_nt_deep_k   3-buffer counted-vmcnt experiment (variant 1)
//   gemm_bf16_nt_8phase_k 256^2 half-tile 8-phase — THE default (variant 2)
//   gemm_bf16_nt_8phase2/4/32_k  measured alternatives (variants 3/4/5)
// NN and TN lower onto NT via the vectorized transpose (design rationale:
// docs/design.md "why NT is the canonical form"); tiny-output large-K
// shapes split K into f32 slabs + a fused reduce.
#include "src/hip/hip_common.h"
#include "src/hip/hip_ops.h"

namespace nz {
namespace hip {

using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int LDS_STRIDE = BK + 8;  // 72 elems = 144 B, 16-B aligned rows
constexpr int THREADS = 256;

struct GemmArgs {
  const uint16_t* A;
  const uint16_t* B;
  uint16_t* C;
  float* out_f32 = nullptr;  // split-K partial mode: raw f32, no epilogue
  const uint16_t* bias;   // null or [N]
  uint16_t* save_pre;     // null or [M,N]
  int64_t M, N, K;
  int64_t lda, ldb;       // row strides of the STORED matrices
  int64_t sA, sB, sC;     // batch strides (elements)
  float alpha, beta;
  int fuse_gelu;
  // dgrad-side fused GELU derivative: C = (alpha*acc) * gelu'(dgelu_pre)
  // [+ beta*C + bias] — set only for routes whose kernels implement it
  // (gemm_can_fuse_dgelu mirrors the dispatch below).
  const uint16_t* dgelu_pre;
  int diag_no_store;  // NEZHA_PERS_DIAG=1: skip pers epilogue stores
                      // (DIAGNOSTIC ONLY — output garbage; cost attribution)
  int group_m = 0;    // >0: GROUP_M-blocked tile raster (see tile_coords)
};

// wgid -> 256^2 tile coordinates. group_m == 0: plain m-major raster.
// group_m > 0: GROUP_M-blocked raster (the CUTLASS/Triton "grouped" swizzle)
// so that a CONTIGUOUS run of wgids forms a group_m x k tile rectangle.
// Why (measured, gpurun_out/rocprof_r2): the XCD remap gives each XCD a
// contiguous wgid chunk, so its 32 co-resident WGs used to cover a 1x32
// m-major run — one shared A panel but 32 DISTINCT B panels streaming
// through a 4 MiB XCD L2 = no reuse; the request stream plateaus at
// ~6.6 TB/s and the MFMA pipe sits 57% idle. With group_m=8 the same 32
// WGs cover an 8x4 tile block: 8 A + 4 B panels, every fetched line
// consumed 4-8x from L2.
__device__ __forceinline__ void tile_coords(int wgid, int nwg_m, int nwg_n,
                                            int group_m, int64_t& tm,
                                            int64_t& tn) {
  if (group_m <= 1) {
    tm = wgid / nwg_n;
    tn = wgid % nwg_n;
    return;
  }
  int per_group = group_m * nwg_n;
  int gid = wgid / per_group;
  int first_m = gid * group_m;
  int gsz = nwg_m - first_m;
  if (gsz > group_m) gsz = group_m;
  int pos = wgid % per_group;
  tm = first_m + pos % gsz;
  tn = pos / gsz;
}

// Stage op(A) tile [BM][BK] (rows m, cols k) into lds with row stride
// LDS_STRIDE. TRANS=false: stored [.,lda] row-major [M,K] -> vector loads
// along k. TRANS=true: stored [K,M] -> vector loads along m, scatter to lds.
template <bool TRANS>
__device__ void stage_a(const uint16_t* __restrict__ A, uint16_t* __restrict__ lds,
                        int64_t m0, int64_t k0, int64_t M, int64_t K, int64_t lda) {
  constexpr int ELEMS = BM * BK / THREADS;  // 32 elems per thread, as 4 x 8
#pragma unroll
  for (int rep = 0; rep < ELEMS / 8; ++rep) {
    int e = (rep * THREADS + (int)threadIdx.x) * 8;
    if (!TRANS) {
      int m = e / BK, k = e % BK;
      int64_t gm = m0 + m, gk = k0 + k;
      short8 v = {};
      if (gm < M) {
        if (gk + 7 < K) {
          v = *(const short8*)(A + gm * lda + gk);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i)
            v[i] = (gk + i < K) ? (short)A[gm * lda + gk + i] : (short)0;
        }
      }
      *(short8*)(lds + m * LDS_STRIDE + k) = v;
    } else {
      int k = e / BM, m = e % BM;
      int64_t gk = k0 + k, gm = m0 + m;
      short8 v = {};
      if (gk >= K) {
        // ragged-K zero row
      } else if (gm + 7 < M) {
        v = *(const short8*)(A + gk * lda + gm);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          v[i] = (gm + i < M) ? (short)A[gk * lda + gm + i] : (short)0;
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) lds[(m + i) * LDS_STRIDE + k] = (uint16_t)v[i];
    }
  }
}

// Stage op(B) tile into lds TRANSPOSED: lds[n][k] = op(B)[k][n].
// TRANS=false: stored [K,N] -> vector loads along n, scatter.
// TRANS=true:  stored [N,K] -> vector loads along k, direct.
template <bool TRANS>
__device__ void stage_b(const uint16_t* __restrict__ B, uint16_t* __restrict__ lds,
                        int64_t n0, int64_t k0, int64_t N, int64_t K, int64_t ldb) {
  constexpr int ELEMS = BN * BK / THREADS;
#pragma unroll
  for (int rep = 0; rep < ELEMS / 8; ++rep) {
    int e = (rep * THREADS + (int)threadIdx.x) * 8;
    if (!TRANS) {
      int k = e / BN, n = e % BN;
      int64_t gk = k0 + k, gn = n0 + n;
      short8 v = {};
      if (gk >= K) {
        // ragged-K zero row
      } else if (gn + 7 < N) {
        v = *(const short8*)(B + gk * ldb + gn);
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          v[i] = (gn + i < N) ? (short)B[gk * ldb + gn + i] : (short)0;
      }
#pragma unroll
      for (int i = 0; i < 8; ++i) lds[(n + i) * LDS_STRIDE + k] = (uint16_t)v[i];
    } else {
      int n = e / BK, k = e % BK;
      int64_t gn = n0 + n, gk = k0 + k;
      short8 v = {};
      if (gn < N) {
        if (gk + 7 < K) {
          v = *(const short8*)(B + gn * ldb + gk);
        } else {
#pragma unroll
          for (int i = 0; i < 8; ++i)
            v[i] = (gk + i < K) ? (short)B[gn * ldb + gk + i] : (short)0;
        }
      }
      *(short8*)(lds + n * LDS_STRIDE + k) = v;
    }
  }
}

template <bool TA, bool TB>
__launch_bounds__(THREADS) __global__ void gemm_bf16_k(GemmArgs args) {
  __shared__ uint16_t As[2][BM * LDS_STRIDE];
  __shared__ uint16_t Bs[2][BN * LDS_STRIDE];

  const int64_t batch = blockIdx.z;
  const uint16_t* A = args.A + batch * args.sA;
  const uint16_t* B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  uint16_t* save_pre = args.save_pre ? args.save_pre + batch * args.sC : nullptr;

  const int64_t m0 = (int64_t)blockIdx.y * BM;
  const int64_t n0 = (int64_t)blockIdx.x * BN;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;      // 4 waves
  const int wr = wid >> 1, wc = wid & 1; // 2x2 wave grid, 64x64 each

  f32x4 acc[4][4] = {};

  const int64_t ntiles = (args.K + BK - 1) / BK;  // ragged K: zero-padded
  stage_a<TA>(A, As[0], m0, 0, args.M, args.K, args.lda);
  stage_b<TB>(B, Bs[0], n0, 0, args.N, args.K, args.ldb);
  __syncthreads();

  int cur = 0;
  for (int64_t t = 0; t < ntiles; ++t) {
    // Issue next tile's staging AFTER compute reads current LDS? No:
    // stage into the other buffer now (loads overlap MFMA via scheduler),
    // barrier afterwards protects buffer reuse two iterations out.
    if (t + 1 < ntiles) {
      stage_a<TA>(A, As[cur ^ 1], m0, (t + 1) * BK, args.M, args.K, args.lda);
      stage_b<TB>(B, Bs[cur ^ 1], n0, (t + 1) * BK, args.N, args.K, args.ldb);
    }
    const uint16_t* Ab = As[cur];
    const uint16_t* Bb = Bs[cur];
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      short8 a_frag[4], b_frag[4];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        a_frag[fm] = *(const short8*)(Ab + row * LDS_STRIDE + kbase);
      }
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = wc * 64 + fn * 16 + (lane & 15);
        b_frag[fn] = *(const short8*)(Bb + col * LDS_STRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // Epilogue: alpha/beta, +bias, save_pre, gelu, bf16 store.
  // C/D fragment map (guide §3): col = lane&15, row = (lane>>4)*4 + r.
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int64_t grow = m0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + r;
        int64_t gcol = n0 + wc * 64 + fn * 16 + (lane & 15);
        if (grow >= args.M || gcol >= args.N) continue;
        int64_t idx = grow * args.N + gcol;
        float v = args.alpha * acc[fm][fn][r];
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Tier B: fast NT kernel (C[M,N] = A[M,K] . B[N,K]^T), both operands
// k-contiguous, staged by direct global->LDS DMA (global_load_lds_dwordx4,
// guide §5 ladder step 3: width=16 took 517->874 TF). 128x128 tile, BK=64,
// double-buffered 64 KB LDS (2 blocks/CU), issue-next-tile-before-compute
// (T3 minimum 2-phase), s_setprio(1) around the MFMA cluster (T5), and a
// bijective XCD-aware block swizzle (T1) so neighbor tiles share an XCD L2.
// Out-of-range rows/cols are CLAMPED at the staging loads (glds cannot mask
// lanes); the epilogue masks stores, so garbage lanes never escape.
// ---------------------------------------------------------------------------

__device__ __forceinline__ void glds16(const uint16_t* gsrc, uint16_t* lds_dst) {
  // direct global->LDS DMA, 16 B per lane (global_load_lds_dwordx4).
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t*)gsrc,
      (__attribute__((address_space(3))) uint32_t*)lds_dst, 16, 0, 0);
}

__launch_bounds__(THREADS) __global__ void gemm_bf16_nt_k(GemmArgs args,
                                                          int nwg_n) {
  constexpr int NT_BK = 64;
  // ONE shared object (guide §5 ".s-level traps" (a): a second __shared__
  // array would force a vmcnt(0) before every ds_read, de-pipelining glds).
  // Layout: [buf][A tile 128x64 | B tile 128x64].
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * (BM + BN) * NT_BK];
  constexpr int BUF_STRIDE = (BM + BN) * NT_BK;
  constexpr int B_OFF = BM * NT_BK;
#define NT_AS(buf) (smem + (buf) * BUF_STRIDE)
#define NT_BS(buf) (smem + (buf) * BUF_STRIDE + B_OFF)

  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  uint16_t* save_pre = args.save_pre ? args.save_pre + batch * args.sC : nullptr;

  // XCD-aware bijective block swizzle (guide T1): contiguous grid chunk per
  // XCD so neighboring tiles (sharing A rows / B panels) hit the same L2.
  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  const int64_t m0 = (int64_t)(wgid / nwg_n) * BM;
  const int64_t n0 = (int64_t)(wgid % nwg_n) * BN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;

  // Per-lane global element offsets for the 4 glds chunks of each operand:
  // chunk c covers bytes [wid*4096 + c*1024 + lane*16]; elem e = byte/2,
  // row = e/64, k = e%64 (lane-linear LDS image [128][64], rows of 128 B).
  int e0 = (wid * 2048) + lane * 8;  // elements; +c*512 per chunk

  f32x4 acc[4][4] = {};
  const int64_t ntiles = args.K / NT_BK;

  // LDS XOR swizzle (guide §6 G4 / §5.4 rule 21): without it every 16-lane
  // ds_read_b128 group of the fragment reads lands on 2 bank groups (8-way
  // conflict, rows are 128 B). The involution kbyte ^= (row&7)<<4 spreads a
  // group over 8 slots; with glds the LDS image is lane-linear, so the
  // permutation is applied to the per-lane global SOURCE address here and
  // to the ds_read byte address below (same involution both sides).
  auto stage = [&](int buf, int64_t t) {
    int64_t k0 = t * NT_BK;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = e0 + c * 512;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);  // element swizzle (16B = 8 elems)
      int64_t ar = m0 + row;
      if (ar > Mc) ar = Mc;  // clamp: masked at the epilogue
      glds16(A + ar * lda + k0 + k, NT_AS(buf) + e);
      int64_t br = n0 + row;
      if (br > Nc) br = Nc;
      glds16(B + br * ldb + k0 + k, NT_BS(buf) + e);
    }
  };

  stage(0, 0);
  __syncthreads();  // carries vmcnt(0): tile 0 landed

  int cur = 0;
  for (int64_t t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) stage(cur ^ 1, t + 1);  // issue BEFORE compute (T3)
#pragma unroll
    for (int kk = 0; kk < NT_BK; kk += 32) {
      short8 a_frag[4], b_frag[4];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);  // read-side of the involution
        a_frag[fm] = *(const short8*)(NT_AS(cur) + row * NT_BK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = wc * 64 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(NT_BS(cur) + col * NT_BK + ks);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();  // drains the in-flight glds (vmcnt(0)) + barrier
    cur ^= 1;
  }

  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  const int64_t N = args.N;
  if (m0 + BM <= args.M && n0 + BN <= args.N) {
    // interior fast path (hoisted addressing; see gemm_bf16_nt_pers_k)
    const int64_t base =
        (m0 + wr * 64 + (lane >> 4) * 4) * N + n0 + wc * 64 + (lane & 15);
    const uint16_t* bias_l =
        args.bias ? args.bias + n0 + wc * 64 + (lane & 15) : nullptr;
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        const int64_t rowbase = base + (int64_t)(fm * 16 + rI) * N;
#pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          const int64_t idx = rowbase + fn * 16;
          if (out32) {
            out32[idx] = acc[fm][fn][rI];
          } else {
            float v = args.alpha * acc[fm][fn][rI];
            if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
            if (bias_l) v += b2f(bias_l[fn * 16]);
            if (save_pre) save_pre[idx] = f2b(v);
            if (args.fuse_gelu) v = gelu_fwd_scalar(v);
            C[idx] = f2b(v);
          }
        }
      }
    return;
  }
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int64_t grow = m0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
        int64_t gcol = n0 + wc * 64 + fn * 16 + (lane & 15);
        if (grow >= args.M || gcol >= args.N) continue;
        int64_t idx = grow * args.N + gcol;
        if (out32) {
          out32[idx] = acc[fm][fn][rI];
          continue;
        }
        float v = args.alpha * acc[fm][fn][rI];
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Persistent NT kernel: fixed 512-WG grid (2 blocks/CU), each WG sweeps a
// strided run of (batch, m, n) tiles with the SAME 2-buffer glds pipeline as
// gemm_bf16_nt_k, staging tile t+1 / the NEXT tile's first k-block during the
// current compute. Motivation (measured): at K=768 a 128^2 tile is only 12
// k-iterations (~5 us of MFMA) against a ~2 us stage prologue + drain, so the
// one-tile-per-WG kernel tops out near 60% of its long-K rate (536 vs 880+
// TF/s on the BERT qkv-fwd shape); tiny batched tiles (attention backward,
// K=64, ONE k-iteration) are almost pure prologue. Amortizing one prologue
// over cnt/64 tiles per WG and making every WG process the same tile count
// removes both the ramp and the grid-quantization waste (e.g. N=768 ->
// 1536 WGs = exactly 3 rounds, but 576 -> 1.125 rounds = 12.5% idle).
// Tile order: XCD-chunked like the block swizzle (WG w = slot w/8 on XCD
// w%8 walks chunk tiles slot, slot+64, ... so the 64 co-resident WGs of an
// XCD always cover ~64 consecutive m-major tiles -> A rows and B panels
// stay hot in that XCD's L2).
// ---------------------------------------------------------------------------
__launch_bounds__(THREADS) __global__ void gemm_bf16_nt_pers_k(
    GemmArgs args, int nwg_n, int64_t per_b, int64_t total) {
  constexpr int NT_BK = 64;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * (BM + BN) * NT_BK];
  constexpr int BUF_STRIDE = (BM + BN) * NT_BK;
  constexpr int B_OFF = BM * NT_BK;

  const int orig = blockIdx.x;
  const int xcd = orig & 7, slot = orig >> 3;  // 64 slots per XCD
  const int64_t q = total / 8, r8 = total % 8;
  const int64_t lo = xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q;
  const int64_t cnt = xcd < r8 ? q + 1 : q;
  if (slot >= cnt) return;
  constexpr int S = 64;  // slot stride inside the chunk

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  int e0 = (wid * 2048) + lane * 8;

  f32x4 acc[4][4] = {};
  const int64_t ntiles = args.K / NT_BK;

  // Staging-tile state, advanced ONCE per tile: the two 64-bit div/mods and
  // the 8 address rebuilds must stay out of the per-k-iteration stage call
  // (first version recomputed them every iteration and lost ~15% on every
  // shape). The per-lane A/B source addresses for the 4 glds chunks are
  // fully precomputed (clamp included); stage() only adds k0.
  const uint16_t* asrc[4];
  const uint16_t* bsrc[4];
  auto set_tile = [&](int64_t t_l) {
    const int64_t b = t_l / per_b, rtl = t_l % per_b;
    const uint16_t* __restrict__ Ab = args.A + b * args.sA;
    const uint16_t* __restrict__ Bb = args.B + b * args.sB;
    const int64_t sm0 = (rtl / nwg_n) * BM;
    const int64_t sn0 = (rtl % nwg_n) * BN;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = e0 + c * 512;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);  // XOR involution, source side
      int64_t ar = sm0 + row;
      if (ar > Mc) ar = Mc;  // clamp: masked at the epilogue
      asrc[c] = Ab + ar * lda + k;
      int64_t br = sn0 + row;
      if (br > Nc) br = Nc;
      bsrc[c] = Bb + br * ldb + k;
    }
  };
  auto stage = [&](int buf, int64_t k0) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = e0 + c * 512;
      glds16(asrc[c] + k0, smem + buf * BUF_STRIDE + e);
      glds16(bsrc[c] + k0, smem + buf * BUF_STRIDE + B_OFF + e);
    }
  };

  // The first version of this epilogue did per-element int64 bounds checks
  // and full grow*N+gcol address math: ~4000 VALU + exec-mask churn per
  // tile = ~6 us — DOMINANT at short K (a 1-k-iteration tile measured
  // 11.6 us wall against 1.15 us of MFMA). Interior tiles now take a
  // hoisted-address path: one branch per tile, one row mul per (fm,rI),
  // adds only inside.
  auto epilogue = [&](int64_t tl) {
    const int64_t b = tl / per_b, rtl = tl % per_b;
    const int64_t em0 = (rtl / nwg_n) * BM;
    const int64_t en0 = (rtl % nwg_n) * BN;
    uint16_t* C = args.C + b * args.sC;
    float* out32 = args.out_f32 ? args.out_f32 + b * args.sC : nullptr;
    uint16_t* save_pre = args.save_pre ? args.save_pre + b * args.sC : nullptr;
    const int64_t N = args.N;
    if (args.diag_no_store) return;
    if (em0 + BM <= args.M && en0 + BN <= args.N) {
      const int64_t base =
          (em0 + wr * 64 + (lane >> 4) * 4) * N + en0 + wc * 64 + (lane & 15);
      const uint16_t* bias_l =
          args.bias ? args.bias + en0 + wc * 64 + (lane & 15) : nullptr;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          const int64_t rowbase = base + (int64_t)(fm * 16 + rI) * N;
#pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            const int64_t idx = rowbase + fn * 16;
            if (out32) {
              out32[idx] = acc[fm][fn][rI];
            } else {
              float v = args.alpha * acc[fm][fn][rI];
              if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
              if (bias_l) v += b2f(bias_l[fn * 16]);
              if (save_pre) save_pre[idx] = f2b(v);
              if (args.fuse_gelu) v = gelu_fwd_scalar(v);
              C[idx] = f2b(v);
            }
          }
        }
    } else {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int64_t grow = em0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int64_t gcol = en0 + wc * 64 + fn * 16 + (lane & 15);
            if (grow < args.M && gcol < args.N) {
              int64_t idx = grow * args.N + gcol;
              if (out32) {
                out32[idx] = acc[fm][fn][rI];
              } else {
                float v = args.alpha * acc[fm][fn][rI];
                if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                if (args.bias) v += b2f(args.bias[gcol]);
                if (save_pre) save_pre[idx] = f2b(v);
                if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                C[idx] = f2b(v);
              }
            }
          }
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) acc[fm][fn] = f32x4{};
  };

  int64_t tl = lo + slot;
  set_tile(tl);
  stage(0, 0);
  __syncthreads();  // vmcnt(0): first tile landed

  int cur = 0;
  for (; tl < lo + cnt; tl += S) {
    for (int64_t t = 0; t < ntiles; ++t) {
      // prefetch: next k-block of this tile, else the next tile's block 0
      if (t + 1 < ntiles) {
        stage(cur ^ 1, (t + 1) * NT_BK);
      } else if (tl + S < lo + cnt) {
        set_tile(tl + S);  // once per tile, off the steady-state path
        stage(cur ^ 1, 0);
      }
#pragma unroll
      for (int kk = 0; kk < NT_BK; kk += 32) {
        short8 a_frag[4], b_frag[4];
        const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
        for (int fm = 0; fm < 4; ++fm) {
          int row = wr * 64 + fm * 16 + (lane & 15);
          int ks = kbase ^ ((row & 7) << 3);
          a_frag[fm] =
              *(const short8*)(smem + cur * BUF_STRIDE + row * NT_BK + ks);
        }
#pragma unroll
        for (int fn = 0; fn < 4; ++fn) {
          int col = wc * 64 + fn * 16 + (lane & 15);
          int ks = kbase ^ ((col & 7) << 3);
          b_frag[fn] = *(const short8*)(smem + cur * BUF_STRIDE + B_OFF +
                                        col * NT_BK + ks);
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 4; ++fn)
            acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      if (t + 1 == ntiles) {
        epilogue(tl);  // acc complete; stores issue behind the next stage
        // Counted wait: outstanding = 8 glds (FIFO-older) + <=64 epilogue
        // stores. vmcnt(63) guarantees the glds retired (>=9 retirements)
        // while the stores keep draining under the next tile's compute —
        // a full __syncthreads here paid the store-drain once per tile.
        asm volatile("s_waitcnt vmcnt(63)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
      } else {
        __syncthreads();
      }
      cur ^= 1;
    }
  }
}

// ---------------------------------------------------------------------------
// RING persistent NT kernel: the latency-cover kernel at 2 WGs/CU.
// Diagnosis (docs/perf): the 2-buffer persistent kernel's steady state is
// ~1.2 us/iteration against 0.43 us of MFMA — each stage gets only ONE
// compute-slot (~0.86 us across the 2 co-resident WGs) to cover ~1-2 us of
// HBM latency. Deeper whole-tile buffering busts the 80 KB/WG budget that
// 2 WGs/CU requires; HALF-tile rings fit exactly:
//   A ring: 6 half-slots (64 rows x 64 k, 8 KB) = 3 k-tiles deep — A rows
//           are compulsory HBM misses, they get ~2.5 iterations of cover;
//   B ring: 4 half-slots = 2 k-tiles deep — B panels are L2-resident.
// Total 48 + 32 = 80 KB. Iteration g: wait vmcnt(12) (the 6 younger
// in-flight half-units, 2 glds each) -> compute k-tile g (4 waves, 64x64
// quadrants, same XOR swizzle) -> lgkm barrier (all waves done reading g's
// slots) -> stage A(g+3) both halves + B(g+2) both halves into the slots
// g's data just vacated ((2g+h) mod ring); steady wait vmcnt(8). The
// ring flows ACROSS tile
// boundaries (persistent sweep, same XCD-chunked order as
// gemm_bf16_nt_pers_k); at a boundary the wait is vmcnt(63) so the
// epilogue's scalar stores (FIFO-younger than every needed glds) drain
// under the next tile's compute. True tail: vmcnt(0).
// ---------------------------------------------------------------------------
__launch_bounds__(THREADS) __global__ void gemm_bf16_nt_ring_pers_k(
    GemmArgs args, int nwg_n, int64_t per_b, int64_t total) {
  constexpr int RBK = 64;
  constexpr int HALF = 64 * RBK;              // elements per half-slot (8 KB)
  __shared__ __attribute__((aligned(16))) uint16_t smem[10 * HALF];  // 80 KB
  uint16_t* Aring = smem;                      // 6 slots
  uint16_t* Bring = smem + 6 * HALF;           // 4 slots

  const int orig = blockIdx.x;
  const int xcd = orig & 7, slotq = orig >> 3;  // 64 slots per XCD
  const int64_t q = total / 8, r8 = total % 8;
  const int64_t lo = xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q;
  const int64_t cnt = xcd < r8 ? q + 1 : q;
  if (slotq >= cnt) return;
  constexpr int S = 64;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / RBK;  // k-tiles per output tile
  const int e0 = (int)threadIdx.x * 8;

  f32x4 acc[4][4] = {};

  // Tile states: staging reaches g+3 (A) so at most one boundary ahead.
  struct TileSt {
    const uint16_t* A;
    const uint16_t* B;
    int32_t m0, n0;
  };
  auto make_state = [&](int64_t tl) {
    TileSt ts;
    const int64_t b = tl / per_b, rtl = tl % per_b;
    ts.A = args.A + b * args.sA;
    ts.B = args.B + b * args.sB;
    ts.m0 = (int32_t)((rtl / nwg_n) * BM);
    ts.n0 = (int32_t)((rtl % nwg_n) * BN);
    return ts;
  };

  // stage one half (h = 0/1) of operand A or B for LOCAL k-tile kt of
  // tile-state ts, into ring slot rs. 2 glds per thread.
  auto stage_half = [&](const TileSt& ts, int isB, int h, int rs, int64_t kt) {
    const int64_t k0 = kt * RBK;
    uint16_t* dst = (isB ? Bring : Aring) + rs * HALF;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = e0 + c * 2048;
      int row = e >> 6;  // 0..63 within the half
      int k = (e & 63) ^ (((h * 64 + row) & 7) << 3);
      if (!isB) {
        int64_t ar = ts.m0 + h * 64 + row;
        if (ar > Mc) ar = Mc;
        glds16(ts.A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = ts.n0 + h * 64 + row;
        if (br > Nc) br = Nc;
        glds16(ts.B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  auto epilogue = [&](int64_t tl) {
    const int64_t b = tl / per_b, rtl = tl % per_b;
    const int64_t em0 = (rtl / nwg_n) * BM;
    const int64_t en0 = (rtl % nwg_n) * BN;
    uint16_t* C = args.C + b * args.sC;
    float* out32 = args.out_f32 ? args.out_f32 + b * args.sC : nullptr;
    uint16_t* save_pre = args.save_pre ? args.save_pre + b * args.sC : nullptr;
    const int64_t N = args.N;
    if (em0 + BM <= args.M && en0 + BN <= args.N) {
      const int64_t base =
          (em0 + wr * 64 + (lane >> 4) * 4) * N + en0 + wc * 64 + (lane & 15);
      const uint16_t* bias_l =
          args.bias ? args.bias + en0 + wc * 64 + (lane & 15) : nullptr;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          const int64_t rowbase = base + (int64_t)(fm * 16 + rI) * N;
#pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            const int64_t idx = rowbase + fn * 16;
            if (out32) {
              out32[idx] = acc[fm][fn][rI];
            } else {
              float v = args.alpha * acc[fm][fn][rI];
              if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
              if (bias_l) v += b2f(bias_l[fn * 16]);
              if (save_pre) save_pre[idx] = f2b(v);
              if (args.fuse_gelu) v = gelu_fwd_scalar(v);
              C[idx] = f2b(v);
            }
          }
        }
    } else {
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int64_t grow = em0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int64_t gcol = en0 + wc * 64 + fn * 16 + (lane & 15);
            if (grow < args.M && gcol < args.N) {
              int64_t idx = grow * args.N + gcol;
              if (out32) {
                out32[idx] = acc[fm][fn][rI];
              } else {
                float v = args.alpha * acc[fm][fn][rI];
                if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                if (args.bias) v += b2f(args.bias[gcol]);
                if (save_pre) save_pre[idx] = f2b(v);
                if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                C[idx] = f2b(v);
              }
            }
          }
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) acc[fm][fn] = f32x4{};
  };

  // ---- persistent sweep with a GLOBAL k-iteration counter g (ring phase) --
  int64_t tl = lo + slotq;
  TileSt cur = make_state(tl);
  const bool have2 = (tl + S) < (lo + cnt);
  TileSt nxt = have2 ? make_state(tl + S) : cur;

  // map global k-iter g (< 2 boundaries ahead) to (state, local kt);
  // returns false when past the WG's work
  const int64_t wg_iters = ((cnt - slotq + S - 1) / S) * ntiles;

  // prologue: A(0) B(0) A(1) B(1) A(2) — 10 half-units, 20 glds; the first
  // wait vmcnt(12) retires exactly A(0)+B(0)
  stage_half(cur, 0, 0, 0, 0);
  stage_half(cur, 0, 1, 1, 0);
  stage_half(cur, 1, 0, 0, 0);
  stage_half(cur, 1, 1, 1, 0);
  stage_half(cur, 0, 0, 2, 1);
  stage_half(cur, 0, 1, 3, 1);
  stage_half(cur, 1, 0, 2, 1);
  stage_half(cur, 1, 1, 3, 1);
  stage_half(cur, 0, 0, 4, 2);
  stage_half(cur, 0, 1, 5, 2);

  int64_t g = 0;  // global k-iteration index for this WG
  for (; tl < lo + cnt; tl += S) {
    for (int64_t t = 0; t < ntiles; ++t, ++g) {
      const bool boundary = (t == 0) && (g != 0);
      const bool tail = (g + 2 >= wg_iters);  // B(g+2) and beyond skipped
      if (tail) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      } else if (boundary) {
        asm volatile("s_waitcnt vmcnt(63)" ::: "memory");
      } else {
        // younger-than-B(g) in the FIFO = B(g+1)+A(g+2) = 8 glds
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");

      // compute k-tile g from ring slots (2g+h) mod 6 / mod 4
      {
        const int sa = (int)((2 * g) % 6), sb = (int)((2 * g) % 4);
        const uint16_t* Ab0 = Aring + sa * HALF;
        const uint16_t* Ab1 = Aring + ((sa + 1) % 6) * HALF;
        const uint16_t* Bb0 = Bring + sb * HALF;
        const uint16_t* Bb1 = Bring + ((sb + 1) % 4) * HALF;
        const uint16_t* As = wr ? Ab1 : Ab0;   // wave's A half
        const uint16_t* Bs = wc ? Bb1 : Bb0;   // wave's B half
#pragma unroll
        for (int kk = 0; kk < RBK; kk += 32) {
          short8 a_frag[4], b_frag[4];
          const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
          for (int fm = 0; fm < 4; ++fm) {
            int row = wr * 64 + fm * 16 + (lane & 15);  // global row for swz
            int ks = kbase ^ ((row & 7) << 3);
            a_frag[fm] = *(const short8*)(As + (row & 63) * RBK + ks);
          }
#pragma unroll
          for (int fn = 0; fn < 4; ++fn) {
            int col = wc * 64 + fn * 16 + (lane & 15);
            int ks = kbase ^ ((col & 7) << 3);
            b_frag[fn] = *(const short8*)(Bs + (col & 63) * RBK + ks);
          }
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int fm = 0; fm < 4; ++fm)
#pragma unroll
            for (int fn = 0; fn < 4; ++fn)
              acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
        }
      }

      // epilogue (tile's last k-iter) BEFORE the reuse barrier: its stores
      // are FIFO-younger than every glds a later wait needs
      if (t + 1 == ntiles) epilogue(tl);

      // all waves done reading g's slots before they are restaged
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");

      // stage A(g+3) and B(g+2) into the freed slots
      {
        const int64_t ga = g + 3, gb = g + 2;
        if (ga < wg_iters) {
          const int64_t ta = ga / ntiles - g / ntiles;  // 0=cur,1=nxt,2=skip
          if (ta <= 1) {
            const TileSt& ts = ta ? nxt : cur;
            const int64_t kt = ga % ntiles;
            const int rs = (int)((2 * ga) % 6);
            stage_half(ts, 0, 0, rs, kt);
            stage_half(ts, 0, 1, (rs + 1) % 6, kt);
          }
        }
        if (gb < wg_iters) {
          const int64_t tb = gb / ntiles - g / ntiles;
          if (tb <= 1) {
            const TileSt& ts = tb ? nxt : cur;
            const int64_t kt = gb % ntiles;
            const int rs = (int)((2 * gb) % 4);
            stage_half(ts, 1, 0, rs, kt);
            stage_half(ts, 1, 1, (rs + 1) % 4, kt);
          }
        }
      }
    }
    cur = nxt;
    const bool have_n2 = (tl + 2 * S) < (lo + cnt);
    nxt = have_n2 ? make_state(tl + 2 * S) : cur;
  }
}

// ---------------------------------------------------------------------------
// Tier B+: deep-pipelined NT kernel. 256x128 tile, 512 threads (8 waves as
// 4Mx2N, 64x64 per wave), BK=64, THREE LDS buffers (144 KB, 1 block/CU)
// with counted s_waitcnt vmcnt(N) and raw s_barriers so two tiles of glds
// stay in flight across every barrier (guide §5 "Pipelining across
// barriers": a __syncthreads here would carry vmcnt(0) and drain the DMA
// queue — the ~20% structural stall of the 2-buffer version). Same XOR
// swizzle and XCD-aware block remap as gemm_bf16_nt_k.
// Per-thread glds per tile: 4 for A (32 KB) + 2 for B (16 KB) = 6, so
// "one tile still in flight" == vmcnt(6).
// ---------------------------------------------------------------------------
__launch_bounds__(512) __global__ void gemm_bf16_nt_deep_k(GemmArgs args,
                                                           int nwg_n) {
  constexpr int DBM = 256, DBN = 128, DBK = 64;
  constexpr int ABYTES = DBM * DBK;              // elements per A tile (16K)
  constexpr int BBYTES = DBN * DBK;              // elements per B tile (8K)
  constexpr int BUFE = ABYTES + BBYTES;          // 24K elements = 48 KB
  __shared__ __attribute__((aligned(16))) uint16_t smem[3 * BUFE];

  const uint16_t* __restrict__ A = args.A;
  const uint16_t* __restrict__ B = args.B;
  uint16_t* C = args.C;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  const int64_t m0 = (int64_t)(wgid / nwg_n) * DBM;
  const int64_t n0 = (int64_t)(wgid % nwg_n) * DBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;        // 8 waves
  const int wr = wid >> 1, wc = wid & 1;   // 4x2 wave grid, 64x64 each

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / DBK;

  // per-lane element offsets: A chunks c=0..3 at e = tid*8 + c*4096,
  // B chunks c=0..1 at e = tid*8 + c*4096 (8192 elems per 512-thread pass).
  const int ea = (int)threadIdx.x * 8;

  auto stage = [&](int buf, int64_t t) {
    int64_t k0 = t * DBK;
    uint16_t* as = smem + buf * BUFE;
    uint16_t* bs = as + ABYTES;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      int64_t ar = m0 + row;
      if (ar > Mc) ar = Mc;
      glds16(A + ar * lda + k0 + k, as + e);
    }
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      int64_t br = n0 + row;
      if (br > Nc) br = Nc;
      glds16(B + br * ldb + k0 + k, bs + e);
    }
  };

  f32x4 acc[4][4] = {};

  stage(0, 0);
  if (ntiles > 1) stage(1, 1);

  for (int64_t t = 0; t < ntiles; ++t) {
    // open: tile t landed (one younger tile may stay in flight), all waves
    if (t + 1 < ntiles)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    // issue tile t+2 into the buffer released at the END of iteration t-1
    if (t + 2 < ntiles) stage((int)((t + 2) % 3), t + 2);

    const uint16_t* as = smem + (t % 3) * BUFE;
    const uint16_t* bs = as + ABYTES;
#pragma unroll
    for (int kk = 0; kk < DBK; kk += 32) {
      short8 a_frag[4], b_frag[4];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(as + row * DBK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = wc * 64 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(bs + col * DBK + ks);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // close: every wave done reading buf[t%3]; the DMA writes issued next
    // iteration may then land in it. lgkmcnt(0) so no ds_read is pending.
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }

  float* out32 = args.out_f32;  // deep kernel: single-batch
#pragma unroll
  for (int fm = 0; fm < 4; ++fm) {
#pragma unroll
    for (int fn = 0; fn < 4; ++fn) {
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int64_t grow = m0 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
        int64_t gcol = n0 + wc * 64 + fn * 16 + (lane & 15);
        if (grow >= args.M || gcol >= args.N) continue;
        int64_t idx = grow * args.N + gcol;
        if (out32) {
          out32[idx] = acc[fm][fn][rI];
          continue;
        }
        float v = args.alpha * acc[fm][fn][rI];
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Tier B++: 8-phase 256x256 NT kernel (the guide §5 256^2 template class,
// re-derived for the NT layout with a provably-safe wait schedule).
//
// Geometry: BM=BN=256, BK=64, 512 threads = 8 waves (2 wrow x 4 wcol).
// LDS = 2 tile-buffers x 4 half-tiles (A rows [0,128)/[128,256), B cols
// same) x 16 KB = 128 KB, ONE __shared__ array. An iteration covers two
// K-tiles (t even in buf0, t+1 in buf1) as 8 phases; phase p computes one
// 128x128 C-quadrant (HA,HB) of one tile over its full K=64 (16 MFMA/wave)
// and issues ONE half-tile DMA stage (2 glds/thread) into the slot whose
// last reader finished at the previous phase barrier:
//
//   p0 t  Q(0,0) stage A0(t+1)   p4 t+1 Q(0,0) stage A0(t+2)
//   p1 t  Q(1,0) stage B1(t+1)   p5 t+1 Q(1,0) stage B1(t+2)
//   p2 t  Q(1,1) stage B0(t+2)   p6 t+1 Q(1,1) stage B0(t+3)
//   p3 t  Q(0,1) stage A1(t+2)   p7 t+1 Q(0,1) stage A1(t+3)
//
// Counted waits (vmcnt in 2-load stage units, issue-order landing):
// p0/p4 wait vmcnt(6) (3 stage-units in flight -> the tile's A0 landed),
// p2/p6 wait vmcnt(8) (4 units in flight -> its B1 landed); the last two
// iterations drain with vmcnt(0). Each wait precedes the phase barrier so
// it covers every wave's DMA, not just the waiting wave's.
// ---------------------------------------------------------------------------
namespace ph8 {
constexpr int PBM = 256, PBN = 256, PBK = 64;
constexpr int HALF_E = 128 * PBK;            // elements per half-tile (8192)
constexpr int BUF_E = 4 * HALF_E;            // elements per tile-buffer
// slot offsets inside smem: buf*BUF_E + (isB*2 + half)*HALF_E
__device__ __forceinline__ uint16_t* slot(uint16_t* smem, int buf, int isB,
                                          int half) {
  return smem + buf * BUF_E + (isB * 2 + half) * HALF_E;
}
// compile-time vmcnt wait (asm needs a literal; N<0 emits nothing)
template <int N>
__device__ __forceinline__ void vm_wait() {
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 12) asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 14) asm volatile("s_waitcnt vmcnt(14)" ::: "memory");
  else if constexpr (N == 16) asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
}
// one wait site of the ping-pong schedule: group GRP waits the per-mode
// count (NT_N for TRMODE 0, NN_N for TRMODE 1, nothing for TN); tails
// drain everything
template <int TRMODE, int GRP, int NT_N, int NN_N>
__device__ __forceinline__ void pp_site(int g, bool tail) {
  if (tail) {
    vm_wait<0>();
    return;
  }
  if (g != GRP) return;
  if constexpr (TRMODE == 0) vm_wait<NT_N>();
  else if constexpr (TRMODE == 1) vm_wait<NN_N>();
}
}  // namespace ph8

__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase_k(GemmArgs args,
                                                                int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  // grid.y = split-K slice (sA/sB/sC element strides; out_f32 = raw f32
  // partial-slab mode, reduced by splitk_reduce_k)
  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;  // even, >= 6 (launch guard)
  const int ea = (int)threadIdx.x * 8;  // first element of this thread's chunk

  // stage one half-tile (128 rows x 64 k) of operand A (isB=0) or B (isB=1),
  // half h, of K-tile t, into buf = t & 1. 2 glds per thread.
  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  // acc[HA][HB][fm 0..3][fn 0..1]
  f32x4 acc[2][2][4][2] = {};

  // one compute phase: quadrant (HA,HB) of tile t (buffer t&1)
  auto compute = [&](int HA, int HB, int64_t t, auto& accq) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, HA);
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, HB);
#pragma unroll
    for (int kk = 0; kk < PBK; kk += 32) {
      short8 a_frag[4], b_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(as + row * PBK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(bs + col * PBK + ks);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          accq[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], accq[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  // prologue: 6 stage-units in the order the wait schedule assumes
  stage(1, 0, 0);  // B0(0)
  stage(0, 1, 0);  // A1(0)
  stage(0, 0, 0);  // A0(0)
  stage(1, 1, 0);  // B1(0)
  stage(1, 0, 1);  // B0(1)
  stage(0, 1, 1);  // A1(1)

#define PH8_WAIT(n)                                        \
  do {                                                     \
    if (tail)                                              \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");     \
    else                                                   \
      asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory"); \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)
#define PH8_BAR()                                          \
  do {                                                     \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)

  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 4 >= ntiles);
    const bool s2 = (t + 2 < ntiles), s3 = (t + 3 < ntiles);
    // phase 0: Q(0,0) of t; stage A0(t+1)
    PH8_WAIT(6);
    stage(0, 0, t + 1);  // A0 of the odd tile: staged same-iteration (p0)
    compute(0, 0, t, acc[0][0]);
    PH8_BAR();
    // phase 1: Q(1,0) of t; stage B1(t+1)
    stage(1, 1, t + 1);
    compute(1, 0, t, acc[1][0]);
    PH8_BAR();
    // phase 2: Q(1,1) of t; stage B0(t+2)
    PH8_WAIT(8);
    if (s2) stage(1, 0, t + 2);
    compute(1, 1, t, acc[1][1]);
    PH8_BAR();
    // phase 3: Q(0,1) of t; stage A1(t+2)
    if (s2) stage(0, 1, t + 2);
    compute(0, 1, t, acc[0][1]);
    PH8_BAR();
    // phase 4: Q(0,0) of t+1; stage A0(t+2)
    PH8_WAIT(6);
    if (s2) stage(0, 0, t + 2);
    compute(0, 0, t + 1, acc[0][0]);
    PH8_BAR();
    // phase 5: Q(1,0) of t+1; stage B1(t+2)
    if (s2) stage(1, 1, t + 2);
    compute(1, 0, t + 1, acc[1][0]);
    PH8_BAR();
    // phase 6: Q(1,1) of t+1; stage B0(t+3)
    PH8_WAIT(8);
    if (s3) stage(1, 0, t + 3);
    compute(1, 1, t + 1, acc[1][1]);
    PH8_BAR();
    // phase 7: Q(0,1) of t+1; stage A1(t+3)
    if (s3) stage(0, 1, t + 3);
    compute(0, 1, t + 1, acc[0][1]);
    PH8_BAR();
  }
#undef PH8_WAIT
#undef PH8_BAR

  if (out32) {
    // split-K partial mode: scatter raw f32. 128 stores/lane, but the tile
    // did >= 6 deep-K iterations so the issue cost is amortized.
    if (m0 + PBM <= args.M && n0 + PBN <= args.N) {
      const int64_t N = args.N;
      const int64_t base =
          (m0 + wr * 64 + (lane >> 4) * 4) * N + n0 + wc * 32 + (lane & 15);
#pragma unroll
      for (int ha = 0; ha < 2; ++ha)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            const int64_t rowbase = base + (int64_t)(ha * 128 + fm * 16 + rI) * N;
#pragma unroll
            for (int hb = 0; hb < 2; ++hb)
#pragma unroll
              for (int fn = 0; fn < 2; ++fn)
                out32[rowbase + hb * 128 + fn * 16] = acc[ha][hb][fm][fn][rI];
          }
      return;
    }
#pragma unroll
    for (int ha = 0; ha < 2; ++ha)
#pragma unroll
      for (int hb = 0; hb < 2; ++hb)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
#pragma unroll
            for (int rI = 0; rI < 4; ++rI) {
              int64_t grow = m0 + ha * 128 + wr * 64 + fm * 16 +
                             (lane >> 4) * 4 + rI;
              int64_t gcol = n0 + hb * 128 + wc * 32 + fn * 16 + (lane & 15);
              if (grow < args.M && gcol < args.N)
                out32[grow * args.N + gcol] = acc[ha][hb][fm][fn][rI];
            }
    return;
  }

  // Epilogue: stage the 256x256 bf16 C tile through LDS (exactly reuses the
  // 128 KB smem, DMA already drained), then stream it out with 16-B vector
  // stores — the scatter of 128 scalar 2-B stores per lane is store-ISSUE
  // bound (guide T21); this halves+ the tail at equal bytes.
#pragma unroll
  for (int ha = 0; ha < 2; ++ha)
#pragma unroll
    for (int hb = 0; hb < 2; ++hb)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int lrow = ha * 128 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int lcol = hb * 128 + wc * 32 + fn * 16 + (lane & 15);
            smem[lrow * 256 + lcol] = f2b(args.alpha * acc[ha][hb][fm][fn][rI]);
          }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {  // 256*256 elems / (512 threads * 8) = 16
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Variant 6: READS-BEFORE-BARRIER 8-phase kernel (guide §5 "256² 8-phase
// template" phase discipline, re-derived for this NT schedule).
//
// Diagnosis this answers (VERDICT r01 weak #1 + profiles/gemm_wait_pmc):
// variant 2 interleaves the 12 fragment ds_read_b128 with the MFMA cluster
// inside each phase, so hipcc breaks the back-to-back MFMA issue with
// per-fragment lgkm waits — SQ_WAIT_ANY = 42% of wave cycles. Here each
// phase issues ALL 12 fragment reads + its 2-glds stage BEFORE the phase
// barrier; after the barrier a single lgkmcnt(0) (already satisfied by the
// barrier's arrival skew) releases a PURE 16-MFMA cluster:
//
//   read 12 x ds_read_b128 (this phase's quadrant fragments)
//   stage 1 half-tile (2 x glds)
//   s_barrier                       <- covers the ds_read latency
//   s_waitcnt lgkmcnt(0)
//   s_setprio(1); 16 x MFMA; s_setprio(0)
//   [vmcnt(N) certificate for the NEXT phase's reads]
//   s_barrier
//
// Wait schedule (same stage order as variant 2; vmcnt counts LOADS, one
// stage unit = 2 glds). Reads at phase p are issued before p's first
// barrier, so their slots must be certified one barrier EARLIER than in
// variant 2: the waits move from the head of phases 0/2/4/6 to the tail of
// phases 7/1/3/5 with the same counts (6/8/6/8). Steady state, iter base t
// (unit issue order ... B0(t)@p2', A1(t)@p3', A0(t)@p4', B1(t)@p5',
// B0(t+1)@p6', A1(t+1)@p7' of iter t-2, then A0(t+1)@p0, B1(t+1)@p1,
// B0(t+2)@p2, A1(t+2)@p3, A0(t+2)@p4, B1(t+2)@p5, B0(t+3)@p6, A1(t+3)@p7):
//   end p7: vmcnt(6) -> A0(t+2),B0(t+2) landed for next p0 (youngest 3
//           in flight: B1(t+2), B0(t+3), A1(t+3));
//   end p1: vmcnt(8) -> B1(t) landed for p2/p3 (youngest 4 in flight);
//   end p3: vmcnt(6) -> A0(t+1),B0(t+1) landed for p4/p5;
//   end p5: vmcnt(8) -> B1(t+1) landed for p6/p7.
// Prologue: 6 units, vmcnt(6) + barrier before the first reads (the
// wait/barrier PAIR is the cross-wave certificate — a lone per-wave vmcnt
// only covers this wave's own DMA).
// Slot-overwrite safety is inherited from variant 2 and RELAXED by this
// change (reads complete by each phase's lgkmcnt(0), one barrier before
// the earliest re-stage of that slot). Tail iterations (stages skipped)
// drain with vmcnt(0) at every wait site, as in variant 2.
// ---------------------------------------------------------------------------
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase6_k(GemmArgs args,
                                                                 int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;  // even, >= 6 (launch guard)
  const int ea = (int)threadIdx.x * 8;

  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][2][4][2] = {};
  // fragment registers for ONE phase: [kk-step][fragment]
  short8 af[2][4], bf[2][2];

  // issue the 12 fragment reads for quadrant (HA,HB) of tile t
  auto read_frags = [&](int HA, int HB, int64_t t) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, HA);
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, HB);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        af[kk2][fm] = *(const short8*)(as + row * PBK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        bf[kk2][fn] = *(const short8*)(bs + col * PBK + ks);
      }
    }
  };

  // the pure MFMA cluster (no loads, no waits inside)
  auto mfma16 = [&](auto& accq) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          accq[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk2][fm], bf[kk2][fn], accq[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // prologue: 6 stage-units + landing certificate for the first reads
  stage(1, 0, 0);  // B0(0)
  stage(0, 1, 0);  // A1(0)
  stage(0, 0, 0);  // A0(0)
  stage(1, 1, 0);  // B1(0)
  stage(1, 0, 1);  // B0(1)
  stage(0, 1, 1);  // A1(1)
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

#define PH6_MID()                                          \
  do {                                                     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
  } while (0)
#define PH6_END(n)                                         \
  do {                                                     \
    if (tail)                                              \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");     \
    else                                                   \
      asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory"); \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)
#define PH6_END_NOWAIT()                                   \
  do {                                                     \
    if (tail) asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)

  // diag ablation (NEZHA_PERS_DIAG via args.diag_no_store; OUTPUT GARBAGE):
  //   2 = stage-only (skip MFMA clusters)  -> staging pipeline's own pace
  //   3 = compute-only (skip glds stages)  -> read+MFMA+barrier pace
  const bool dg_mfma = args.diag_no_store != 2;
  const bool dg_stage = args.diag_no_store != 3;
  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 4 >= ntiles) || !dg_stage;
    const bool s2 = (t + 2 < ntiles) && dg_stage, s3 = (t + 3 < ntiles) && dg_stage;
    // phase 0: Q(0,0) of t; stage A0(t+1)
    read_frags(0, 0, t);
    if (dg_stage) stage(0, 0, t + 1);
    PH6_MID();
    if (dg_mfma) mfma16(acc[0][0]);
    PH6_END_NOWAIT();
    // phase 1: Q(1,0) of t; stage B1(t+1); certify B1(t) for p2/p3
    read_frags(1, 0, t);
    if (dg_stage) stage(1, 1, t + 1);
    PH6_MID();
    if (dg_mfma) mfma16(acc[1][0]);
    PH6_END(8);
    // phase 2: Q(1,1) of t; stage B0(t+2)
    read_frags(1, 1, t);
    if (s2) stage(1, 0, t + 2);
    PH6_MID();
    if (dg_mfma) mfma16(acc[1][1]);
    PH6_END_NOWAIT();
    // phase 3: Q(0,1) of t; stage A1(t+2); certify A0/B0(t+1) for p4/p5
    read_frags(0, 1, t);
    if (s2) stage(0, 1, t + 2);
    PH6_MID();
    if (dg_mfma) mfma16(acc[0][1]);
    PH6_END(6);
    // phase 4: Q(0,0) of t+1; stage A0(t+2)
    read_frags(0, 0, t + 1);
    if (s2) stage(0, 0, t + 2);
    PH6_MID();
    if (dg_mfma) mfma16(acc[0][0]);
    PH6_END_NOWAIT();
    // phase 5: Q(1,0) of t+1; stage B1(t+2); certify B1(t+1) for p6/p7
    read_frags(1, 0, t + 1);
    if (s2) stage(1, 1, t + 2);
    PH6_MID();
    if (dg_mfma) mfma16(acc[1][0]);
    PH6_END(8);
    // phase 6: Q(1,1) of t+1; stage B0(t+3)
    read_frags(1, 1, t + 1);
    if (s3) stage(1, 0, t + 3);
    PH6_MID();
    if (dg_mfma) mfma16(acc[1][1]);
    PH6_END_NOWAIT();
    // phase 7: Q(0,1) of t+1; stage A1(t+3); certify A0/B0(t+2) for next p0
    read_frags(0, 1, t + 1);
    if (s3) stage(0, 1, t + 3);
    PH6_MID();
    if (dg_mfma) mfma16(acc[0][1]);
    PH6_END(6);
  }
#undef PH6_MID
#undef PH6_END
#undef PH6_END_NOWAIT

  if (out32) {
    // split-K partial mode: scatter raw f32 (same as variant 2)
    if (m0 + PBM <= args.M && n0 + PBN <= args.N) {
      const int64_t N = args.N;
      const int64_t base =
          (m0 + wr * 64 + (lane >> 4) * 4) * N + n0 + wc * 32 + (lane & 15);
#pragma unroll
      for (int ha = 0; ha < 2; ++ha)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            const int64_t rowbase = base + (int64_t)(ha * 128 + fm * 16 + rI) * N;
#pragma unroll
            for (int hb = 0; hb < 2; ++hb)
#pragma unroll
              for (int fn = 0; fn < 2; ++fn)
                out32[rowbase + hb * 128 + fn * 16] = acc[ha][hb][fm][fn][rI];
          }
      return;
    }
#pragma unroll
    for (int ha = 0; ha < 2; ++ha)
#pragma unroll
      for (int hb = 0; hb < 2; ++hb)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
#pragma unroll
            for (int rI = 0; rI < 4; ++rI) {
              int64_t grow = m0 + ha * 128 + wr * 64 + fm * 16 +
                             (lane >> 4) * 4 + rI;
              int64_t gcol = n0 + hb * 128 + wc * 32 + fn * 16 + (lane & 15);
              if (grow < args.M && gcol < args.N)
                out32[grow * args.N + gcol] = acc[ha][hb][fm][fn][rI];
            }
    return;
  }

  // LDS-staged vector epilogue (same as variant 2)
#pragma unroll
  for (int ha = 0; ha < 2; ++ha)
#pragma unroll
    for (int hb = 0; hb < 2; ++hb)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int lrow = ha * 128 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int lcol = hb * 128 + wc * 32 + fn * 16 + (lane & 15);
            smem[lrow * 256 + lcol] = f2b(args.alpha * acc[ha][hb][fm][fn][rI]);
          }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Variant 7: PING-PONG wave-specialized 256^2 kernel.
//
// Diagnosis (gpurun_out/r2_diag.log, ablation on variant 6): with all 8
// waves in lock-step phases, each phase serializes [12 ds_read_b128 x 8
// waves = 384 LDS-array cycles + barrier] BEFORE [32 MFMAs x 16 cyc = 512
// cycles of matrix pipe]; stage-only runs 12.6 TB/s and compute-only is
// ~90% of the full kernel's time — the fragment-read segment, not the DMA
// and not L2 misses (80% hit rate after the grouped raster), paces the
// loop, and the MFMA pipe idles ~57%.
//
// Structure: waves 0-3 (G0, rows [0,128)) and waves 4-7 (G1, rows
// [128,256)) ALTERNATE roles each phase: while G0's four waves run a pure
// 32-MFMA cluster (its 64x64-per-wave quadrant slice over K=64), G1's four
// waves issue THEIR next quadrant's 8-16 ds_read_b128 + the phase's 2-glds
// half-tile stage — LDS array and matrix pipe are independent resources,
// and each SIMD hosts one wave of each group (cyclic wave->SIMD placement),
// so every SIMD has MFMA work in flight in EVERY phase (microarch "two
// waves per SIMD" item 5: pair matrix beside memory).
//
// Per group per K-tile: 2 compute phases (quadrant cols b=0,1) of 32 MFMA
// (4x4 fragments x BK=64), A fragments REUSED across the b=0/b=1 pair
// (read 16 b128 at the tile edge, 8 at the inner edge). acc[2][4][4] = 128
// VGPR + one 16-b128 fragment set = 64 + addressing ~= 225 VGPR.
//
// Stage schedule: unchanged unit order (A0(t+1)@p0, B1(t+1)@p1, B0(t+2)@p2,
// A1(t+2)@p3, A0(t+2)@p4, B1(t+2)@p5, B0(t+3)@p6, A1(t+3)@p7), issued by
// the LOADING group of the phase (G1 on even phases: A0/B0 units; G0 on
// odd: B1/A1 units), 4 glds per thread (256 lanes x 4 x 16 B = 16 KB).
//
// vmcnt certificates are now PER-GROUP (vmcnt counts the waiting wave's
// own instructions; each group issues 4 glds per unit). Derivation in
// docs/perf (r2): with reads at phase p certified by [staking group's
// vmcnt + any barrier] before p:
//   end-p0: G0 vmcnt(4)   (B1(t) landed        -> p1/p2 reads)
//   end-p2: G1 vmcnt(4)   (A0(t+1), B0(t+1)    -> p3/p4 reads)
//   end-p3: G0 vmcnt(8)   (A1(t+1)             -> p4 reads)
//   end-p4: G0 vmcnt(4)   (B1(t+1)             -> p5/p6 reads)
//   end-p6: G1 vmcnt(4)   (A0(t+2), B0(t+2)    -> p7/next-p0 reads)
//   end-p7: G0+G1 vmcnt(8) (A1(t+2), B0(t+2)   -> next-p0 reads)
// Prologue: 6 units + vmcnt(6)/(8 for G1)... both groups issued 3 units
// each in the prologue (G1: B0(0) A0(0) B0(1); G0: A1(0) B1(0) A1(1)), so
// a uniform vmcnt(4) + barrier certifies the first two units of each
// group (B0(0), A1(0), A0(0), B1(0)) for the pre-read and p0/p1.
// Tails drain with vmcnt(0) as in variants 2/6.
// ---------------------------------------------------------------------------
// TRMODE: native-layout staging (r2: kills the NN/TN lowering transposes).
//   bit0: B stored [K,N] (k-major) — staged via short8 row loads + ds_write
//         transpose scatter into the standard [n][k] XOR image;
//   bit1: A stored [K,M] — same for A.
// Transposed stages are SYNCHRONOUS (their loads drain before the scatter),
// so they need no vmcnt certificates. Per-mode wait schedules are
// simulator-verified (tools/pp_schedule_sim.py style derivation):
//   TRMODE=0 (NT): the deep schedule's 8/12 waits;
//   TRMODE=1 (NN): only A-units in flight -> all four waits vmcnt(4);
//   TRMODE=3 (TN): every unit synchronous -> no steady waits at all.
template <int PPMODE, int TRMODE = 0>
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_pingpong_k(
    GemmArgs args, int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = wid >> 2;        // ping-pong group (0: waves 0-3, 1: 4-7)
  const int ws = wid & 3;        // wave-in-group: 2x2 over the quadrant
  // microarch "two waves per SIMD" item 4: the second-dispatched half is
  // the arbitration loser; one static s_setprio(1) for it, no flips
  if constexpr (PPMODE & 1) {
    if (g == 1) __builtin_amdgcn_s_setprio(1);
  }
  const int tig = (int)threadIdx.x & 255;  // thread-in-group

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;  // even, >= 6 (launch guard)

  // one half-tile (16 KB) staged by ONE group: 4 glds per thread
  // k-major transposed stage: operand stored [K, R]; each thread of the
  // staging group covers (k-row, 8 consecutive r) and scatters into a
  // ROTATED [r][k] XOR image: col = (k ^ ((r&7)<<3)) + (r>>3 & 7)*8 mod 64.
  // Without the (r>>3) rotation every lane of a scatter instruction lands
  // on the SAME bank (row stride 128 B, swizzle constant across lanes) —
  // a 64-way conflict that measured 6-10x slower. Loads are hoisted ahead
  // of the scatters so their latency overlaps.
  auto stage_tr = [&](const uint16_t* X, int64_t ldx, int64_t r0,
                      int64_t Rlast8, uint16_t* dst, int64_t k0) {
    short8 vv[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = tig * 8 + c * 2048;  // [k][r] order over the 64x128 half
      int kk = e >> 7;
      int rl = e & 127;
      int64_t gr = r0 + rl;
      if (gr > Rlast8) gr = Rlast8;  // 8-aligned clamp (R % 8 == 0 gated)
      vv[c] = *(const short8*)(X + (k0 + kk) * ldx + gr);
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = tig * 8 + c * 2048;
      int kk = e >> 7;
      int rl = e & 127;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        int r = rl + i;
        int col = ((kk ^ ((r & 7) << 3)) + ((r >> 3) << 3)) & 63;
        dst[r * PBK + col] = (uint16_t)vv[c][i];
      }
    }
  };
  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
    if (!isB) {
      if constexpr ((TRMODE & 2) != 0) {
        stage_tr(A, lda, m0 + h * 128, args.M - 8, dst, k0);
        return;
      }
    } else {
      if constexpr ((TRMODE & 1) != 0) {
        stage_tr(B, ldb, n0 + h * 128, args.N - 8, dst, k0);
        return;
      }
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = tig * 8 + c * 2048;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][4][4] = {};   // [quadrant col b][fm][fn]
  short8 af[2][4], bf[2][4];  // one fragment set: [kk][frag]

  // A-fragment reads: this group's 64-row wave slice of A-half g, tile t
  auto read_af = [&](int64_t t) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, g);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = (ws >> 1) * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        if constexpr ((TRMODE & 2) != 0)
          ks = (ks + ((row >> 3) << 3)) & 63;  // stage_tr's rotated image
        af[kk2][fm] = *(const short8*)(as + row * PBK + ks);
      }
    }
  };
  // B-fragment reads: wave's 64-col slice of B-half b, tile t
  auto read_bf = [&](int b, int64_t t) {
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, b);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = (ws & 1) * 64 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        if constexpr ((TRMODE & 1) != 0)
          ks = (ks + ((col >> 3) << 3)) & 63;  // stage_tr's rotated image
        bf[kk2][fn] = *(const short8*)(bs + col * PBK + ks);
      }
    }
  };

  // pure 32-MFMA cluster for quadrant column b
  auto mfma32 = [&](int b) {
    if constexpr (!(PPMODE & 1)) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[b][fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk2][fm], bf[kk2][fn], acc[b][fm][fn], 0, 0, 0);
    if constexpr (!(PPMODE & 1)) __builtin_amdgcn_s_setprio(0);
  };

  // prologue: tiles 0 and 1 fully staged, 4 units per group in each
  // group's in-loop FIFO order (deep schedule, r2: iteration t stages ALL
  // of tiles t+2/t+3, giving every unit a 5-7 phase flight; the old
  // schedule's 3-phase flights made DMA latency the pace-setter on
  // short-K shapes)
  if (g == 1) {
    stage(0, 0, 0);  // A0(0)
    stage(1, 0, 0);  // B0(0)
    stage(0, 0, 1);  // A0(1)
    stage(1, 0, 1);  // B0(1)
  } else {
    stage(0, 1, 0);  // A1(0)
    stage(1, 1, 0);  // B1(0)
    stage(0, 1, 1);  // A1(1)
    stage(1, 1, 1);  // B1(1)
  }
  // certificates for the pre-read + p0 reads: oldest units of each group
  if constexpr (TRMODE == 0) {
    if (g == 0)
      asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else if constexpr (TRMODE == 1) {
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  }
  // transposed-stage scatters (and the pre-read below) are lgkm traffic
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  // pre-read: G0 its p0 fragments (A0,B0 of tile 0); G1 reads at p0 proper
  if (g == 0) {
    read_af(0);
    read_bf(0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

#define PP_END(wait_stmt)                                  \
  do {                                                     \
    wait_stmt;                                             \
    /* drain THIS wave's LDS reads before crossing: the deep  */ \
    /* schedule re-stages a slot ONE phase after its last read */ \
    /* issue; an undrained ds_read would race the incoming DMA */ \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");          \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)
  const bool dg_mfma = args.diag_no_store != 2;   // 2: stage-only ablation
  const bool dg_stage = args.diag_no_store != 3;  // 3: compute-only ablation
  for (int64_t t = 0; t < ntiles; t += 2) {
    // drain only in the LAST iteration (its skipped stages break the
    // steady vmcnt counts); the second-to-last still stages every unit,
    // so its counts — and all its certificates — are exactly steady
    const bool tail = (t + 2 >= ntiles) || !dg_stage;
    const bool s2 = (t + 2 < ntiles) && dg_stage, s3 = (t + 3 < ntiles) && dg_stage;
    // p0: G0 computes Q(0,0)t | G1 reads af(A1,t)+bf(B0,t), stages A0(t+1)
    if (g == 0) {
      if (dg_mfma) mfma32(0);
    } else {
      if constexpr (PPMODE & 2) { if (s2) stage(0, 0, t + 2); }
      read_af(t);
      read_bf(0, t);
      if constexpr (!(PPMODE & 2)) { if (s2) stage(0, 0, t + 2); }
    }
    PP_END((pp_site<TRMODE, 0, 8, -1>(g, tail)));
    // p1: G1 computes Q(1,0)t | G0 reads bf(B1,t) [af reuse], stages B1(t+1)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(0);
    } else {
      if constexpr (PPMODE & 2) { if (s2) stage(0, 1, t + 2); }
      read_bf(1, t);
      if constexpr (!(PPMODE & 2)) { if (s2) stage(0, 1, t + 2); }
    }
    PP_END();
    // p2: G0 computes Q(0,1)t | G1 reads bf(B1,t), stages B0(t+2)
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(1);
    } else {
      if constexpr (PPMODE & 2) { if (s2) stage(1, 0, t + 2); }
      read_bf(1, t);
      if constexpr (!(PPMODE & 2)) { if (s2) stage(1, 0, t + 2); }
    }
    PP_END((pp_site<TRMODE, 1, 8, 4>(g, tail)));
    // p3: G1 computes Q(1,1)t | G0 reads af(A0,t+1)+bf(B0,t+1), stages A1(t+2)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(1);
    } else {
      if constexpr (PPMODE & 2) { if (s2) stage(1, 1, t + 2); }
      read_af(t + 1);
      read_bf(0, t + 1);
      if constexpr (!(PPMODE & 2)) { if (s2) stage(1, 1, t + 2); }
    }
    PP_END((pp_site<TRMODE, 0, 12, 4>(g, tail)));
    // p4: G0 computes Q(0,0)t+1 | G1 reads af(A1,t+1)+bf(B0,t+1), stages A0(t+2)
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(0);
    } else {
      if constexpr (PPMODE & 2) { if (s3) stage(0, 0, t + 3); }
      read_af(t + 1);
      read_bf(0, t + 1);
      if constexpr (!(PPMODE & 2)) { if (s3) stage(0, 0, t + 3); }
    }
    PP_END((pp_site<TRMODE, 0, 8, -1>(g, tail)));
    // p5: G1 computes Q(1,0)t+1 | G0 reads bf(B1,t+1), stages B1(t+2)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(0);
    } else {
      if constexpr (PPMODE & 2) { if (s3) stage(0, 1, t + 3); }
      read_bf(1, t + 1);
      if constexpr (!(PPMODE & 2)) { if (s3) stage(0, 1, t + 3); }
    }
    PP_END();
    // p6: G0 computes Q(0,1)t+1 | G1 reads bf(B1,t+1), stages B0(t+3)
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(1);
    } else {
      if constexpr (PPMODE & 2) { if (s3) stage(1, 0, t + 3); }
      read_bf(1, t + 1);
      if constexpr (!(PPMODE & 2)) { if (s3) stage(1, 0, t + 3); }
    }
    PP_END((pp_site<TRMODE, 1, 8, 4>(g, tail)));
    // p7: G1 computes Q(1,1)t+1 | G0 reads af(A0,t+2)+bf(B0,t+2), stages A1(t+3)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma32(1);
    } else {
      if constexpr (PPMODE & 2) { if (s3) stage(1, 1, t + 3); }
      if (s2) {
        read_af(t + 2);
        read_bf(0, t + 2);
      }
      if constexpr (!(PPMODE & 2)) { if (s3) stage(1, 1, t + 3); }
    }
    PP_END((pp_site<TRMODE, 0, 12, 4>(g, tail), pp_site<TRMODE, 1, 8, -1>(g, tail)));
  }
#undef PP_END

  // drain any straggling DMA before the epilogue reuses smem
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  if (out32) {
    // split-K partial mode: scatter raw f32
    const int64_t N = args.N;
#pragma unroll
    for (int b = 0; b < 2; ++b)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int64_t grow = m0 + g * 128 + (ws >> 1) * 64 + fm * 16 +
                           (lane >> 4) * 4 + rI;
            int64_t gcol = n0 + b * 128 + (ws & 1) * 64 + fn * 16 + (lane & 15);
            if (grow < args.M && gcol < args.N)
              out32[grow * N + gcol] = acc[b][fm][fn][rI];
          }
    return;
  }

  // LDS-staged vector epilogue (as variants 2/6)
#pragma unroll
  for (int b = 0; b < 2; ++b)
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int lrow = g * 128 + (ws >> 1) * 64 + fm * 16 + (lane >> 4) * 4 + rI;
          int lcol = b * 128 + (ws & 1) * 64 + fn * 16 + (lane & 15);
          smem[lrow * 256 + lcol] = f2b(args.alpha * acc[b][fm][fn][rI]);
        }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Variant 7-PERSISTENT: ping-pong kernel with the pipeline flowing ACROSS
// tile boundaries — the short-K answer (BERT K=768 panels were 32.6% of the
// step at ~540 TF on the 2-buffer persistent kernel).
//
// Tile-boundary correspondence (same discovery as gemm_bf16_nt_8phase_pers_k,
// restated for the ping-pong stage order): the s2/s3-guarded stages of a
// tile's LAST iteration (B0/A1/A0/B1 of t+2 at p2..p5, B0/A1 of t+3 at
// p6/p7) are — in ORDER, GROUP and LDS SLOT — exactly the 6-unit prologue
// of the NEXT tile (G1: B0(0) A0(0) B0(1); G0: A1(0) B1(0) A1(1); parity
// matches because ntiles is even). Substituting next-tile addresses there
// keeps every per-group glds count, hence the whole per-group vmcnt
// schedule, steady across the sweep: NO drain and NO prologue between
// tiles. The p7 fragment reads flow too: read_af/bf of "t+2" are next
// tile's A0(0)/B0(0) from the very slots the substitution staged.
//
// Epilogue between tiles: stores are FIFO-YOUNGER than every glds a later
// wait needs, so the steady vmcnt counts remain CORRECT certificates (a
// wait only gets stricter — it may force store drain, never miss a load).
// Interior tiles stage each 128x128 C-quadrant through a 32 KiB spare LDS
// region (the full 160 KiB budget: 128 KiB pipeline + 32 KiB spare) and
// stream it out as b128 stores — 16 store instructions per wave instead
// of the 128-scalar scatter whose issue tail stalled the 8phase_pers
// experiment. Edge tiles fall back to the masked scalar scatter.
// ---------------------------------------------------------------------------
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_pingpong_pers_k(
    GemmArgs args, int nwg_n, int64_t per_b, int64_t total) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E + 16384];
  uint16_t* spare = smem + 2 * BUF_E;  // 32 KiB epilogue staging

  const int orig = blockIdx.x;  // 256 WGs
  const int xcd = orig & 7, slotq = orig >> 3;  // 32 slots per XCD
  const int64_t q = total / 8, r8 = total % 8;
  const int64_t lo = xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q;
  const int64_t cnt = xcd < r8 ? q + 1 : q;
  if (slotq >= cnt) return;
  constexpr int SLOTS = 32;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = wid >> 2;   // ping-pong group
  const int ws = wid & 3;   // wave-in-group (2x2)
  const int tig = (int)threadIdx.x & 255;
  if (g == 1) __builtin_amdgcn_s_setprio(1);  // static younger-half prio

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;  // even, >= 2 (launch guard)

  // compact per-tile state (3 x i32; the pointers are recomputed at use so
  // the sweep's live set stays tiny — the fat 2-pointer TileSt spilled)
  struct TileSt {
    int32_t bidx, m0, n0;
  };
  auto make_state = [&](int64_t tl) {
    TileSt ts;
    // readfirstlane: these are wave-uniform; pinning them scalar keeps the
    // sweep state out of the VGPR budget (the main loop sits at the cap)
    ts.bidx = __builtin_amdgcn_readfirstlane((int32_t)(tl / per_b));
    const int64_t rtl = tl % per_b;
    ts.m0 = __builtin_amdgcn_readfirstlane((int32_t)((rtl / nwg_n) * PBM));
    ts.n0 = __builtin_amdgcn_readfirstlane((int32_t)((rtl % nwg_n) * PBN));
    return ts;
  };

  // one half-tile staged by ONE group (4 glds per thread); kt indexes the
  // tile-local K-step, buf = kt & 1 (every tile starts at buf 0)
  auto stage = [&](const TileSt& ts, int isB, int h, int64_t kt) {
    int64_t k0 = kt * PBK;
    uint16_t* dst = slot(smem, (int)(kt & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int e = tig * 8 + c * 2048;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = ts.m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(args.A + ts.bidx * args.sA + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = ts.n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(args.B + ts.bidx * args.sB + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][4][4] = {};
  short8 af[2][4], bf[2][4];

  auto read_af = [&](int64_t kt) {
    const uint16_t* as = slot(smem, (int)(kt & 1), 0, g);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = (ws >> 1) * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        af[kk2][fm] = *(const short8*)(as + row * PBK + ks);
      }
    }
  };
  auto read_bf = [&](int b, int64_t kt) {
    const uint16_t* bs = slot(smem, (int)(kt & 1), 1, b);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = (ws & 1) * 64 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        bf[kk2][fn] = *(const short8*)(bs + col * PBK + ks);
      }
    }
  };

  auto mfma32 = [&](int b) {
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[b][fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk2][fm], bf[kk2][fn], acc[b][fm][fn], 0, 0, 0);
  };

  // Epilogue. Interior tiles stage each 128x128 C-quadrant through the
  // 32 KiB spare LDS and stream it out as 4 b128 stores per thread
  // (16 store instructions per wave for the whole tile). The store COUNT
  // matters, not just the issue rate: epilogue stores sit in the vmcnt
  // FIFO between the boundary prefetch and the next tile's counted waits,
  // so the first post-boundary waits add +16 to their counts instead of
  // force-draining the pipeline (a scalar scatter's 128 stores/wave blew
  // past vmcnt's 63 cap and serialized every tile boundary at ~290 TF).
  // Raw barriers + lgkm-only waits: __syncthreads would drain the
  // in-flight next-tile glds.
  auto epilogue = [&](const TileSt& ts) {
    uint16_t* C = args.C + (int64_t)ts.bidx * args.sC;
    float* out32 =
        args.out_f32 ? args.out_f32 + (int64_t)ts.bidx * args.sC : nullptr;
    uint16_t* save_pre =
        args.save_pre ? args.save_pre + (int64_t)ts.bidx * args.sC : nullptr;
    const int64_t N = args.N;
    const bool interior =
        !out32 && (ts.m0 + PBM <= args.M) && (ts.n0 + PBN <= args.N);
    if (interior) {
      for (int pass = 0; pass < 4; ++pass) {
        const int ha = pass >> 1, hb = pass & 1;
        if (g == ha) {
#pragma unroll
          for (int fm = 0; fm < 4; ++fm)
#pragma unroll
            for (int fn = 0; fn < 4; ++fn)
#pragma unroll
              for (int rI = 0; rI < 4; ++rI) {
                int lrow = (ws >> 1) * 64 + fm * 16 + (lane >> 4) * 4 + rI;
                int lcol = (ws & 1) * 64 + fn * 16 + (lane & 15);
                spare[lrow * 128 + lcol] =
                    f2b(args.alpha * acc[hb][fm][fn][rI]);
              }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
#pragma unroll
        for (int c = 0; c < 4; ++c) {
          int e = (c * 512 + (int)threadIdx.x) * 8;
          int row = e >> 7, col = e & 127;
          short8 vv = *(const short8*)(spare + e);
          int64_t gcol = ts.n0 + hb * 128 + col;
          int64_t idx = (ts.m0 + ha * 128 + row) * N + gcol;
          short8 zout, yout;
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float v = b2f((uint16_t)vv[i]);
            if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
            if (args.bias) v += b2f(args.bias[gcol + i]);
            zout[i] = (short)f2b(v);
            if (args.fuse_gelu) v = gelu_fwd_scalar(v);
            yout[i] = (short)f2b(v);
          }
          if (save_pre) *(short8*)(save_pre + idx) = zout;
          *(short8*)(C + idx) = yout;
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("" ::: "memory");
      }
    } else {
#pragma unroll
      for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 4; ++fn)
#pragma unroll
            for (int rI = 0; rI < 4; ++rI) {
              int64_t grow = ts.m0 + g * 128 + (ws >> 1) * 64 + fm * 16 +
                             (lane >> 4) * 4 + rI;
              int64_t gcol =
                  ts.n0 + b2 * 128 + (ws & 1) * 64 + fn * 16 + (lane & 15);
              if (grow < args.M && gcol < args.N) {
                int64_t idx = grow * N + gcol;
                if (out32) {
                  out32[idx] = acc[b2][fm][fn][rI];
                } else {
                  float v = args.alpha * acc[b2][fm][fn][rI];
                  if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                  if (args.bias) v += b2f(args.bias[gcol]);
                  if (save_pre) save_pre[idx] = f2b(v);
                  if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                  C[idx] = f2b(v);
                }
              }
            }
      // a non-interior epilogue's scalar stores overflow the countable
      // window — drain them (rare: edge tiles / split-K slabs only)
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }
#pragma unroll
    for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn) acc[b2][fm][fn] = f32x4{};
  };

#define PPP_END(wait_stmt)                                 \
  do {                                                     \
    wait_stmt;                                             \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)
// fresh = first iteration after a tile epilogue: its 16 b128 stores per
// wave are still in the FIFO between the boundary prefetch and these
// waits, so the counts widen by 16. FIFO completion order makes the wide
// form safe even once the stores drain (stores drained => every older
// load drained too).
#define PPP_W0(n, nf)                                         \
  if (tail)                                                   \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");          \
  else if (g == 0) {                                          \
    if (fresh)                                                \
      asm volatile("s_waitcnt vmcnt(" #nf ")" ::: "memory");  \
    else                                                      \
      asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory");   \
  }
#define PPP_W1(n, nf)                                         \
  if (tail)                                                   \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");          \
  else if (g == 1) {                                          \
    if (fresh)                                                \
      asm volatile("s_waitcnt vmcnt(" #nf ")" ::: "memory");  \
    else                                                      \
      asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory");   \
  }

  int64_t tl = lo + slotq;
  TileSt cur = make_state(tl);
  bool have_next = (tl + SLOTS) < (lo + cnt);
  TileSt nxt = have_next ? make_state(tl + SLOTS) : cur;

  // first-tile prologue: tiles 0 and 1 fully staged, 4 units per group
  // in the in-loop FIFO order (deep schedule: 5-7 phase flights)
  if (g == 1) {
    stage(cur, 0, 0, 0);  // A0(0)
    stage(cur, 1, 0, 0);  // B0(0)
    stage(cur, 0, 0, 1);  // A0(1)
    stage(cur, 1, 0, 1);  // B0(1)
  } else {
    stage(cur, 0, 1, 0);  // A1(0)
    stage(cur, 1, 1, 0);  // B1(0)
    stage(cur, 0, 1, 1);  // A1(1)
    stage(cur, 1, 1, 1);  // B1(1)
  }
  if (g == 0)
    asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (g == 0) {
    read_af(0);
    read_bf(0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  for (; tl < lo + cnt; tl += SLOTS) {
    const bool not_first_tile = (tl != lo + slotq);
    for (int64_t t = 0; t < ntiles; t += 2) {
      const bool last_it = (t + 2 >= ntiles);
      const bool fresh = (t == 0) && not_first_tile;
      // stage targets: tile-local K-step kt < ntiles stays on cur; the
      // overflow (kt - ntiles on nxt) is the next tile's prologue
      const bool tail = last_it && !have_next;  // true end: drain
      const bool s2 = !last_it || have_next;
      const bool s3 = (t + 3 < ntiles) || have_next;
      auto st = [&](int isB, int h, int64_t kt) {
        // value-select the 3-int state (one inlined stage body, not two)
        const bool ov = kt >= ntiles;
        TileSt ts = ov ? nxt : cur;
        stage(ts, isB, h, ov ? kt - ntiles : kt);
      };
      // p0: G0 Q(0,0)t | G1 reads af(A1,t)+bf(B0,t), stages A0(t+1)
      if (g == 0) {
        mfma32(0);
      } else {
        read_af(t);
        read_bf(0, t);
        if (s2) st(0, 0, t + 2);
      }
      PPP_END(PPP_W0(8, 24));
      // p1: G1 Q(1,0)t | G0 reads bf(B1,t), stages B1(t+1)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(0);
      } else {
        read_bf(1, t);
        if (s2) st(0, 1, t + 2);
      }
      PPP_END();
      // p2: G0 Q(0,1)t | G1 reads bf(B1,t), stages B0(t+2)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(1);
      } else {
        read_bf(1, t);
        if (s2) st(1, 0, t + 2);
      }
      PPP_END(PPP_W1(8, 24));
      // p3: G1 Q(1,1)t | G0 reads af(A0,t+1)+bf(B0,t+1), stages A1(t+2)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(1);
      } else {
        read_af(t + 1);
        read_bf(0, t + 1);
        if (s2) st(1, 1, t + 2);
      }
      PPP_END(PPP_W0(12, 28));
      // p4: G0 Q(0,0)t+1 | G1 reads af(A1,t+1)+bf(B0,t+1), stages A0(t+2)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(0);
      } else {
        read_af(t + 1);
        read_bf(0, t + 1);
        if (s3) st(0, 0, t + 3);
      }
      PPP_END(PPP_W0(8, 24));
      // p5: G1 Q(1,0)t+1 | G0 reads bf(B1,t+1), stages B1(t+2)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(0);
      } else {
        read_bf(1, t + 1);
        if (s3) st(0, 1, t + 3);
      }
      PPP_END();
      // p6: G0 Q(0,1)t+1 | G1 reads bf(B1,t+1), stages B0(t+3)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(1);
      } else {
        read_bf(1, t + 1);
        if (s3) st(1, 0, t + 3);
      }
      PPP_END(PPP_W1(8, 24));
      // p7: G1 Q(1,1)t+1 | G0 reads af+bf of t+2 (within-tile only: the
      // cross-boundary read happens AFTER the epilogue so af/bf are not
      // live across it — keeping them live spilled 1.2 KB of scratch),
      // stages A1(t+3)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma32(1);
      } else {
        if (!last_it) {
          read_af(t + 2);
          read_bf(0, t + 2);
        }
        if (s3) st(1, 1, t + 3);
      }
      PPP_END(PPP_W0(12, 28); PPP_W1(8, 24));
    }
    epilogue(cur);
    cur = nxt;
    have_next = (tl + 2 * SLOTS) < (lo + cnt);
    nxt = have_next ? make_state(tl + 2 * SLOTS) : cur;
    if (tl + SLOTS < lo + cnt) {
      if (g == 0) {
        // next tile's p0 fragments (slots staged by the boundary overflow;
        // certified by the last steady end-p6 wait + the epilogue barriers)
        read_af(0);
        read_bf(0, 0);
      }
      // drain + barrier: the new tile's p0 stage overwrites these slots
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }
  }
#undef PPP_END
#undef PPP_W0
#undef PPP_W1
}

// ---------------------------------------------------------------------------
// pp128: BM=256 x BN=128 PERSISTENT ping-pong, template TRMODE (r2).
//
// The 256^2 ping-pong held 128 acc VGPRs per lane and sat exactly at the
// 255-register cap: both the persistent sweep AND the transpose-in-stage
// experiments spilled and fragmented the MFMA clusters. Halving BN halves
// the accumulator (64 VGPR) and leaves room for everything:
//   - persistent sweep state (short-K BERT panels — no per-tile ramp),
//   - native NN/TN staging (no lowering transposes),
//   - a 64 KiB spare for a fully countable LDS-staged epilogue
//     (8 b128 stores per wave; boundary waits widen by +8).
// Intensity drops to 87 FLOP/B (demand ~12 TB/s at 1.05 PF), so DEEP-K NT
// stays on the 256^2 single-tile kernel; this one serves short-K NT and
// all native-TR shapes (incl. split-K TN via the batch dim).
//
// Stage schedule: the deep pattern (iteration t stages all of t+2/t+3;
// the last iteration's stages are the next tile's prologue). Units are
// MIXED SIZE (A half 16 KiB = 4 glds/thread, B half 8 KiB = 2), so the
// per-group vmcnt certificates were re-derived and simulator-verified:
//   NT: end-p0 W0(6), p2 W1(6), p3 W0(8), p4 W0(6), p6 W1(6),
//       p7 W0(8)+W1(6); prologue vmcnt(6) both groups
//   NN (B transposed, sync): all four waits vmcnt(4); prologue vmcnt(4)
//   TN (both transposed): no steady waits at all
// fresh (first iteration after an epilogue): counts widen by the 8
// epilogue stores per wave (6->14, 8->16, 4->12); FIFO completion order
// keeps the wide form safe once the stores drain.
// ---------------------------------------------------------------------------
namespace pp128 {
constexpr int QBM = 256, QBN = 128, QBK = 64;
constexpr int AHALF_E = 128 * QBK;  // 8192 elems (16 KiB)
constexpr int BHALF_E = 64 * QBK;   // 4096 elems (8 KiB)
constexpr int QBUF_E = 2 * AHALF_E + 2 * BHALF_E;  // 24576 elems (48 KiB)
__device__ __forceinline__ uint16_t* qslot(uint16_t* smem, int buf, int isB,
                                           int half) {
  return smem + buf * QBUF_E +
         (isB ? 2 * AHALF_E + half * BHALF_E : half * AHALF_E);
}
// one wait site; mode-selected count, tails drain
template <int TRMODE, int GRP, int NT_N, int NN_N>
__device__ __forceinline__ void site(int g, bool tail, bool fresh) {
  if (tail) {
    ph8::vm_wait<0>();
    return;
  }
  if (g != GRP) return;
  if constexpr (TRMODE == 0) {
    if (fresh) ph8::vm_wait<NT_N + 8>();
    else ph8::vm_wait<NT_N>();
  } else if constexpr (TRMODE == 1) {
    if constexpr (NN_N >= 0) {
      if (fresh) ph8::vm_wait<NN_N + 8>();
      else ph8::vm_wait<NN_N>();
    }
  }
}
}  // namespace pp128

template <int TRMODE>
__launch_bounds__(512, 2) __global__ void gemm_bf16_pp128_pers_k(
    GemmArgs args, int nwg_n, int64_t per_b, int64_t total) {
  using namespace pp128;
  using ph8::vm_wait;
  // [2 pipeline buffers 96 KiB][spare 64 KiB] = the full 160 KiB
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * QBUF_E + 32768];
  uint16_t* spare = smem + 2 * QBUF_E;

  const int orig = blockIdx.x;  // 256 WGs
  const int xcd = orig & 7, slotq = orig >> 3;
  const int64_t q = total / 8, r8 = total % 8;
  const int64_t lo = xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q;
  const int64_t cnt = xcd < r8 ? q + 1 : q;
  if (slotq >= cnt) return;
  constexpr int SLOTS = 32;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = wid >> 2;   // ping-pong group: G0 rows [0,128), G1 [128,256)
  const int ws = wid & 3;   // 2x2 wave grid over a 128x64 quadrant
  const int tig = (int)threadIdx.x & 255;
  if (g == 1) __builtin_amdgcn_s_setprio(1);

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / QBK;  // even, >= 2 (launch guard)

  struct TileSt {
    int32_t bidx, m0, n0;
  };
  auto make_state = [&](int64_t tl) {
    TileSt ts;
    ts.bidx = __builtin_amdgcn_readfirstlane((int32_t)(tl / per_b));
    const int64_t rtl = tl % per_b;
    ts.m0 = __builtin_amdgcn_readfirstlane((int32_t)((rtl / nwg_n) * QBM));
    ts.n0 = __builtin_amdgcn_readfirstlane((int32_t)((rtl % nwg_n) * QBN));
    return ts;
  };

  // k-major transposed stage into the bank-rotated XOR image (see the
  // 256^2 kernel's stage_tr comment); NELEM = half-tile elems, RW = r width
  auto stage_tr = [&](const uint16_t* X, int64_t ldx, int64_t r0,
                      int64_t Rlast8, uint16_t* dst, int64_t k0, int rw) {
    const int chunks = rw == 128 ? 4 : 2;
    for (int c = 0; c < chunks; ++c) {
      int e = tig * 8 + c * 2048;
      int kk = rw == 128 ? (e >> 7) : (e >> 6);
      int rl = e & (rw - 1);
      int64_t gr = r0 + rl;
      if (gr > Rlast8) gr = Rlast8;
      short8 vv = *(const short8*)(X + (k0 + kk) * ldx + gr);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        int r = rl + i;
        int col = ((kk ^ ((r & 7) << 3)) + ((r >> 3) << 3)) & 63;
        dst[r * QBK + col] = (uint16_t)vv[i];
      }
    }
  };
  auto stage = [&](const TileSt& ts, int isB, int h, int64_t kt) {
    int64_t k0 = kt * QBK;
    uint16_t* dst = qslot(smem, (int)(kt & 1), isB, h);
    if (!isB) {
      if constexpr ((TRMODE & 2) != 0) {
        stage_tr(args.A + (int64_t)ts.bidx * args.sA, lda, ts.m0 + h * 128,
                 args.M - 8, dst, k0, 128);
        return;
      }
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        int e = tig * 8 + c * 2048;
        int row = e >> 6;
        int k = (e & 63) ^ ((row & 7) << 3);
        int64_t ar = ts.m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(args.A + (int64_t)ts.bidx * args.sA + ar * lda + k0 + k,
               dst + e);
      }
    } else {
      if constexpr ((TRMODE & 1) != 0) {
        stage_tr(args.B + (int64_t)ts.bidx * args.sB, ldb, ts.n0 + h * 64,
                 args.N - 8, dst, k0, 64);
        return;
      }
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        int e = tig * 8 + c * 2048;
        int row = e >> 6;
        int k = (e & 63) ^ ((row & 7) << 3);
        int64_t br = ts.n0 + h * 64 + row;
        if (br > Nc) br = Nc;
        glds16(args.B + (int64_t)ts.bidx * args.sB + br * ldb + k0 + k,
               dst + e);
      }
    }
  };

  f32x4 acc[2][4][2] = {};    // [B-half][fm][fn]
  short8 af[2][4], bf[2][2];  // fragment set: [kk][frag]

  auto read_af = [&](int64_t kt) {
    const uint16_t* as = qslot(smem, (int)(kt & 1), 0, g);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = (ws >> 1) * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        if constexpr ((TRMODE & 2) != 0)
          ks = (ks + ((row >> 3) << 3)) & 63;
        af[kk2][fm] = *(const short8*)(as + row * QBK + ks);
      }
    }
  };
  auto read_bf = [&](int b, int64_t kt) {
    const uint16_t* bs = qslot(smem, (int)(kt & 1), 1, b);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (ws & 1) * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        if constexpr ((TRMODE & 1) != 0)
          ks = (ks + ((col >> 3) << 3)) & 63;
        bf[kk2][fn] = *(const short8*)(bs + col * QBK + ks);
      }
    }
  };

  auto mfma16 = [&](int b) {
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[b][fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk2][fm], bf[kk2][fn], acc[b][fm][fn], 0, 0, 0);
  };

  // countable epilogue: whole 256x128 C tile staged through the 64 KiB
  // spare, streamed as 8 b128 stores per thread. Raw barriers + lgkm-only
  // waits (never drain the in-flight next-tile pipeline).
  auto epilogue = [&](const TileSt& ts) {
    uint16_t* C = args.C + (int64_t)ts.bidx * args.sC;
    float* out32 =
        args.out_f32 ? args.out_f32 + (int64_t)ts.bidx * args.sC : nullptr;
    uint16_t* save_pre =
        args.save_pre ? args.save_pre + (int64_t)ts.bidx * args.sC : nullptr;
    const int64_t N = args.N;
    const bool interior =
        !out32 && (ts.m0 + QBM <= args.M) && (ts.n0 + QBN <= args.N);
    if (interior) {
#pragma unroll
      for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
#pragma unroll
            for (int rI = 0; rI < 4; ++rI) {
              int lrow = g * 128 + (ws >> 1) * 64 + fm * 16 +
                         (lane >> 4) * 4 + rI;
              int lcol = b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
              spare[lrow * 128 + lcol] = f2b(args.alpha * acc[b2][fm][fn][rI]);
            }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int e = (c * 512 + (int)threadIdx.x) * 8;
        int row = e >> 7, col = e & 127;
        short8 vv = *(const short8*)(spare + e);
        int64_t gcol = ts.n0 + col;
        int64_t idx = (ts.m0 + row) * N + gcol;
        short8 zout, yout;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float v = b2f((uint16_t)vv[i]);
          if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
          if (args.bias) v += b2f(args.bias[gcol + i]);
          zout[i] = (short)f2b(v);
          if (args.fuse_gelu) v = gelu_fwd_scalar(v);
          yout[i] = (short)f2b(v);
        }
        if (save_pre) *(short8*)(save_pre + idx) = zout;
        *(short8*)(C + idx) = yout;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    } else {
#pragma unroll
      for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
#pragma unroll
            for (int rI = 0; rI < 4; ++rI) {
              int64_t grow = ts.m0 + g * 128 + (ws >> 1) * 64 + fm * 16 +
                             (lane >> 4) * 4 + rI;
              int64_t gcol =
                  ts.n0 + b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
              if (grow < args.M && gcol < args.N) {
                int64_t idx = grow * N + gcol;
                if (out32) {
                  out32[idx] = acc[b2][fm][fn][rI];
                } else {
                  float v = args.alpha * acc[b2][fm][fn][rI];
                  if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                  if (args.bias) v += b2f(args.bias[gcol]);
                  if (save_pre) save_pre[idx] = f2b(v);
                  if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                  C[idx] = f2b(v);
                }
              }
            }
      // uncountable scalar stores: drain before the next tile's waits
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }
#pragma unroll
    for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn) acc[b2][fm][fn] = f32x4{};
  };

#define Q_END(wait_stmt)                                   \
  do {                                                     \
    wait_stmt;                                             \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)

  int64_t tl = lo + slotq;
  TileSt cur = make_state(tl);
  bool have_next = (tl + SLOTS) < (lo + cnt);
  TileSt nxt = have_next ? make_state(tl + SLOTS) : cur;

  // prologue: tiles 0 and 1 fully staged, 4 units per group (FIFO order)
  if (g == 1) {
    stage(cur, 0, 0, 0);  // A0(0)
    stage(cur, 1, 0, 0);  // B0(0)
    stage(cur, 0, 0, 1);  // A0(1)
    stage(cur, 1, 0, 1);  // B0(1)
  } else {
    stage(cur, 0, 1, 0);  // A1(0)
    stage(cur, 1, 1, 0);  // B1(0)
    stage(cur, 0, 1, 1);  // A1(1)
    stage(cur, 1, 1, 1);  // B1(1)
  }
  if constexpr (TRMODE == 0) vm_wait<6>();
  else if constexpr (TRMODE == 1) vm_wait<4>();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (g == 0) {
    read_af(0);
    read_bf(0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  for (; tl < lo + cnt; tl += SLOTS) {
    const bool not_first_tile = (tl != lo + slotq);
    for (int64_t t = 0; t < ntiles; t += 2) {
      const bool last_it = (t + 2 >= ntiles);
      const bool fresh = (t == 0) && not_first_tile;
      const bool tail = last_it && !have_next;
      const bool s2 = !last_it || have_next;
      const bool s3 = (t + 3 < ntiles) || have_next;
      auto st = [&](int isB, int h, int64_t kt) {
        const bool ov = kt >= ntiles;
        TileSt ts = ov ? nxt : cur;
        stage(ts, isB, h, ov ? kt - ntiles : kt);
      };
      // p0: G0 Q(0,0)t | G1 reads af(A1,t)+bf(B0,t), stages A0(t+2)
      if (g == 0) {
        mfma16(0);
      } else {
        read_af(t);
        read_bf(0, t);
        if (s2) st(0, 0, t + 2);
      }
      Q_END((site<TRMODE, 0, 6, -1>(g, tail, fresh)));
      // p1: G1 Q(1,0)t | G0 reads bf(B1,t), stages A1(t+2)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(0);
      } else {
        read_bf(1, t);
        if (s2) st(0, 1, t + 2);
      }
      Q_END();
      // p2: G0 Q(0,1)t | G1 reads bf(B1,t), stages B0(t+2)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(1);
      } else {
        read_bf(1, t);
        if (s2) st(1, 0, t + 2);
      }
      Q_END((site<TRMODE, 1, 6, 4>(g, tail, fresh)));
      // p3: G1 Q(1,1)t | G0 reads af(A0,t+1)+bf(B0,t+1), stages B1(t+2)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(1);
      } else {
        read_af(t + 1);
        read_bf(0, t + 1);
        if (s2) st(1, 1, t + 2);
      }
      Q_END((site<TRMODE, 0, 8, 4>(g, tail, fresh)));
      // p4: G0 Q(0,0)t+1 | G1 reads af(A1,t+1)+bf(B0,t+1), stages A0(t+3)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(0);
      } else {
        read_af(t + 1);
        read_bf(0, t + 1);
        if (s3) st(0, 0, t + 3);
      }
      Q_END((site<TRMODE, 0, 6, -1>(g, tail, fresh)));
      // p5: G1 Q(1,0)t+1 | G0 reads bf(B1,t+1), stages A1(t+3)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(0);
      } else {
        read_bf(1, t + 1);
        if (s3) st(0, 1, t + 3);
      }
      Q_END();
      // p6: G0 Q(0,1)t+1 | G1 reads bf(B1,t+1), stages B0(t+3)
      if (g == 0) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(1);
      } else {
        read_bf(1, t + 1);
        if (s3) st(1, 0, t + 3);
      }
      Q_END((site<TRMODE, 1, 6, 4>(g, tail, fresh)));
      // p7: G1 Q(1,1)t+1 | G0 reads af+bf of t+2 (within-tile only; the
      // cross-boundary read happens after the epilogue), stages B1(t+3)
      if (g == 1) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        mfma16(1);
      } else {
        if (!last_it) {
          read_af(t + 2);
          read_bf(0, t + 2);
        }
        if (s3) st(1, 1, t + 3);
      }
      Q_END((site<TRMODE, 0, 8, 4>(g, tail, fresh),
             site<TRMODE, 1, 6, -1>(g, tail, fresh)));
    }
    epilogue(cur);
    cur = nxt;
    have_next = (tl + 2 * SLOTS) < (lo + cnt);
    nxt = have_next ? make_state(tl + 2 * SLOTS) : cur;
    if (tl + SLOTS < lo + cnt) {
      if (g == 0) {
        read_af(0);
        read_bf(0, 0);
      }
      // the pre-read must complete before the new tile's p0 stage
      // overwrites the A0/B0 slots it is reading (race found by the
      // screen: flaky NONDET on 32768x2304x768)
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      asm volatile("" ::: "memory");
    }
  }
#undef Q_END
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

// ---------------------------------------------------------------------------
// pp64q: 128x128 ping-pong at 2 WORKGROUPS PER CU — the short-K kernel.
//
// Short-K diagnosis (profiles/r2): every 1-WG/CU structure plateaus near
// 540 TF on the BERT K=768 panels while hipBLASLt reaches ~1014-1043 —
// with one workgroup the per-tile prologue fill (~2 us of DMA against
// ~9 us of tile work at K=768) and every wait bubble are exposed. At
// 64 KiB LDS and a <=128-VGPR budget (launch_bounds(512,4)) TWO
// independent 8-wave ping-pong workgroups co-reside per CU: 4 waves per
// SIMD, and the de-synchronized pair fills each other's prologue/wait/
// epilogue bubbles — the cross-WG hiding the 128^2 persistent kernel
// relied on, combined with the ping-pong's pure 8-MFMA clusters.
// acc[2][2][2] (32 VGPR) + 4+4 fragment b128s (32) leave margin.
// Deep schedule, uniform 2-glds units; waits simulator-verified:
//   p0 W0(4), p2 W1(4), p3 W0(6), p4 W0(4), p6 W1(4), p7 W0(6)+W1(4);
// prologue vmcnt(4) both groups; last iteration drains.
// ---------------------------------------------------------------------------
namespace pp64q {
constexpr int QM = 128, QN = 128, QK = 64;
constexpr int HALF_E = 64 * QK;  // 4096 elems (8 KiB)
constexpr int BUF_E = 4 * HALF_E;
__device__ __forceinline__ uint16_t* qslot(uint16_t* smem, int buf, int isB,
                                           int half) {
  return smem + buf * BUF_E + (isB * 2 + half) * HALF_E;
}
template <int GRP, int NT_N>
__device__ __forceinline__ void site(int g, bool tail) {
  if (tail) {
    ph8::vm_wait<0>();
    return;
  }
  if (g == GRP) ph8::vm_wait<NT_N>();
}
}  // namespace pp64q

__launch_bounds__(512, 4) __global__ void gemm_bf16_pp64q_k(GemmArgs args,
                                                            int nwg_n) {
  using namespace pp64q;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int64_t tm, tn;
  if (args.group_m < 0) {
    // XCD-slab raster (short-K): consecutive blockIdx dispatch round-robin
    // over the 8 XCDs, so XCD x = orig%8 owns M-slabs {x, x+8, ...} and
    // sweeps ALL N-tiles of a slab before the next (N-first order). The
    // ~64 live WGs of an XCD then share ~4 A-slabs (~0.8 MB) in that
    // XCD's L2 — the cold A stream becomes L2-hit and the glds stage
    // rate is no longer HBM-latency-starved (requires nwg_m % 8 == 0;
    // dispatch guards it).
    int i = orig >> 3, x = orig & 7;
    tm = (int64_t)(i / nwg_n) * 8 + x;
    tn = i % nwg_n;
  } else {
    int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
    int wgid =
        (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
    tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  }
  const int64_t m0 = tm * QM;
  const int64_t n0 = tn * QN;
  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  uint16_t* save_pre = args.save_pre;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = wid >> 2;   // group: A-half (rows g*64..+64)
  const int ws = wid & 3;   // 2x2 wave grid over a 64x64 quadrant
  const int tig = (int)threadIdx.x & 255;
  if (g == 1) __builtin_amdgcn_s_setprio(1);

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / QK;  // even, >= 4 (launch guard)

  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * QK;
    uint16_t* dst = qslot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = tig * 8 + c * 2048;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 64 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 64 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][2][2] = {};    // [B-half][fm][fn]
  short8 af[2][2], bf[2][2];  // [kk][frag]

  auto read_af = [&](int64_t t) {
    const uint16_t* as = qslot(smem, (int)(t & 1), 0, g);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = (ws >> 1) * 32 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        af[kk2][fm] = *(const short8*)(as + row * QK + ks);
      }
    }
  };
  auto read_bf = [&](int b, int64_t t) {
    const uint16_t* bs = qslot(smem, (int)(t & 1), 1, b);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (ws & 1) * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        bf[kk2][fn] = *(const short8*)(bs + col * QK + ks);
      }
    }
  };

  auto mfma8 = [&](int b) {
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[b][fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[kk2][fm], bf[kk2][fn], acc[b][fm][fn], 0, 0, 0);
  };

  // prologue: tiles 0 and 1, 4 units per group in FIFO order
  if (g == 1) {
    stage(0, 0, 0);  // A0(0)
    stage(1, 0, 0);  // B0(0)
    stage(0, 0, 1);  // A0(1)
    stage(1, 0, 1);  // B0(1)
  } else {
    stage(0, 1, 0);  // A1(0)
    stage(1, 1, 0);  // B1(0)
    stage(0, 1, 1);  // A1(1)
    stage(1, 1, 1);  // B1(1)
  }
  ph8::vm_wait<4>();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (g == 0) {
    read_af(0);
    read_bf(0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

#define Q4_END(wait_stmt)                                  \
  do {                                                     \
    wait_stmt;                                             \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)

  // diag ablation (NEZHA_PERS_DIAG, same contract as the v7 kernel;
  // OUTPUT GARBAGE): 2 = stage-only, 3 = compute-only
  const bool dg_mfma = args.diag_no_store != 2;
  const bool dg_stage = args.diag_no_store != 3;
  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 2 >= ntiles) || !dg_stage;
    const bool s2 = (t + 2 < ntiles) && dg_stage, s3 = (t + 3 < ntiles) && dg_stage;
    // p0
    if (g == 0) {
      if (dg_mfma) mfma8(0);
    } else {
      read_af(t);
      read_bf(0, t);
      if (s2) stage(0, 0, t + 2);
    }
    Q4_END((site<0, 4>(g, tail)));
    // p1
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(0);
    } else {
      read_bf(1, t);
      if (s2) stage(0, 1, t + 2);
    }
    Q4_END();
    // p2
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(1);
    } else {
      read_bf(1, t);
      if (s2) stage(1, 0, t + 2);
    }
    Q4_END((site<1, 4>(g, tail)));
    // p3
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(1);
    } else {
      read_af(t + 1);
      read_bf(0, t + 1);
      if (s2) stage(1, 1, t + 2);
    }
    Q4_END((site<0, 6>(g, tail)));
    // p4
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(0);
    } else {
      read_af(t + 1);
      read_bf(0, t + 1);
      if (s3) stage(0, 0, t + 3);
    }
    Q4_END((site<0, 4>(g, tail)));
    // p5
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(0);
    } else {
      read_bf(1, t + 1);
      if (s3) stage(0, 1, t + 3);
    }
    Q4_END();
    // p6
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(1);
    } else {
      read_bf(1, t + 1);
      if (s3) stage(1, 0, t + 3);
    }
    Q4_END((site<1, 4>(g, tail)));
    // p7
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma8(1);
    } else {
      if (s2) {
        read_af(t + 2);
        read_bf(0, t + 2);
      }
      if (s3) stage(1, 1, t + 3);
    }
    Q4_END((site<0, 6>(g, tail), site<1, 4>(g, tail)));
  }
#undef Q4_END

  // drain, then the LDS-staged epilogue reuses smem (32 KiB C tile)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  if (out32) {
#pragma unroll
    for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int64_t grow = m0 + g * 64 + (ws >> 1) * 32 + fm * 16 +
                           (lane >> 4) * 4 + rI;
            int64_t gcol =
                n0 + b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
            if (grow < args.M && gcol < args.N)
              out32[grow * args.N + gcol] = acc[b2][fm][fn][rI];
          }
    return;
  }
#pragma unroll
  for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int lrow = g * 64 + (ws >> 1) * 32 + fm * 16 + (lane >> 4) * 4 + rI;
          int lcol = b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
          smem[lrow * 128 + lcol] = f2b(args.alpha * acc[b2][fm][fn][rI]);
        }
  __syncthreads();
  const bool interior = (m0 + QM <= args.M) && (n0 + QN <= args.N);
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 7, col = e & 127;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// 4-PHASE variant of the 2-WG/CU 128^2 ping-pong (NEZHA_GEMM_64Q=4): same
// tile geometry and epilogue as gemm_bf16_pp64q_k, but 4 fat phases per 2
// k-iterations instead of 8 thin ones — each compute phase is a 16-MFMA
// cluster (both B-halves) and the barrier/lgkm machinery per FLOP halves.
// The pp64q diag ablation showed compute-only pace 4x above the MFMA issue
// floor (0.370 vs 0.092 ms on the BERT qkv panel) with the thin phases;
// this variant attacks exactly that.
//
// Staging plan (verified by tools/pp64q4_sim.py — run it after ANY edit
// to the schedule below):
//   - A0/B0 (read by G0/both, staged by G1): DOUBLE buffered, slot t%2.
//     G1 stages A0B0(t+2) at P0 and A0B0(t+3) at P2.
//   - A1/B1 (read by G1/both, staged by G0): TRIPLE buffered, slot t%3 —
//     G0 stages A1B1(t+3) at P1 and A1B1(t+4) at P3; the third buffer is
//     what lets it run a full iteration deeper without overwriting a slot
//     another phase still reads. 2x2+3x2 half-units = 80 KiB = the whole
//     per-WG LDS budget at 2 WG/CU.
//   - Certificates: BOTH groups wait vmcnt<=4 (keep 2 stage-calls in
//     flight) at end-P0 and end-P2; end-P1/P3 are barrier-only. Tail
//     iteration drains vmcnt(0). Every read-then-stage phase drains lgkm
//     between its ds_reads and its glds issues (same-parity slot reuse).
// ---------------------------------------------------------------------------
__launch_bounds__(512, 4) __global__ void gemm_bf16_pp64q4_k(GemmArgs args,
                                                             int nwg_n) {
  using namespace pp64q;
  __shared__ __attribute__((aligned(16))) uint16_t smem[10 * HALF_E];

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * QM;
  const int64_t n0 = tn * QN;
  const int64_t batch = blockIdx.y;
  const uint16_t* __restrict__ A = args.A + batch * args.sA;
  const uint16_t* __restrict__ B = args.B + batch * args.sB;
  uint16_t* C = args.C + batch * args.sC;
  float* out32 = args.out_f32 ? args.out_f32 + batch * args.sC : nullptr;
  uint16_t* save_pre = args.save_pre;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int g = wid >> 2;   // group: A-half (rows g*64..+64)
  const int ws = wid & 3;   // 2x2 wave grid over a 64x64 quadrant
  const int tig = (int)threadIdx.x & 255;
  if (g == 1) __builtin_amdgcn_s_setprio(1);

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / QK;  // even, >= 4 (launch guard)

  // slot index: A0 -> 0 + t%2, B0 -> 2 + t%2, A1 -> 4 + t%3, B1 -> 7 + t%3
  auto slot4 = [&](int isB, int h, int64_t t) -> uint16_t* {
    int s = h ? (4 + 3 * isB + (int)(t % 3)) : (2 * isB + (int)(t & 1));
    return smem + s * HALF_E;
  };
  auto stage4 = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * QK;
    uint16_t* dst = slot4(isB, h, t);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = tig * 8 + c * 2048;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 64 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 64 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][2][2] = {};       // [B-half][fm][fn]
  short8 af[2][2], bf[2][2][2];  // af[kk2][fm]; bf[b][kk2][fn]

  auto read_af4 = [&](int64_t t) {
    const uint16_t* as = slot4(0, g, t);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = (ws >> 1) * 32 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        af[kk2][fm] = *(const short8*)(as + row * QK + ks);
      }
    }
  };
  auto read_bf4 = [&](int b, int64_t t) {
    const uint16_t* bs = slot4(1, b, t);
#pragma unroll
    for (int kk2 = 0; kk2 < 2; ++kk2) {
      const int kbase = kk2 * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (ws & 1) * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        bf[b][kk2][fn] = *(const short8*)(bs + col * QK + ks);
      }
    }
  };

  auto mfma16 = [&]() {
#pragma unroll
    for (int b = 0; b < 2; ++b)
#pragma unroll
      for (int kk2 = 0; kk2 < 2; ++kk2)
#pragma unroll
        for (int fm = 0; fm < 2; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn)
            acc[b][fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[kk2][fm], bf[b][kk2][fn], acc[b][fm][fn], 0, 0, 0);
  };

  // prologue: G1 stages A0B0 of tiles 0,1; G0 stages A1B1 of 0,1,2
  // (the triple buffer starts one tile deeper); full drain, then G0
  // pre-reads its tile-0 fragments.
  if (g == 1) {
    stage4(0, 0, 0); stage4(1, 0, 0);
    stage4(0, 0, 1); stage4(1, 0, 1);
  } else {
    stage4(0, 1, 0); stage4(1, 1, 0);
    stage4(0, 1, 1); stage4(1, 1, 1);
    stage4(0, 1, 2); stage4(1, 1, 2);
  }
  ph8::vm_wait<0>();
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (g == 0) {
    read_af4(0);
    read_bf4(0, 0);
    read_bf4(1, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

#define Q4P_END(vmstmt)                                    \
  do {                                                     \
    vmstmt;                                                \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");     \
    __builtin_amdgcn_s_barrier();                          \
    asm volatile("" ::: "memory");                         \
  } while (0)

  const bool dg_mfma = args.diag_no_store != 2;
  const bool dg_stage = args.diag_no_store != 3;
  const bool dg_drain = args.diag_no_store == 9;  // debug: vm(0) every phase
  // debug bisect: 8 = add vm(4) at the barrier-only ends (P1/P3);
  //               7 = tighten P0/P2 to vm(0), keep P1/P3 barrier-only
  const bool dg_w13 = args.diag_no_store == 8;
  const bool dg_w02 = args.diag_no_store == 7;
  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 2 >= ntiles) || !dg_stage || dg_drain;
    const bool s2 = (t + 2 < ntiles) && dg_stage;
    const bool s3 = (t + 3 < ntiles) && dg_stage;
    const bool s4 = (t + 4 < ntiles) && dg_stage;
    // P0: G0 computes t | G1 reads t-frags, stages A0B0(t+2)
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma16();
    } else {
      read_af4(t);
      read_bf4(0, t);
      read_bf4(1, t);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (s2) { stage4(0, 0, t + 2); stage4(1, 0, t + 2); }
    }
    Q4P_END(((tail || dg_w02) ? ph8::vm_wait<0>() : ph8::vm_wait<4>()));
    // P1: G1 computes t | G0 reads (t+1)-frags, stages A1B1(t+3)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma16();
    } else {
      read_af4(t + 1);
      read_bf4(0, t + 1);
      read_bf4(1, t + 1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (s3) { stage4(0, 1, t + 3); stage4(1, 1, t + 3); }
    }
    Q4P_END((tail ? ph8::vm_wait<0>() : (dg_w13 ? ph8::vm_wait<4>() : (void)0)));
    // P2: G0 computes t+1 | G1 reads (t+1)-frags, stages A0B0(t+3)
    if (g == 0) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma16();
    } else {
      read_af4(t + 1);
      read_bf4(0, t + 1);
      read_bf4(1, t + 1);
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (s3) { stage4(0, 0, t + 3); stage4(1, 0, t + 3); }
    }
    Q4P_END(((tail || dg_w02) ? ph8::vm_wait<0>() : ph8::vm_wait<4>()));
    // P3: G1 computes t+1 | G0 reads (t+2)-frags, stages A1B1(t+4)
    if (g == 1) {
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (dg_mfma) mfma16();
    } else {
      if (s2) {
        read_af4(t + 2);
        read_bf4(0, t + 2);
        read_bf4(1, t + 2);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (s4) { stage4(0, 1, t + 4); stage4(1, 1, t + 4); }
    }
    Q4P_END((tail ? ph8::vm_wait<0>() : (dg_w13 ? ph8::vm_wait<4>() : (void)0)));
  }
#undef Q4P_END

  // drain, then the LDS-staged epilogue reuses smem (32 KiB C tile)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  if (out32) {
#pragma unroll
    for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int64_t grow = m0 + g * 64 + (ws >> 1) * 32 + fm * 16 +
                           (lane >> 4) * 4 + rI;
            int64_t gcol =
                n0 + b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
            if (grow < args.M && gcol < args.N)
              out32[grow * args.N + gcol] = acc[b2][fm][fn][rI];
          }
    return;
  }
#pragma unroll
  for (int b2 = 0; b2 < 2; ++b2)
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int lrow = g * 64 + (ws >> 1) * 32 + fm * 16 + (lane >> 4) * 4 + rI;
          int lcol = b2 * 64 + (ws & 1) * 32 + fn * 16 + (lane & 15);
          smem[lrow * 128 + lcol] = f2b(args.alpha * acc[b2][fm][fn][rI]);
        }
  __syncthreads();
  const bool interior = (m0 + QM <= args.M) && (n0 + QN <= args.N);
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 7, col = e & 127;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// PERSISTENT 8-phase kernel: 256 WGs (1/CU), each sweeping a strided run of
// (batch, m, n) 256^2 tiles with the SAME 8-phase wait schedule as
// gemm_bf16_nt_8phase_k — and the pipeline flowing ACROSS tile boundaries.
// The key observation: the s2/s3-guarded stages of a tile's LAST iteration
// (B0/A1/A0/B1 of t+2 at p2..p5, B0/A1 of t+3 at p6/p7) are, in both ORDER
// and LDS SLOT, exactly the 6-unit prologue of the next tile (B0(0) A1(0)
// A0(0) B1(0) B0(1) A1(1), k0 -> buf0, k1 -> buf1 since ntiles is even).
// Substituting next-tile stages there keeps every per-phase glds count, and
// therefore the whole counted-vmcnt schedule, unchanged. Only two waits
// differ from the single-tile kernel:
//   - phase 0 of each tile waits vmcnt(0): it absorbs the previous tile's
//     epilogue stores (which share the vmcnt FIFO) and re-establishes the
//     counting invariant from a clean slate;
//   - the true tail (last tile's last iteration, nothing left to stage)
//     keeps the original vmcnt(0)-everywhere behavior.
// This removes the 8-phase ramp at short K (K=768 = 3 iterations was ramp-
// dominated) AND the 256-WG round-balance restriction — one prologue and
// one drain per WG instead of per tile. Epilogue is a direct scatter
// (interior fast path); the LDS-staged epilogue of the single-tile kernel
// would clobber the buffers holding the next tile's prefetched halves.
// ---------------------------------------------------------------------------
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase_pers_k(
    GemmArgs args, int nwg_n, int64_t per_b, int64_t total) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const int orig = blockIdx.x;  // 256 WGs
  const int xcd = orig & 7, slotq = orig >> 3;  // 32 slots per XCD
  const int64_t q = total / 8, r8 = total % 8;
  const int64_t lo = xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q;
  const int64_t cnt = xcd < r8 ? q + 1 : q;
  if (slotq >= cnt) return;
  constexpr int S = 32;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;  // even, >= 2 (launch guard)
  const int ea = (int)threadIdx.x * 8;

  struct TileSt {
    const uint16_t* A;  // batch-offset bases
    const uint16_t* B;
    int32_t m0, n0;     // 32-bit: tile coords fit easily
  };
  auto make_state = [&](int64_t tl) {
    TileSt ts;
    const int64_t b = tl / per_b;
    const int64_t rtl = tl % per_b;
    ts.A = args.A + b * args.sA;
    ts.B = args.B + b * args.sB;
    ts.m0 = (int32_t)((rtl / nwg_n) * PBM);
    ts.n0 = (int32_t)((rtl % nwg_n) * PBN);
    return ts;
  };

  auto stage = [&](const TileSt& ts, int isB, int h, int buf, int64_t kt) {
    int64_t k0 = kt * PBK;
    uint16_t* dst = slot(smem, buf, isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = ts.m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(ts.A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = ts.n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(ts.B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][2][4][2] = {};

  auto compute = [&](int HA, int HB, int buf, auto& accq) {
    const uint16_t* as = slot(smem, buf, 0, HA);
    const uint16_t* bs = slot(smem, buf, 1, HB);
#pragma unroll
    for (int kk = 0; kk < PBK; kk += 32) {
      short8 a_frag[4], b_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(as + row * PBK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(bs + col * PBK + ks);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          accq[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], accq[fm][fn], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  // scatter epilogue (interior fast path); stores stay in flight — the next
  // tile's phase-0 vmcnt(0) drains them
  auto epilogue = [&](int64_t tl) {
    const int64_t b = tl / per_b;
    const int64_t rtl = tl % per_b;
    uint16_t* C = args.C + b * args.sC;
    float* out32 = args.out_f32 ? args.out_f32 + b * args.sC : nullptr;
    uint16_t* save_pre = args.save_pre ? args.save_pre + b * args.sC : nullptr;
    const int64_t N = args.N;
    const int64_t m0 = (rtl / nwg_n) * PBM, n0 = (rtl % nwg_n) * PBN;
    if (m0 + PBM <= args.M && n0 + PBN <= args.N) {
      const int64_t base =
          (m0 + wr * 64 + (lane >> 4) * 4) * N + n0 + wc * 32 + (lane & 15);
      const uint16_t* bias_l =
          args.bias ? args.bias + n0 + wc * 32 + (lane & 15) : nullptr;
#pragma unroll
      for (int ha = 0; ha < 2; ++ha)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            const int64_t rowbase =
                base + (int64_t)(ha * 128 + fm * 16 + rI) * N;
#pragma unroll
            for (int hb = 0; hb < 2; ++hb)
#pragma unroll
              for (int fn = 0; fn < 2; ++fn) {
                const int64_t idx = rowbase + hb * 128 + fn * 16;
                if (out32) {
                  out32[idx] = acc[ha][hb][fm][fn][rI];
                } else {
                  float v = args.alpha * acc[ha][hb][fm][fn][rI];
                  if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                  if (bias_l) v += b2f(bias_l[hb * 128 + fn * 16]);
                  if (save_pre) save_pre[idx] = f2b(v);
                  if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                  C[idx] = f2b(v);
                }
              }
          }
    } else {
#pragma unroll
      for (int ha = 0; ha < 2; ++ha)
#pragma unroll
        for (int hb = 0; hb < 2; ++hb)
#pragma unroll
          for (int fm = 0; fm < 4; ++fm)
#pragma unroll
            for (int fn = 0; fn < 2; ++fn)
#pragma unroll
              for (int rI = 0; rI < 4; ++rI) {
                int64_t grow =
                    m0 + ha * 128 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
                int64_t gcol =
                    n0 + hb * 128 + wc * 32 + fn * 16 + (lane & 15);
                if (grow < args.M && gcol < args.N) {
                  int64_t idx = grow * N + gcol;
                  if (out32) {
                    out32[idx] = acc[ha][hb][fm][fn][rI];
                  } else {
                    float v = args.alpha * acc[ha][hb][fm][fn][rI];
                    if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
                    if (args.bias) v += b2f(args.bias[gcol]);
                    if (save_pre) save_pre[idx] = f2b(v);
                    if (args.fuse_gelu) v = gelu_fwd_scalar(v);
                    C[idx] = f2b(v);
                  }
                }
              }
    }
#pragma unroll
    for (int ha = 0; ha < 2; ++ha)
#pragma unroll
      for (int hb = 0; hb < 2; ++hb)
#pragma unroll
        for (int fm = 0; fm < 4; ++fm)
#pragma unroll
          for (int fn = 0; fn < 2; ++fn) acc[ha][hb][fm][fn] = f32x4{};
  };

#define P8P_WAIT0()                                     \
  do {                                                  \
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");    \
    __builtin_amdgcn_s_barrier();                       \
    asm volatile("" ::: "memory");                      \
  } while (0)
#define P8P_WAITN(n)                                    \
  do {                                                  \
    if (tail)                                           \
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  \
    else                                                \
      asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory"); \
    __builtin_amdgcn_s_barrier();                       \
    asm volatile("" ::: "memory");                      \
  } while (0)
#define P8P_BAR()                                       \
  do {                                                  \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  \
    __builtin_amdgcn_s_barrier();                       \
    asm volatile("" ::: "memory");                      \
  } while (0)

  int64_t tl = lo + slotq;
  TileSt cur = make_state(tl);
  // prologue of the first tile (order the wait schedule assumes)
  stage(cur, 1, 0, 0, 0);  // B0(0)
  stage(cur, 0, 1, 0, 0);  // A1(0)
  stage(cur, 0, 0, 0, 0);  // A0(0)
  stage(cur, 1, 1, 0, 0);  // B1(0)
  stage(cur, 1, 0, 1, 1);  // B0(1)
  stage(cur, 0, 1, 1, 1);  // A1(1)

  for (; tl < lo + cnt; tl += S) {
    const bool has_next = (tl + S) < (lo + cnt);
    TileSt nxt = has_next ? make_state(tl + S) : cur;
    for (int64_t t = 0; t < ntiles; t += 2) {
      const bool last = (t + 2 >= ntiles);  // note: !last implies t+3 < ntiles
      const bool tail = last && !has_next;
      // phase 0: Q(0,0) of t (buf0); stage A0(t+1) -> buf1
      if (t == 0) {
        P8P_WAIT0();
      } else {
        P8P_WAITN(6);
      }
      stage(cur, 0, 0, 1, t + 1);
      compute(0, 0, 0, acc[0][0]);
      P8P_BAR();
      // phase 1: Q(1,0) of t; stage B1(t+1) -> buf1
      stage(cur, 1, 1, 1, t + 1);
      compute(1, 0, 0, acc[1][0]);
      P8P_BAR();
      // phase 2: Q(1,1) of t; stage B0(t+2) / next-tile B0(0) -> buf0
      P8P_WAITN(8);
      if (!last)
        stage(cur, 1, 0, 0, t + 2);
      else if (has_next)
        stage(nxt, 1, 0, 0, 0);
      compute(1, 1, 0, acc[1][1]);
      P8P_BAR();
      // phase 3: Q(0,1) of t; stage A1(t+2) / next A1(0) -> buf0
      if (!last)
        stage(cur, 0, 1, 0, t + 2);
      else if (has_next)
        stage(nxt, 0, 1, 0, 0);
      compute(0, 1, 0, acc[0][1]);
      P8P_BAR();
      // phase 4: Q(0,0) of t+1 (buf1); stage A0(t+2) / next A0(0) -> buf0
      P8P_WAITN(6);
      if (!last)
        stage(cur, 0, 0, 0, t + 2);
      else if (has_next)
        stage(nxt, 0, 0, 0, 0);
      compute(0, 0, 1, acc[0][0]);
      P8P_BAR();
      // phase 5: Q(1,0) of t+1; stage B1(t+2) / next B1(0) -> buf0
      if (!last)
        stage(cur, 1, 1, 0, t + 2);
      else if (has_next)
        stage(nxt, 1, 1, 0, 0);
      compute(1, 0, 1, acc[1][0]);
      P8P_BAR();
      // phase 6: Q(1,1) of t+1; stage B0(t+3) / next B0(1) -> buf1
      P8P_WAITN(8);
      if (!last)
        stage(cur, 1, 0, 1, t + 3);
      else if (has_next)
        stage(nxt, 1, 0, 1, 1);
      compute(1, 1, 1, acc[1][1]);
      P8P_BAR();
      // phase 7: Q(0,1) of t+1; stage A1(t+3) / next A1(1) -> buf1
      if (!last)
        stage(cur, 0, 1, 1, t + 3);
      else if (has_next)
        stage(nxt, 0, 1, 1, 1);
      compute(0, 1, 1, acc[0][1]);
      P8P_BAR();
    }
    epilogue(tl);
    cur = nxt;
  }
#undef P8P_WAIT0
#undef P8P_WAITN
#undef P8P_BAR
}

// ---------------------------------------------------------------------------
// Variant 3: 8-phase kernel with fragment reads PIPELINED one phase ahead
// (the guide's "per-phase interleave is the lever"): body p issues the
// ds_reads of phase p+1 right before the MFMAs of phase p, so LDS reads,
// in-flight half-tile DMA and the matrix pipe overlap. Body structure:
//   [vmcnt_p?; s_barrier; stage S(p); lgkmcnt(0); R(p+1); MFMA(p)]
// The stage moves AFTER the barrier (all waves' R(p-1) reads completed
// before the slot is overwritten); the wait schedule shifts one phase
// earlier than variant 2 (derivation in the body comments):
//   before p1: vmcnt(6)  before p3: vmcnt(4)
//   before p5: vmcnt(6)  before p7: vmcnt(4)
// Dual fragment register sets (p&1) keep reads and compute independent.
// ---------------------------------------------------------------------------
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase2_k(GemmArgs args,
                                                                 int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const uint16_t* __restrict__ A = args.A;
  const uint16_t* __restrict__ B = args.B;
  uint16_t* C = args.C;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;
  const int ea = (int)threadIdx.x * 8;

  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int k = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + k, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + k, dst + e);
      }
    }
  };

  f32x4 acc[2][2][4][2] = {};
  // two half-phase (one kk step) fragment register sets, rotated A/B
  short8 fa[2][4], fb[2][2];

  // issue the ds_reads for kk-half KH of quadrant (HA,HB) of tile t into
  // register set S; the compiler tracks the dependency and inserts the
  // counted lgkm waits before the consuming MFMAs (guide: hipcc's
  // ds_read->MFMA scheduling is near-optimal — issue early, don't wait).
  auto rd = [&](int S, int KH, int HA, int HB, int64_t t) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, HA);
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, HB);
    const int kbase = KH * 32 + (lane >> 4) * 8;
#pragma unroll
    for (int fm = 0; fm < 4; ++fm) {
      int row = wr * 64 + fm * 16 + (lane & 15);
      int ks = kbase ^ ((row & 7) << 3);
      fa[S][fm] = *(const short8*)(as + row * PBK + ks);
    }
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      int col = wc * 32 + fn * 16 + (lane & 15);
      int ks = kbase ^ ((col & 7) << 3);
      fb[S][fn] = *(const short8*)(bs + col * PBK + ks);
    }
  };
  auto mm = [&](int S, auto& accq) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        accq[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            fa[S][fm], fb[S][fn], accq[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  // prologue: 6 half-tile stages; wait tile0's A0/B0/A1 prefix; read the
  // first kk-half of Q0(t0) into set 0
  stage(1, 0, 0);
  stage(0, 1, 0);
  stage(0, 0, 0);
  stage(1, 1, 0);
  stage(1, 0, 1);
  stage(0, 1, 1);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  rd(0, 0, 0, 0, 0);

#define PH8B(waitn)                                            do {                                                           if (tail)                                                      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");           else if (waitn > 0)                                            asm volatile("s_waitcnt vmcnt(" #waitn ")" ::: "memory");     __builtin_amdgcn_s_barrier();                                asm volatile("" ::: "memory");                             } while (0)

  // Body p (quadrant Q, tile tau): on entry set0/1 alternates holding
  // (p, kk=0); the body issues (p, kk=1) into the other set, runs kk=0
  // MFMAs, issues (p+1, kk=0) back into the first set, runs kk=1 MFMAs.
  // Reads of body p+1's data are guarded by the vmcnt BEFORE body p's
  // barrier (schedule in the header comment); slot reuse is safe because
  // a stage only runs after the barrier that every wave passed having
  // consumed its reads of that slot.
#define PH8_BODY(SETA, SETB, HA, HB, TT, NHA, NHB, NTT, GUARD)          do {                                                                    rd(SETB, 1, HA, HB, TT);                                              mm(SETA, acc[HA][HB]);                                                if (GUARD) rd(SETA, 0, NHA, NHB, NTT);                                mm(SETB, acc[HA][HB]);                                              } while (0)

  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 4 >= ntiles);
    const bool s2 = (t + 2 < ntiles), s3 = (t + 3 < ntiles);
    const bool last = (t + 2 >= ntiles);
    // p0: Q(0,0) of t; stage A0(t+1); next reads Q(1,0) of t
    PH8B(0);
    stage(0, 0, t + 1);
    PH8_BODY(0, 1, 0, 0, t, 1, 0, t, true);
    // p1: Q(1,0) of t; stage B1(t+1); next Q(1,1) of t (needs B1(t): vmcnt)
    PH8B(6);
    stage(1, 1, t + 1);
    PH8_BODY(0, 1, 1, 0, t, 1, 1, t, true);
    // p2: Q(1,1) of t; stage B0(t+2); next Q(0,1) of t
    PH8B(0);
    if (s2) stage(1, 0, t + 2);
    PH8_BODY(0, 1, 1, 1, t, 0, 1, t, true);
    // p3: Q(0,1) of t; stage A1(t+2); next Q(0,0) of t+1 (needs A0(t+1))
    PH8B(4);
    if (s2) stage(0, 1, t + 2);
    PH8_BODY(0, 1, 0, 1, t, 0, 0, t + 1, true);
    // p4: Q(0,0) of t+1; stage A0(t+2); next Q(1,0) of t+1
    PH8B(0);
    if (s2) stage(0, 0, t + 2);
    PH8_BODY(0, 1, 0, 0, t + 1, 1, 0, t + 1, true);
    // p5: Q(1,0) of t+1; stage B1(t+2); next Q(1,1) of t+1 (needs B1(t+1))
    PH8B(6);
    if (s2) stage(1, 1, t + 2);
    PH8_BODY(0, 1, 1, 0, t + 1, 1, 1, t + 1, true);
    // p6: Q(1,1) of t+1; stage B0(t+3); next Q(0,1) of t+1
    PH8B(0);
    if (s3) stage(1, 0, t + 3);
    PH8_BODY(0, 1, 1, 1, t + 1, 0, 1, t + 1, true);
    // p7: Q(0,1) of t+1; stage A1(t+3); next Q(0,0) of t+2 (needs A0(t+2))
    PH8B(4);
    if (s3) stage(0, 1, t + 3);
    PH8_BODY(0, 1, 0, 1, t + 1, 0, 0, t + 2, !last);
  }
#undef PH8_BODY
#undef PH8B

  // LDS-staged vector epilogue (same as variant 2)
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
#pragma unroll
  for (int ha = 0; ha < 2; ++ha)
#pragma unroll
    for (int hb = 0; hb < 2; ++hb)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int lrow = ha * 128 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int lcol = hb * 128 + wc * 32 + fn * 16 + (lane & 15);
            smem[lrow * 256 + lcol] = f2b(args.alpha * acc[ha][hb][fm][fn][rI]);
          }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Variant 4 (template-faithful phase shape): single fragment set, the
// ds_reads of phase p issue BEFORE the phase barrier — in-wave they hide
// under the in-flight MFMAs of phase p-1, cross-wave under the barrier
// wait. Phase p body:
//     reads_p ; stage S(p) ; BAR_A ; MFMA_p ; [vmcnt?] ; BAR_B
// Slot safety: stage S(p) writes the slot last read by reads_{p-1}, which
// every wave completed before its MFMA_{p-1} (compiler lgkm), i.e. before
// BAR_B of p-1 — and stage S(p) issues after that barrier. The vmcnt
// guarding reads_{p+1} sits before BAR_B of phase p:
//   vmcnt(8) at p1/p5 (guards Q(1,1)'s B1), vmcnt(6) at p3/p7 (guards the
//   next tile's Q(0,0) A0). Same stage schedule as variant 2.
// ---------------------------------------------------------------------------
__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase4_k(GemmArgs args,
                                                                 int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const uint16_t* __restrict__ A = args.A;
  const uint16_t* __restrict__ B = args.B;
  uint16_t* C = args.C;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;
  const int ea = (int)threadIdx.x * 8;

  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int kx = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + kx, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + kx, dst + e);
      }
    }
  };

  f32x4 acc[2][2][4][2] = {};
  short8 fa[2][4], fb[2][2];  // [kk][frag], ONE phase's fragments

  auto rd = [&](int HA, int HB, int64_t t) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, HA);
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, HB);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int kbase = kk * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        fa[kk][fm] = *(const short8*)(as + row * PBK + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        fb[kk][fn] = *(const short8*)(bs + col * PBK + ks);
      }
    }
  };
  auto mm = [&](auto& accq) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          accq[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[kk][fm], fb[kk][fn], accq[fm][fn], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  };

  stage(1, 0, 0);
  stage(0, 1, 0);
  stage(0, 0, 0);
  stage(1, 1, 0);
  stage(1, 0, 1);
  stage(0, 1, 1);
  asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

#define P4BAR()                                    do {                                               asm volatile("" ::: "memory");                   __builtin_amdgcn_s_barrier();                    asm volatile("" ::: "memory");                 } while (0)
#define P4WAIT(n)                                             do {                                                          if (tail)                                                     asm volatile("s_waitcnt vmcnt(0)" ::: "memory");          else                                                          asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory");   } while (0)
#define P4PHASE(HA, HB, TT, SGUARD, SB, SH, ST, WAITC)    do {                                                      rd(HA, HB, TT);                                         if (SGUARD) stage(SB, SH, ST);                          P4BAR();                                                mm(acc[HA][HB]);                                        WAITC;                                                  P4BAR();                                              } while (0)

  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 4 >= ntiles);
    const bool s2 = (t + 2 < ntiles), s3 = (t + 3 < ntiles);
    P4PHASE(0, 0, t, true, 0, 0, t + 1, (void)0);            // p0 +A0(t+1)
    P4PHASE(1, 0, t, true, 1, 1, t + 1, P4WAIT(8));          // p1 +B1(t+1)
    P4PHASE(1, 1, t, s2, 1, 0, t + 2, (void)0);              // p2 +B0(t+2)
    P4PHASE(0, 1, t, s2, 0, 1, t + 2, P4WAIT(6));            // p3 +A1(t+2)
    P4PHASE(0, 0, t + 1, s2, 0, 0, t + 2, (void)0);          // p4 +A0(t+2)
    P4PHASE(1, 0, t + 1, s2, 1, 1, t + 2, P4WAIT(8));        // p5 +B1(t+2)
    P4PHASE(1, 1, t + 1, s3, 1, 0, t + 3, (void)0);          // p6 +B0(t+3)
    P4PHASE(0, 1, t + 1, s3, 0, 1, t + 3, P4WAIT(6));        // p7 +A1(t+3)
  }
#undef P4PHASE
#undef P4WAIT
#undef P4BAR

  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
#pragma unroll
  for (int ha = 0; ha < 2; ++ha)
#pragma unroll
    for (int hb = 0; hb < 2; ++hb)
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
#pragma unroll
          for (int rI = 0; rI < 4; ++rI) {
            int lrow = ha * 128 + wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
            int lcol = hb * 128 + wc * 32 + fn * 16 + (lane & 15);
            smem[lrow * 256 + lcol] = f2b(args.alpha * acc[ha][hb][fm][fn][rI]);
          }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}


// ---------------------------------------------------------------------------
// Variant 5: the variant-2 phase structure on v_mfma_f32_32x32x16_bf16.
// Same half-tile DMA schedule and waits; each phase's quadrant is computed
// as 2x1 fragments of 32x32 over four K=16 steps — half the MFMA
// instruction slots at the higher 32x32 pipe ceiling (2382 vs 2075 TF
// µbench), with identical ds_read count. C/D fragment map (guide §3):
// col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5), reg in [0,16).
// ---------------------------------------------------------------------------
using f32x16 = __attribute__((ext_vector_type(16))) float;

__launch_bounds__(512, 2) __global__ void gemm_bf16_nt_8phase32_k(
    GemmArgs args, int nwg_n) {
  using namespace ph8;
  __shared__ __attribute__((aligned(16))) uint16_t smem[2 * BUF_E];

  const uint16_t* __restrict__ A = args.A;
  const uint16_t* __restrict__ B = args.B;
  uint16_t* C = args.C;
  uint16_t* save_pre = args.save_pre;

  int nwg = gridDim.x;
  int orig = blockIdx.x;
  int xcd = orig % 8, q = nwg / 8, r = nwg % 8;
  int wgid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + orig / 8;
  int64_t tm, tn;
  tile_coords(wgid, nwg / nwg_n, nwg_n, args.group_m, tm, tn);
  const int64_t m0 = tm * PBM;
  const int64_t n0 = tn * PBN;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2x4 wave grid, 64x32 per quadrant

  const int64_t lda = args.lda, ldb = args.ldb;
  const int64_t Mc = args.M - 1, Nc = args.N - 1;
  const int64_t ntiles = args.K / PBK;
  const int ea = (int)threadIdx.x * 8;

  auto stage = [&](int isB, int h, int64_t t) {
    int64_t k0 = t * PBK;
    uint16_t* dst = slot(smem, (int)(t & 1), isB, h);
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int e = ea + c * 4096;
      int row = e >> 6;
      int kx = (e & 63) ^ ((row & 7) << 3);
      if (!isB) {
        int64_t ar = m0 + h * 128 + row;
        if (ar > Mc) ar = Mc;
        glds16(A + ar * lda + k0 + kx, dst + e);
      } else {
        int64_t br = n0 + h * 128 + row;
        if (br > Nc) br = Nc;
        glds16(B + br * ldb + k0 + kx, dst + e);
      }
    }
  };

  // acc[ha][hb]: one 32-col fragment pair (2 fm of 32 rows) per quadrant
  f32x16 acc[2][2][2] = {};

  auto compute = [&](int HA, int HB, int64_t t, auto& accq) {
    const uint16_t* as = slot(smem, (int)(t & 1), 0, HA);
    const uint16_t* bs = slot(smem, (int)(t & 1), 1, HB);
#pragma unroll
    for (int kk = 0; kk < PBK; kk += 16) {
      short8 a_frag[2], b_frag;
      const int kbase = kk + (lane >> 5) * 8;  // 32x32x16: k = (lane>>5)*8+j
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = wr * 64 + fm * 32 + (lane & 31);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(as + row * PBK + ks);
      }
      {
        int col = wc * 32 + (lane & 31);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag = *(const short8*)(bs + col * PBK + ks);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        accq[fm] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a_frag[fm], b_frag,
                                                           accq[fm], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  stage(1, 0, 0);
  stage(0, 1, 0);
  stage(0, 0, 0);
  stage(1, 1, 0);
  stage(1, 0, 1);
  stage(0, 1, 1);

#define PH32_WAIT(n)                                           do {                                                           if (tail)                                                      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");           else                                                           asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory");      __builtin_amdgcn_s_barrier();                                asm volatile("" ::: "memory");                             } while (0)
#define PH32_BAR()                                             do {                                                           asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");           __builtin_amdgcn_s_barrier();                                asm volatile("" ::: "memory");                             } while (0)

  for (int64_t t = 0; t < ntiles; t += 2) {
    const bool tail = (t + 4 >= ntiles);
    const bool s2 = (t + 2 < ntiles), s3 = (t + 3 < ntiles);
    PH32_WAIT(6);
    stage(0, 0, t + 1);
    compute(0, 0, t, acc[0][0]);
    PH32_BAR();
    stage(1, 1, t + 1);
    compute(1, 0, t, acc[1][0]);
    PH32_BAR();
    PH32_WAIT(8);
    if (s2) stage(1, 0, t + 2);
    compute(1, 1, t, acc[1][1]);
    PH32_BAR();
    if (s2) stage(0, 1, t + 2);
    compute(0, 1, t, acc[0][1]);
    PH32_BAR();
    PH32_WAIT(6);
    if (s2) stage(0, 0, t + 2);
    compute(0, 0, t + 1, acc[0][0]);
    PH32_BAR();
    if (s2) stage(1, 1, t + 2);
    compute(1, 0, t + 1, acc[1][0]);
    PH32_BAR();
    PH32_WAIT(8);
    if (s3) stage(1, 0, t + 3);
    compute(1, 1, t + 1, acc[1][1]);
    PH32_BAR();
    if (s3) stage(0, 1, t + 3);
    compute(0, 1, t + 1, acc[0][1]);
    PH32_BAR();
  }
#undef PH32_WAIT
#undef PH32_BAR

  // epilogue via LDS staging; 32x32 C/D map: col=lane&31,
  // row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
#pragma unroll
  for (int ha = 0; ha < 2; ++ha)
#pragma unroll
    for (int hb = 0; hb < 2; ++hb)
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) {
          int lrow = ha * 128 + wr * 64 + fm * 32 + (reg & 3) +
                     8 * (reg >> 2) + 4 * (lane >> 5);
          int lcol = hb * 128 + wc * 32 + (lane & 31);
          smem[lrow * 256 + lcol] = f2b(args.alpha * acc[ha][hb][fm][reg]);
        }
  __syncthreads();
  const bool interior = (m0 + PBM <= args.M) && (n0 + PBN <= args.N);
#pragma unroll 4
  for (int c = 0; c < 16; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 8, col = e & 255;
    int64_t grow = m0 + row;
    int64_t gcol = n0 + col;
    short8 vv = *(const short8*)(smem + e);
    if (interior) {
      int64_t idx = grow * args.N + gcol;
      short8 zout, yout;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx + i]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx + i]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        zout[i] = (short)f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        yout[i] = (short)f2b(v);
      }
      if (save_pre) *(short8*)(save_pre + idx) = zout;
      *(short8*)(C + idx) = yout;
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        if (grow >= args.M || gcol + i >= args.N) continue;
        int64_t idx = grow * args.N + gcol + i;
        float v = b2f((uint16_t)vv[i]);
        if (args.dgelu_pre) v *= gelu_bwd_scalar(b2f(args.dgelu_pre[idx]));
        if (args.beta != 0.f) v += args.beta * b2f(C[idx]);
        if (args.bias) v += b2f(args.bias[gcol + i]);
        if (save_pre) save_pre[idx] = f2b(v);
        if (args.fuse_gelu) v = gelu_fwd_scalar(v);
        C[idx] = f2b(v);
      }
    }
  }
}

// split-K reduce: bf16 C with the full epilogue from S f32 partial slabs
__global__ void splitk_reduce_k(const float* __restrict__ parts, GemmArgs args,
                                int slices) {
  int64_t total = args.M * args.N;
  // fast path: 16-B slab loads + packed stores (the scalar version with its
  // per-element i%N ran ~2.4x off the traffic floor); bias needs per-column
  // addressing and keeps the scalar path (split-K callers never pass it)
  if (!args.bias && (total & 3) == 0) {
    int64_t i4 = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4;
    int64_t stride4 = (int64_t)gridDim.x * blockDim.x * 4;
    for (; i4 < total; i4 += stride4) {
      f32x4 v = {};
      for (int s_ = 0; s_ < slices; ++s_) {
        f32x4 p = *(const f32x4*)(parts + s_ * total + i4);
        v[0] += p[0];
        v[1] += p[1];
        v[2] += p[2];
        v[3] += p[3];
      }
      ushort4 o;
      ushort4 prev;
      if (args.beta != 0.f) prev = *(const ushort4*)(args.C + i4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float x = v[j] * args.alpha;
        if (args.beta != 0.f)
          x += args.beta * b2f(((const uint16_t*)&prev)[j]);
        if (args.save_pre) args.save_pre[i4 + j] = f2b(x);
        if (args.fuse_gelu) x = gelu_fwd_scalar(x);
        ((uint16_t*)&o)[j] = f2b(x);
      }
      *(ushort4*)(args.C + i4) = o;
    }
    return;
  }
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    float v = 0.f;
    for (int s_ = 0; s_ < slices; ++s_) v += parts[s_ * total + i];
    v *= args.alpha;
    if (args.beta != 0.f) v += args.beta * b2f(args.C[i]);
    if (args.bias) v += b2f(args.bias[i % args.N]);
    if (args.save_pre) args.save_pre[i] = f2b(v);
    if (args.fuse_gelu) v = gelu_fwd_scalar(v);
    args.C[i] = f2b(v);
  }
}

thread_local Context* g_splitk_ctx = nullptr;

void launch_gemm(const GemmArgs& args, bool transA, bool transB, int64_t batch,
                 hipStream_t s) {
  if (args.K % BK != 0) {
    // ragged K (non-contract dims): the generic tiled kernel zero-pads the
    // last k-block at the staging loads; all four transpose forms
    dim3 grid((args.N + BN - 1) / BN, (args.M + BM - 1) / BM, batch);
    dim3 block(THREADS);
    if (!transA && !transB)
      gemm_bf16_k<false, false><<<grid, block, 0, s>>>(args);
    else if (!transA && transB)
      gemm_bf16_k<false, true><<<grid, block, 0, s>>>(args);
    else if (transA && !transB)
      gemm_bf16_k<true, false><<<grid, block, 0, s>>>(args);
    else
      gemm_bf16_k<true, true><<<grid, block, 0, s>>>(args);
    NZ_KERNEL_CHECK();
    return;
  }
  static const int kNtVariant = [] {
    const char* v = getenv("NEZHA_GEMM_VARIANT");
    return v ? atoi(v) : 7;  // 7=ping-pong 256^2 (r2 default); 2=8-phase
  }();
  static const int kNtMinWG = [] {
    const char* v = getenv("NEZHA_GEMM_MIN_WG");  // test hook: force 8-phase
    return v ? atoi(v) : 224;
  }();
  static const bool kMinWgForced = getenv("NEZHA_GEMM_MIN_WG") != nullptr;
  static const int kPersDiag = [] {
    const char* v = getenv("NEZHA_PERS_DIAG");
    return v ? atoi(v) : 0;
  }();
  const_cast<GemmArgs&>(args).diag_no_store = kPersDiag;
  static const int kGroupM = [] {
    const char* v = getenv("NEZHA_GEMM_GROUP_M");
    return v ? atoi(v) : 8;  // 0 = legacy m-major raster
  }();
  const_cast<GemmArgs&>(args).group_m = kGroupM;
  // The 256^2 kernel runs ~1 workgroup per CU, so a grid that is not close
  // to a multiple of 256 leaves whole CUs idle for a full round (N=768 ->
  // 384 WGs = 25% idle; measured 679 vs 884 TF against the 128^2 kernel).
  auto balanced_256 = [&](int nwg) {
    return kMinWgForced || nwg % 256 == 0 || nwg >= 1024;
  };
  if (!transA && transB && batch == 1 && kNtVariant == 5 &&
      args.K % 128 == 0 && args.K >= 1536) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG) {
      dim3 grid(nwg_m * nwg_n);
      gemm_bf16_nt_8phase32_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // short-K shapes (BERT K=768 panels): pp128 persistent ping-pong —
  // BN=128 frees the registers the 256^2 variant lacked; pipeline flows
  // across tiles (no per-tile ramp), countable LDS-staged epilogue.
  // short-K NT (BERT K=768 panels): 128^2 ping-pong at 2 WGs/CU —
  // the de-synchronized pair hides prologue/wait/epilogue bubbles
  // NEZHA_GEMM_64Q: 0 = off (fall through to the r1 persistent kernel),
  // 1 = 8-thin-phase pp64q (DEFAULT), 4 = 4-fat-phase pp64q4.
  // The 4-phase variant measured 501-523 TF vs the 8-phase's 519-533 on
  // the BERT panels (profiles/r2/r2_pp64q4.log) — merging phases does not
  // help because the pace-setter is DMA-service latency surfacing through
  // the stage-wait certificates, not barrier count — AND it has a rare
  // box-dependent nondeterminism (profiles/r2/r2_q4race*.log: a single
  // 128 B glds beat of a B0 fragment goes stale; full-drain mode is clean
  // but run-to-run variance defeated the wait-site bisection). Kept as an
  // opt-in research artifact; NOT on any default path.
  static const int k64qMode = [] {
    const char* v = getenv("NEZHA_GEMM_64Q");
    return v ? atoi(v) : 1;
  }();
  static const bool k64qOff = k64qMode == 0;
  if (!transA && transB && batch == 1 && !k64qOff && kNtVariant >= 7 &&
      args.K % 128 == 0 && args.K >= 256 && args.K < 1536) {
    int nwg_m4 = (int)((args.M + 127) / 128);
    int nwg_n4 = (int)((args.N + 127) / 128);
    int64_t nwg4 = (int64_t)nwg_m4 * nwg_n4;
    if (nwg4 >= 1024) {  // >= 2 rounds of the 512-WG chip
      // XCD-slab raster (see kernel comment): measured a wash on the
      // BERT panels (511 vs 532 TF, profiles/r2/r2_skraster.log) — the
      // per-WG glds stream rate, not L2 locality, paces short-K staging.
      // Kept as an opt-in (NEZHA_GEMM_SKRASTER=1) for future A/B.
      static const bool kSkRaster = [] {
        const char* v = getenv("NEZHA_GEMM_SKRASTER");
        return v && atoi(v) == 1;
      }();
      GemmArgs a4 = args;  // inherits dgelu_pre (route supports it)
      // slab raster only implemented in the 8-phase kernel
      if (kSkRaster && nwg_m4 % 8 == 0 && k64qMode != 4) a4.group_m = -1;
      dim3 grid((uint32_t)nwg4);
      if (k64qMode == 4)
        gemm_bf16_pp64q4_k<<<grid, dim3(512), 0, s>>>(a4, nwg_n4);
      else
        gemm_bf16_pp64q_k<<<grid, dim3(512), 0, s>>>(a4, nwg_n4);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // opt-in (NEZHA_GEMM_PP128=1): at 1 WG/CU its 16-MFMA phases leave the
  // SIMD pipe idle between barriers and it measures 397-416 TF vs the r1
  // persistent kernel's 537-542 on the BERT panels (profiles/r2/r2_pp128)
  static const bool kPP128On = getenv("NEZHA_GEMM_PP128") &&
                               atoi(getenv("NEZHA_GEMM_PP128")) == 1;
  if (!transA && transB && kPP128On && kNtVariant >= 7 &&
      args.K % 128 == 0 && args.K >= 256 && args.K < 1536) {
    int nwg_m8 = (int)((args.M + 255) / 256);
    int nwg_n8 = (int)((args.N + 127) / 128);
    int64_t per_b8 = (int64_t)nwg_m8 * nwg_n8;
    int64_t total8 = per_b8 * batch;
    if (total8 >= 512) {
      gemm_bf16_pp128_pers_k<0><<<dim3(256), dim3(512), 0, s>>>(
          args, nwg_n8, per_b8, total8);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // short-K persistent ping-pong: EXPERIMENTAL opt-in (NEZHA_GEMM_PERS_PP=1).
  // The cross-tile flow + deep schedule are simulator-verified, but the
  // epilogue's register pressure fragments the MFMA clusters (303 scratch
  // ops in the hot span) and it measures at/below the 2-WG/CU persistent
  // kernel; the fix is an acc-halving BN=128 regeometry — future work.
  static const bool kPersPP =
      getenv("NEZHA_GEMM_PERS_PP") && atoi(getenv("NEZHA_GEMM_PERS_PP")) == 1;
  if (!transA && transB && kPersPP && kNtVariant >= 7 && args.K % 128 == 0 &&
      args.K >= 256 && args.K < 1536) {
    int nwg_m8 = (int)((args.M + 255) / 256);
    int nwg_n8 = (int)((args.N + 255) / 256);
    int64_t per_b8 = (int64_t)nwg_m8 * nwg_n8;
    int64_t total8 = per_b8 * batch;
    if (total8 >= 512) {
      gemm_bf16_nt_pingpong_pers_k<<<dim3(256), dim3(512), 0, s>>>(
          args, nwg_n8, per_b8, total8);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // ping-pong single-tile for deep K (NEZHA_GEMM_V7_MINK lowers the K
  // threshold for A/B against the short-K kernels; default keeps 256^2
  // on K >= 1536 where the ramp is amortized)
  static const int kV7MinK = [] {
    const char* v = getenv("NEZHA_GEMM_V7_MINK");
    return v ? atoi(v) : 1536;
  }();
  if (!transA && transB && batch == 1 && kNtVariant == 7 &&
      args.K % 128 == 0 && args.K >= kV7MinK && args.K / 64 >= 4) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG &&
        balanced_256(nwg_m * nwg_n)) {
      dim3 grid(nwg_m * nwg_n);
      static const int kPP = [] {
        const char* v = getenv("NEZHA_GEMM_PP");
        return v ? atoi(v) : 1;  // static prio for the younger half
      }();
      GemmArgs a7 = args;  // inherits dgelu_pre (route supports it)
      switch (kPP & 3) {
        case 1: gemm_bf16_nt_pingpong_k<1><<<grid, dim3(512), 0, s>>>(a7, nwg_n); break;
        case 2: gemm_bf16_nt_pingpong_k<2><<<grid, dim3(512), 0, s>>>(a7, nwg_n); break;
        case 3: gemm_bf16_nt_pingpong_k<3><<<grid, dim3(512), 0, s>>>(a7, nwg_n); break;
        default: gemm_bf16_nt_pingpong_k<0><<<grid, dim3(512), 0, s>>>(a7, nwg_n); break;
      }
      NZ_KERNEL_CHECK();
      return;
    }
  }
  if (!transA && transB && batch == 1 && kNtVariant == 6 &&
      args.K % 128 == 0 && args.K / 64 >= 6) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG &&
        balanced_256(nwg_m * nwg_n)) {
      dim3 grid(nwg_m * nwg_n);
      gemm_bf16_nt_8phase6_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  if (!transA && transB && batch == 1 && kNtVariant == 4 &&
      args.K % 128 == 0 && args.K / 64 >= 6) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG) {
      dim3 grid(nwg_m * nwg_n);
      gemm_bf16_nt_8phase4_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  if (!transA && transB && batch == 1 && kNtVariant == 3 &&
      args.K % 128 == 0 && args.K / 64 >= 6) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG) {
      dim3 grid(nwg_m * nwg_n);
      gemm_bf16_nt_8phase2_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // Persistent 8-phase (NEZHA_GEMM_PERS8=1, measurement opt-in): pipeline
  // flows across tiles via the prologue/s2-s3 stage correspondence — but at
  // 1 WG/CU the per-tile phase-0 drain and scatter epilogue have no
  // co-resident WG to hide under, and it measures BELOW the 2-WG/CU 128^2
  // persistent kernel on every contract shape (qkv-fwd 504 vs 566 TF,
  // ffn2-fwd 712 vs 874; square-4k a wash at 927 vs 908). Kept for A/B
  // evidence; see docs/perf/README.md.
  if (!transA && transB && kNtVariant >= 2 && args.K % 128 == 0 &&
      args.M >= 256 && args.N >= 256) {
    static const bool kP8On =
        getenv("NEZHA_GEMM_PERS8") && atoi(getenv("NEZHA_GEMM_PERS8")) == 1;
    int nwg_m8 = (int)((args.M + 255) / 256);
    int nwg_n8 = (int)((args.N + 255) / 256);
    int64_t per_b8 = (int64_t)nwg_m8 * nwg_n8;
    int64_t total8 = per_b8 * batch;
    if (kP8On && total8 >= 256) {
      gemm_bf16_nt_8phase_pers_k<<<dim3(256), dim3(512), 0, s>>>(
          args, nwg_n8, per_b8, total8);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  // short-K shapes (BERT K=768) amortize the 256^2 pipeline poorly; the
  // 128^2 kernel measures faster there (more workgroups, shorter ramp).
  if (!transA && transB && batch == 1 && kNtVariant >= 2 &&
      args.K % 128 == 0 && args.K >= 1536) {
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 255) / 256);
    // 256^2 tiles only when the grid fills the chip (1 block/CU, 256 CUs);
    // small-N/M shapes run faster on the 128^2 2-buffer kernel below.
    if (nwg_m * (int64_t)nwg_n >= kNtMinWG &&
        balanced_256(nwg_m * nwg_n)) {
      dim3 grid(nwg_m * nwg_n);
      gemm_bf16_nt_8phase_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
      NZ_KERNEL_CHECK();
      return;
    }
  }
  if (!transA && transB && batch == 1 && kNtVariant == 1 && args.K / 64 >= 4) {
    // deep-pipelined NT (kept for A/B comparisons; 2-buf won on occupancy)
    int nwg_m = (int)((args.M + 255) / 256);
    int nwg_n = (int)((args.N + 127) / 128);
    dim3 grid(nwg_m * nwg_n);
    gemm_bf16_nt_deep_k<<<grid, dim3(512), 0, s>>>(args, nwg_n);
    NZ_KERNEL_CHECK();
    return;
  }
  if (!transA && transB) {
    // fast NT path: glds-staged, XCD-swizzled (see gemm_bf16_nt_k)
    int nwg_m = (int)((args.M + BM - 1) / BM);
    int nwg_n = (int)((args.N + BN - 1) / BN);
    int nwg = nwg_m * nwg_n;
    static const bool kPersOff =
        getenv("NEZHA_GEMM_PERS") && atoi(getenv("NEZHA_GEMM_PERS")) == 0;
    // Split-K (BERT wgrads: tiny MxN against K=32k left 85% of the chip
    // idle): run K-slices as the batch dim into f32 partial slabs, then a
    // fused reduce applies the real epilogue.
    if (batch == 1 && nwg < 224 && args.K >= 4096 && !args.out_f32 &&
        g_splitk_ctx) {
      // Prefer 256^2 tiles per slice (8-phase kernel): half the staging
      // bytes per flop of the 128^2 kernel, and the >=768-deep per-slice K
      // amortizes the 8-phase ramp. Measured on the BERT wgrad panels:
      // qkv 464 -> ~800 TF equivalent (see profiles/).
      int nwg_m8 = (int)((args.M + 255) / 256);
      int nwg_n8 = (int)((args.N + 255) / 256);
      int64_t nwg8 = (int64_t)nwg_m8 * nwg_n8;
      if (kNtVariant >= 2 && args.K % 256 == 0 && args.K / 2 >= 768 &&
          nwg8 >= 9) {
        int sl = 1;
        while (sl < 16 && nwg8 * sl * 2 <= 288 &&
               args.K % ((int64_t)sl * 256) == 0 && args.K / (sl * 2) >= 768)
          sl *= 2;
        if (sl > 1 && nwg8 * sl >= 96) {
          int64_t ksl = args.K / sl;
          Tensor parts(g_splitk_ctx, DType::F32,
                       Shape{(int64_t)sl, args.M, args.N});
          GemmArgs pa = args;
          pa.K = ksl;
          pa.sA = ksl;
          pa.sB = ksl;
          pa.sC = args.M * args.N;
          pa.out_f32 = (float*)parts.data();
          dim3 g8((uint32_t)nwg8, sl);
          if (kNtVariant >= 7)  // ping-pong slices (+15% on deep-K tiles)
            gemm_bf16_nt_pingpong_k<1><<<g8, dim3(512), 0, s>>>(pa, nwg_n8);
          else
            gemm_bf16_nt_8phase_k<<<g8, dim3(512), 0, s>>>(pa, nwg_n8);
          NZ_KERNEL_CHECK();
          splitk_reduce_k<<<dim3(grid_1d(args.M * args.N, 256)), dim3(256), 0,
                            s>>>((const float*)parts.data(), args, sl);
          NZ_KERNEL_CHECK();
          return;
        }
      }
      // Power-of-two slice count: double until the grid reaches ~2 WGs/CU
      // (the 128^2 kernel's occupancy), keeping K/slices a >=2048 multiple
      // of 64. A non-power-of-two count (the old (448+nwg-1)/nwg) usually
      // failed the divisibility test and silently disabled split-K — the
      // BERT qkv wgrad then ran 108 WGs on 256 CUs at 227 TF.
      static const int kSplitkTarget = [] {
        const char* v = getenv("NEZHA_SPLITK_TARGET");
        return v ? atoi(v) : 512;
      }();
      int slices = 1;
      while (slices < 16 && nwg * slices < kSplitkTarget &&
             args.K % ((int64_t)slices * 128) == 0 &&
             args.K / (slices * 2) >= 2048)
        slices *= 2;
      if (slices > 1 && args.K % ((int64_t)slices * 64) == 0) {
        int64_t ksl = args.K / slices;
        Tensor parts(g_splitk_ctx, DType::F32,
                     Shape{(int64_t)slices, args.M, args.N});
        GemmArgs pa = args;
        pa.K = ksl;
        pa.sA = ksl;  // lda stays K: slice s starts at column s*ksl
        pa.sB = ksl;
        pa.sC = args.M * args.N;
        pa.out_f32 = (float*)parts.data();
        int64_t sk_total = (int64_t)nwg * slices;
        if (!kPersOff && sk_total >= 512) {
          // slices ride the persistent kernel's batch dimension (per-slice
          // strides sA/sB/sC are exactly the batched-pointer contract)
          gemm_bf16_nt_pers_k<<<dim3(512), dim3(THREADS), 0, s>>>(
              pa, nwg_n, nwg, sk_total);
        } else {
          dim3 gridk(nwg, slices);
          gemm_bf16_nt_k<<<gridk, dim3(THREADS), 0, s>>>(pa, nwg_n);
        }
        NZ_KERNEL_CHECK();
        splitk_reduce_k<<<dim3(grid_1d(args.M * args.N, 256)), dim3(256), 0,
                          s>>>((const float*)parts.data(), args, slices);
        NZ_KERNEL_CHECK();
        return;
      }
    }
    // Persistent sweep: amortizes the stage prologue over many tiles and
    // equalizes per-WG work (see gemm_bf16_nt_pers_k). Wins for short-K
    // shapes and small batched tiles; neutral for long K.
    int64_t per_b = (int64_t)nwg_m * nwg_n;
    int64_t total = per_b * batch;
    // Ring variant (half-tile rings, 3-tile A latency cover): measurement
    // opt-in via NEZHA_GEMM_RING=1 — race-screen clean but measured ~1 us
    // per k-iteration SLOWER than the 2-buffer persistent kernel at short
    // K (367 vs 582 TF on qkv-fwd) and only equal at deep K; see
    // docs/perf/README.md for the analysis.
    static const bool kRingOn =
        getenv("NEZHA_GEMM_RING") && atoi(getenv("NEZHA_GEMM_RING")) == 1;
    if (kRingOn && !kPersOff && total >= 512 && !args.out_f32 &&
        args.K / BK >= 3) {
      gemm_bf16_nt_ring_pers_k<<<dim3(512), dim3(THREADS), 0, s>>>(
          args, nwg_n, per_b, total);
      NZ_KERNEL_CHECK();
      return;
    }
    if (!kPersOff && total >= 512 && !args.out_f32) {
      gemm_bf16_nt_pers_k<<<dim3(512), dim3(THREADS), 0, s>>>(args, nwg_n,
                                                              per_b, total);
      NZ_KERNEL_CHECK();
      return;
    }
    dim3 grid(nwg, batch);
    gemm_bf16_nt_k<<<grid, dim3(THREADS), 0, s>>>(args, nwg_n);
    NZ_KERNEL_CHECK();
    return;
  }
  dim3 grid((args.N + BN - 1) / BN, (args.M + BM - 1) / BM, batch);
  dim3 block(THREADS);
  if (!transA && !transB)
    gemm_bf16_k<false, false><<<grid, block, 0, s>>>(args);
  else if (transA && !transB)
    gemm_bf16_k<true, false><<<grid, block, 0, s>>>(args);
  else
    gemm_bf16_k<true, true><<<grid, block, 0, s>>>(args);
  NZ_KERNEL_CHECK();
}

}  // namespace

// The fast kernel is the NT form (both operands k-contiguous -> lane-linear
// glds staging). NN and TN are LOWERED to NT by transposing the k-major
// operand(s) through the vectorized tile-transpose — the transpose costs
// ~2 passes over the operand (~tens of us) vs the GEMM's hundreds, and NT
// then runs ~4x the scatter-staged generic kernel. TT stays on the generic
// tiled kernel (test-only form).
// Mirrors the NT dispatch in hip::gemm: true iff the route chosen for this
// shape is one of the kernels implementing the fused GELU-derivative
// epilogue (pp64q / pp64q4 for short K, the v7 ping-pong for deep K).
// KEEP IN SYNC with the branch conditions in hip::gemm — callers gate the
// dgelu request on this, and hip::gemm NZ_CHECKs it.
bool gemm_can_fuse_dgelu(int64_t M, int64_t N, int64_t K, bool transA,
                         bool transB) {
  if (transA || !transB) return false;
  if (K % 128 != 0) return false;
  static const int variant = [] {
    const char* v = getenv("NEZHA_GEMM_VARIANT");
    return v ? atoi(v) : 7;
  }();
  if (variant < 7) return false;
  if (K >= 256 && K < 1536) {  // pp64q / pp64q4 route
    static const int mode64q = [] {
      const char* v = getenv("NEZHA_GEMM_64Q");
      return v ? atoi(v) : 1;
    }();
    if (mode64q == 0) return false;
    int64_t nwg4 = ((M + 127) / 128) * ((N + 127) / 128);
    return nwg4 >= 1024;
  }
  // deep-K (v7 256^2) route: the kernel implements the epilogue, but the
  // inline pre-activation read stalls its staged epilogue and measured a
  // NET LOSS on the MLP dgrads (2487 vs 2547 k samples/s,
  // profiles/r2/r2_dgelu.log) — the standalone gelu_bwd pass streams at
  // full bandwidth instead. Fusion is therefore short-K (pp64q) only.
  return false;
}

static bool can_fast_transpose(int64_t R, int64_t C) {
  return R % 64 == 0 && C % 64 == 0;
}

void gemm(const Tensor& A, const Tensor& B, Tensor& C, bool transA, bool transB,
          float alpha, float beta, const ops::GemmEpilogue* epi) {
  if (A.dtype() == DType::F32) {
    NZ_CHECK(B.dtype() == DType::F32 && C.dtype() == DType::F32,
             "hip gemm: mixed dtypes");
    gemm_f32(A, B, C, transA, transB, alpha, beta, epi);
    return;
  }
  NZ_CHECK(A.dtype() == DType::BF16 && B.dtype() == DType::BF16 &&
               C.dtype() == DType::BF16,
           "hip gemm: bf16 tensors required");
  GemmArgs args{};
  args.M = C.rows();
  args.N = C.cols();
  args.K = transA ? A.rows() : A.cols();
  NZ_CHECK((transA ? A.cols() : A.rows()) == args.M, "gemm M mismatch");
  NZ_CHECK((transB ? B.rows() : B.cols()) == args.N, "gemm N mismatch");
  NZ_CHECK((transB ? B.cols() : B.rows()) == args.K, "gemm K mismatch");
  args.A = (const uint16_t*)A.data();
  args.B = (const uint16_t*)B.data();
  args.C = (uint16_t*)C.data();
  args.lda = A.cols();
  args.ldb = B.cols();
  args.alpha = alpha;
  args.beta = beta;
  const uint16_t* dgelu_src = nullptr;
  if (epi) {
    args.bias = epi->bias ? (const uint16_t*)epi->bias->data() : nullptr;
    args.save_pre = epi->save_pre ? (uint16_t*)epi->save_pre->data() : nullptr;
    args.fuse_gelu = epi->gelu ? 1 : 0;
    dgelu_src = epi->dgelu_pre ? (const uint16_t*)epi->dgelu_pre->data()
                               : nullptr;
  }
  // dgelu_pre is handed only to routes whose kernels implement the fused
  // derivative; callers must consult gemm_can_fuse_dgelu first (it mirrors
  // the dispatch conditions below — keep the two in sync).
  NZ_CHECK(!dgelu_src || gemm_can_fuse_dgelu(args.M, args.N, args.K, transA,
                                             transB),
           "dgelu epilogue requested on an unsupported GEMM route");
  args.dgelu_pre = dgelu_src;
  // r2: NATIVE NN/TN ping-pong — consume the stored layouts directly via
  // transpose-in-stage (TRMODE), skipping the lowering transposes below
  // entirely (they were 6-9% of both model steps: r2 kernel stats).
  {
    static const int kNtVariantN = [] {
      const char* v = getenv("NEZHA_GEMM_VARIANT");
      return v ? atoi(v) : 7;
    }();
    // EXPERIMENTAL opt-in (NEZHA_GEMM_NATIVE_TR=1): correct on every
    // oracle (race screen NN/TN rows) but the transpose-in-stage register
    // pressure (4 staged short8 + scatter address chains on a kernel
    // already at 255 VGPR) spills ~500 B and fragments the MFMA clusters
    // — 72-90 TF vs 970-1025 via the lowering (profiles/r2/r2_native2).
    // Unblocking needs the BN=128 regeometry (halves acc VGPRs) that the
    // persistent ping-pong also waits on — round-3 work.
    static const bool kNativeOff = !(getenv("NEZHA_GEMM_NATIVE_TR") &&
                                     atoi(getenv("NEZHA_GEMM_NATIVE_TR")) == 1);
    int trmode = 0;
    if (!transA && !transB) trmode = 1;       // NN: B is [K,N]
    else if (transA && !transB) trmode = 3;   // TN: A is [K,M], B is [K,N]
    if (trmode && kNtVariantN >= 7 && !kNativeOff && args.K % 128 == 0 &&
        args.K >= 1536 && args.N % 8 == 0 &&
        (trmode != 3 || args.M % 8 == 0)) {
      int nwg_m = (int)((args.M + 255) / 256);
      hipStream_t st = stream_of(C);
      int nwg_n128 = (int)((args.N + 127) / 128);
      int64_t per_b128 = (int64_t)nwg_m * nwg_n128;
      auto launch_native = [&](const GemmArgs& a, int64_t nslices) {
        // pp128 persistent sweep; split-K slices ride the batch dimension
        int64_t tot = per_b128 * nslices;
        if (trmode == 1)
          gemm_bf16_pp128_pers_k<1><<<dim3(256), dim3(512), 0, st>>>(
              a, nwg_n128, per_b128, tot);
        else
          gemm_bf16_pp128_pers_k<3><<<dim3(256), dim3(512), 0, st>>>(
              a, nwg_n128, per_b128, tot);
        NZ_KERNEL_CHECK();
      };
      if (per_b128 >= 256) {  // big grid: no K split needed
        launch_native(args, 1);
        return;
      }
      // small-grid deep-K (BERT wgrads): split-K slices, f32 partial
      // slabs + fused reduce — no giant operand transposes
      if (args.K >= 4096 && per_b128 >= 9) {
        int sl = 1;
        while (sl < 16 && per_b128 * sl * 2 <= 576 &&
               args.K % ((int64_t)sl * 256) == 0 && args.K / (sl * 2) >= 768)
          sl *= 2;
        if (sl > 1 && per_b128 * sl >= 192) {
          int64_t ksl = args.K / sl;
          Tensor parts(C.ctx(), DType::F32, Shape{(int64_t)sl, args.M, args.N});
          GemmArgs pa = args;
          pa.K = ksl;
          // per-slice k-offsets in ELEMENTS of each stored layout
          pa.sA = (transA ? ksl * args.lda : ksl);
          pa.sB = ksl * args.ldb;  // B is [K,N] in both native modes
          pa.sC = args.M * args.N;
          pa.out_f32 = (float*)parts.data();
          launch_native(pa, sl);
          splitk_reduce_k<<<dim3(grid_1d(args.M * args.N, 256)), dim3(256),
                            0, st>>>((const float*)parts.data(), args, sl);
          NZ_KERNEL_CHECK();
          return;
        }
      }
    }
  }
  Tensor at_scratch, bt_scratch;  // keep lowered operands alive to launch
  if (args.K % 64 == 0) {
    if (!transA && !transB && can_fast_transpose(args.K, args.N)) {
      // NN: C = A.B = NT(A, B^T)
      bt_scratch = Tensor(C.ctx(), DType::BF16, Shape{args.N, args.K});
      transpose2d(B, bt_scratch);
      args.B = (const uint16_t*)bt_scratch.data();
      args.ldb = args.K;
      transB = true;
      transA = false;
    } else if (transA && !transB && can_fast_transpose(args.K, args.M) &&
               can_fast_transpose(args.K, args.N)) {
      // TN: C = A^T.B = NT(A^T, B^T)
      at_scratch = Tensor(C.ctx(), DType::BF16, Shape{args.M, args.K});
      transpose2d(A, at_scratch);
      bt_scratch = Tensor(C.ctx(), DType::BF16, Shape{args.N, args.K});
      transpose2d(B, bt_scratch);
      args.A = (const uint16_t*)at_scratch.data();
      args.B = (const uint16_t*)bt_scratch.data();
      args.lda = args.K;
      args.ldb = args.K;
      transA = false;
      transB = true;
    }
  }
  g_splitk_ctx = C.ctx();
  launch_gemm(args, transA, transB, 1, stream_of(C));
}

void gemm_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                  bool transB, float alpha, int64_t batch, int64_t M, int64_t N,
                  int64_t K, int64_t sA, int64_t sB, int64_t sC) {
  if (A.dtype() == DType::F32) {
    gemm_f32_batched(A, B, C, transA, transB, alpha, batch, M, N, K, sA, sB, sC);
    return;
  }
  NZ_CHECK(A.dtype() == DType::BF16, "hip gemm_batched: bf16 required");
  GemmArgs args{};
  args.A = (const uint16_t*)A.data();
  args.B = (const uint16_t*)B.data();
  args.C = (uint16_t*)C.data();
  args.M = M;
  args.N = N;
  args.K = K;
  args.lda = transA ? M : K;
  args.ldb = transB ? K : N;
  args.sA = sA;
  args.sB = sB;
  args.sC = sC;
  args.alpha = alpha;
  args.beta = 0.f;
  // Lower batched NN / TN to the fast NT kernel via batched transposes
  // (attention P.V, dV, dK paths). Scratch = [batch, rows, cols] per op.
  Tensor at_scratch, bt_scratch;
  if (K % 64 == 0) {
    if (!transA && !transB && can_fast_transpose(K, N)) {
      bt_scratch = Tensor(C.ctx(), DType::BF16, Shape{batch, N, K});
      transpose2d_batched(B, bt_scratch, batch, K, N, sB, N * K);
      args.B = (const uint16_t*)bt_scratch.data();
      args.ldb = K;
      args.sB = N * K;
      transB = true;
    } else if (transA && !transB && can_fast_transpose(K, M) &&
               can_fast_transpose(K, N)) {
      at_scratch = Tensor(C.ctx(), DType::BF16, Shape{batch, M, K});
      transpose2d_batched(A, at_scratch, batch, K, M, sA, M * K);
      bt_scratch = Tensor(C.ctx(), DType::BF16, Shape{batch, N, K});
      transpose2d_batched(B, bt_scratch, batch, K, N, sB, N * K);
      args.A = (const uint16_t*)at_scratch.data();
      args.B = (const uint16_t*)bt_scratch.data();
      args.lda = K;
      args.ldb = K;
      args.sA = M * K;
      args.sB = N * K;
      transA = false;
      transB = true;
    }
  }
  g_splitk_ctx = C.ctx();
  launch_gemm(args, transA, transB, batch, stream_of(C));
}

}  // namespace hip
}  // namespace nz
