// Fused attention (K5), forward AND backward, for the BERT contract shape
// S=128, dh=64: one workgroup per (batch, head). Forward computes
// QK^T -> scaled softmax -> P.V entirely on-chip (the S x S score tile
// never touches HBM; P is written once for backward). Backward
// (attn_bwd_fused_k below) computes dP, the softmax Jacobian, and
// dQ/dK/dV in one kernel. Both kernels address q/k/v/outputs through
// generalized strides so they consume the packed qkv [B,S,3H] projection
// and produce the merged [B,S,H] layout directly.
//
// LDS plan (49.5 KB -> 3 blocks/CU):
//   region Q  [128][64]  16 KB  (glds lane-linear, XOR-swizzled)  \ reused as
//   region K  [128][64]  16 KB  (glds)                            / P [128][136]
//   region VT [64][136]  17 KB  (V transposed via register scatter; +8 pad
//                                gives row stride 272 B = 4 banks mod 64 ->
//                                conflict-free column-of-rows reads)
// Waves: 8 (512 threads). QK^T: 2x4 wave grid, 64x32 each (NT, 16 MFMA).
// Softmax: one wave per 4 rows, row in registers (2 elems/lane), wave
// shuffles for max/sum; writes P to LDS + global. PV: 2x2 wave grid,
// 64x32 each over K=S (NT against VT, 32 MFMA).
#include "src/hip/hip_common.h"
#include "src/hip/hip_ops.h"

namespace nz {
namespace hip {

using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace {

constexpr int AS = 128;   // sequence tile (== S)
constexpr int AD = 64;    // head dim
constexpr int PSTRIDE = AS + 8;  // P/VT row stride in elements (272 B)

__device__ __forceinline__ void att_glds16(const uint16_t* gsrc,
                                           uint16_t* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t*)gsrc,
      (__attribute__((address_space(3))) uint32_t*)lds_dst, 16, 0, 0);
}

// Generalized input/output addressing so the kernel can read q/k/v either
// from per-head-contiguous [BH, S, dh] tensors (qr=64, qh=0, qb=S*64) or
// straight from the packed qkv projection [B, S, 3H] (qr=3H, qh=64,
// qb=S*3H, with k/v base pointers offset by H / 2H) — the latter removes
// the attn_split/attn_merge layout passes entirely. Same for `out`.
__launch_bounds__(512, 2) __global__ void attn_fwd_fused_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ probs,
    uint16_t* __restrict__ out, float scale, int heads, int64_t qb,
    int64_t qh, int64_t qr, int64_t ob, int64_t oh, int64_t orr) {
  // 512 threads (8 waves as 2x4) at the same 49.5 KB LDS / 3 WGs/CU:
  // 6 waves/SIMD to hide the phase barriers (the 256-thread version ran 3).
  __shared__ __attribute__((aligned(16))) uint16_t smem[(AS + AD) * PSTRIDE];
  uint16_t* Qs = smem;
  uint16_t* Ks = smem + AS * AD;
  uint16_t* Ps = smem;                      // overlays Q..K after QK^T
  uint16_t* VTs = smem + AS * PSTRIDE;

  const int64_t bh = blockIdx.x;            // batch*heads index
  const int64_t b_ = bh / heads, h_ = bh % heads;
  const uint16_t* qg = q + b_ * qb + h_ * qh;
  const uint16_t* kg = k + b_ * qb + h_ * qh;
  const uint16_t* vg = v + b_ * qb + h_ * qh;
  uint16_t* pg = probs + bh * AS * AS;
  uint16_t* og = out + b_ * ob + h_ * oh;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  // ---- stage Q,K by glds (swizzled source), V to registers ---------------
  const int e0 = (int)threadIdx.x * 8;
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int e = e0 + c * 4096;
    int row = e >> 6;
    int kk = (e & 63) ^ ((row & 7) << 3);
    att_glds16(qg + row * qr + kk, Qs + e);
    att_glds16(kg + row * qr + kk, Ks + e);
  }
  short8 vreg[2];
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int e = e0 + c * 4096;                  // element in [s][d]
    vreg[c] = *(const short8*)(vg + (e >> 6) * qr + (e & 63));
  }
  // scatter V^T while the DMA flies: (s,d..d+7) -> VT[d+i][s]
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int e = e0 + c * 4096;
    int s = e >> 6, d = e & 63;
#pragma unroll
    for (int i = 0; i < 8; ++i)
      VTs[(d + i) * PSTRIDE + s] = (uint16_t)vreg[c][i];
  }
  __syncthreads();  // drains glds (vmcnt0) + V scatter visible

  // ---- QK^T: wave (wr,wc) of 2x4 computes 64 rows x 32 cols ---------------
  {
    const int wr = wid >> 2, wc = wid & 3;
    f32x4 acc[4][2] = {};
#pragma unroll
    for (int kk = 0; kk < AD; kk += 32) {
      short8 a_frag[4], b_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(Qs + row * AD + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(Ks + col * AD + ks);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();  // everyone done READING Qs/Ks before P overlays them
    // write scores (bf16) into the P overlay
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int row = wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
          int col = wc * 32 + fn * 16 + (lane & 15);
          Ps[row * PSTRIDE + col] = f2b(acc[fm][fn][rI]);
        }
  }
  __syncthreads();

  // ---- softmax: wave w handles rows w, w+8, ... (2 elems per lane) -------
  for (int row = wid; row < AS; row += 8) {
    const uint16_t* pr = Ps + row * PSTRIDE;
    float v0 = scale * b2f(pr[lane * 2]);
    float v1 = scale * b2f(pr[lane * 2 + 1]);
    float m = wave_max(fmaxf(v0, v1));
    v0 = __expf(v0 - m);
    v1 = __expf(v1 - m);
    float inv = 1.0f / wave_sum(v0 + v1);
    uint16_t o0 = f2b(v0 * inv), o1 = f2b(v1 * inv);
    uint16_t* pw = Ps + row * PSTRIDE;
    pw[lane * 2] = o0;
    pw[lane * 2 + 1] = o1;
    // global P write deferred to after PV: a store here sits in the vmcnt
    // FIFO that the next __syncthreads drains (measured SQ_WAIT_ANY =
    // 12.7x MFMA cycles on this kernel)
  }
  __syncthreads();

  // ---- P.V: output 128x64; wave (wr,wc) of 2x4 computes 64x16 ------------
  {
    const int wr = wid >> 2, wc = wid & 3;
    f32x4 acc[4] = {};
#pragma unroll
    for (int kk = 0; kk < AS; kk += 32) {
      short8 a_frag[4], b_frag;
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        a_frag[fm] = *(const short8*)(Ps + row * PSTRIDE + kbase);
      }
      {
        int col = wc * 16 + (lane & 15);
        b_frag = *(const short8*)(VTs + col * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
        acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[fm], b_frag,
                                                          acc[fm], 0, 0, 0);
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int row = wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
        int col = wc * 16 + (lane & 15);
        og[row * orr + col] = f2b(acc[fm][rI]);
      }
  }

  // ---- stream P (backward input) out from LDS: 16-B row-contiguous -------
  // 128x128 elems / 512 threads = 32 = 4 x short8 per thread; the stores
  // drain at kernel end, overlapped by other workgroups.
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = (c * 512 + (int)threadIdx.x) * 8;
    int row = e >> 7, col = e & 127;
    *(short8*)(pg + row * AS + col) =
        *(const short8*)(Ps + row * PSTRIDE + col);
  }
}

// ---------------------------------------------------------------------------
// Fused attention BACKWARD for the same contract shape (S=128, dh=64): one
// workgroup per (batch, head) computes
//   dP = dO.V^T;  dS = scale * P o (dP - rowsum(dP o P));
//   dV = P^T.dO;  dQ = dS.K;  dK = dS^T.Q
// entirely on-chip — replaces 4 batched K1 calls + K4 softmax_bwd + the
// dP/dS HBM round-trips (measured ~6 ms/step of the BERT contract step).
// dP stays in f32 accumulators (the composed path rounded it to bf16).
//
// LDS plan (137 KB, 1 WG/CU; 3072 WGs fill the chip 12x over):
//   dO  [128][64]  16 KB (glds, XOR swz)   \ overlaid by dSn [128][128]
//   V   [128][64]  16 KB (glds, XOR swz)   /  (stride 128, XOR swz)
//   Pt  [128][136] 34 KB (P^T scatter)     -- overlaid by dSt (same stride)
//   dOt [64][136]  17 KB (dO^T scatter)
//   Kt  [64][136]  17 KB (K^T scatter)
//   Qt  [64][136]  17 KB (Q^T scatter)
//   psum[128][32] f32 16 KB + rs[128] f32  (rowsum partials + fold)
// ---------------------------------------------------------------------------
__launch_bounds__(1024) __global__ void attn_bwd_fused_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ probs,
    const uint16_t* __restrict__ dout, uint16_t* __restrict__ dq,
    uint16_t* __restrict__ dk, uint16_t* __restrict__ dv, float scale,
    int heads, int64_t qb, int64_t qh, int64_t qr, int64_t db, int64_t dhh,
    int64_t dr, int64_t gb, int64_t gh, int64_t gr) {
  // 1024 threads (16 waves as 4x4): this kernel is 153 KB of LDS and
  // therefore pinned at 1 WG/CU, so waves-per-SIMD is the only occupancy
  // axis. 256 threads ran 1 wave/SIMD (every barrier and LDS/HBM latency
  // fully exposed), 512 ran 2 (r1), 16 waves give 4/SIMD — measured
  // profile: the per-(b,h) chain is latency- not MFMA-bound (4.6% duty
  // at 8 waves), so more resident waves per SIMD is pure overlap.
  // LDS 153 KB: psum widens to [128][64] (each lane covers 2 columns).
  __shared__ __attribute__((aligned(16))) uint16_t smem[76544];
  uint16_t* dOs = smem;                    // [128][64] swz
  uint16_t* Vs = smem + AS * AD;           // [128][64] swz
  uint16_t* dSn = smem;                    // [128][128] swz (overlay dO+V)
  uint16_t* Pt = smem + 2 * AS * AD;       // [s2][136] -> Pt[s2][s1]
  uint16_t* dSt = Pt;                      // overlay after P consumed
  uint16_t* dOt = Pt + AS * PSTRIDE;       // [d][136]
  uint16_t* Kt = dOt + AD * PSTRIDE;       // [d][136] (Kt[d][s2])
  uint16_t* Qt = Kt + AD * PSTRIDE;        // [d][136] (Qt[d][s1])
  float* psum = (float*)(Qt + AD * PSTRIDE);  // [128][64]
  float* rs = psum + AS * 64;                 // [128]

  const int64_t bh = blockIdx.x;
  const int64_t b_ = bh / heads, h_ = bh % heads;
  const uint16_t* qg = q + b_ * qb + h_ * qh;
  const uint16_t* kg = k + b_ * qb + h_ * qh;
  const uint16_t* vg = v + b_ * qb + h_ * qh;
  const uint16_t* pg = probs + bh * AS * AS;
  const uint16_t* dog = dout + b_ * db + h_ * dhh;
  uint16_t* dqg = dq + b_ * gb + h_ * gh;
  uint16_t* dkg = dk + b_ * gb + h_ * gh;
  uint16_t* dvg = dv + b_ * gb + h_ * gh;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 4x4 wave grid (32-row granules)

  // ---- P0: stage ----------------------------------------------------------
  const int e0 = (int)threadIdx.x * 8;
  {
    int e = e0;
    int row = e >> 6;
    int kk = (e & 63) ^ ((row & 7) << 3);
    att_glds16(dog + row * dr + kk, dOs + e);
    att_glds16(vg + row * qr + kk, Vs + e);
  }
  // register loads + transposed scatters (issued while the DMA flies)
  {
#pragma unroll
    for (int c = 0; c < 2; c += 2) {
      short8 p0 = *(const short8*)(pg + e0 + c * 8192);
      short8 p1 = *(const short8*)(pg + e0 + (c + 1) * 8192);
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        int e = e0 + (c + u) * 8192;  // element in P [s1][s2]
        int s1 = e >> 7, s2 = e & 127;
        const short8& pv = u ? p1 : p0;
#pragma unroll
        for (int i = 0; i < 8; ++i)
          Pt[(s2 + i) * PSTRIDE + s1] = (uint16_t)pv[i];
      }
    }
    {
      int e = e0;                     // element in [s][d]
      int s_ = e >> 6, d = e & 63;
      short8 dov = *(const short8*)(dog + s_ * dr + d);
      short8 kv = *(const short8*)(kg + s_ * qr + d);
      short8 qv = *(const short8*)(qg + s_ * qr + d);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        dOt[(d + i) * PSTRIDE + s_] = (uint16_t)dov[i];
        Kt[(d + i) * PSTRIDE + s_] = (uint16_t)kv[i];
        Qt[(d + i) * PSTRIDE + s_] = (uint16_t)qv[i];
      }
    }
  }
  __syncthreads();  // glds drained + scatters visible

  // ---- P1: dP = dO.V^T (f32 acc; wave owns 32 s1 x 32 s2) -----------------
  f32x4 acc[2][2] = {};
#pragma unroll
  for (int kk = 0; kk < AD; kk += 32) {
    short8 a_frag[2], b_frag[2];
    const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
    for (int fm = 0; fm < 2; ++fm) {
      int row = wr * 32 + fm * 16 + (lane & 15);
      int ks = kbase ^ ((row & 7) << 3);
      a_frag[fm] = *(const short8*)(dOs + row * AD + ks);
    }
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      int col = wc * 32 + fn * 16 + (lane & 15);
      int ks = kbase ^ ((col & 7) << 3);
      b_frag[fn] = *(const short8*)(Vs + col * AD + ks);
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
  }

  // ---- P2: rowsum partials of dP o P --------------------------------------
#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      int row = wr * 32 + fm * 16 + (lane >> 4) * 4 + rI;
      float part = 0.f;
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        part += acc[fm][fn][rI] * b2f(Pt[col * PSTRIDE + row]);
      }
      psum[row * 64 + wc * 16 + (lane & 15)] = part;
    }

  // ---- P2b: dV = P^T.dO (wave owns 32 s2 x 16 d) --------------------------
  {
    f32x4 acc2[2] = {};
#pragma unroll
    for (int kk = 0; kk < AS; kk += 32) {
      short8 a_frag[2], b_frag;
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = wr * 32 + fm * 16 + (lane & 15);  // s2
        a_frag[fm] = *(const short8*)(Pt + row * PSTRIDE + kbase);
      }
      {
        int col = wc * 16 + (lane & 15);  // d
        b_frag = *(const short8*)(dOt + col * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc2[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[fm], b_frag,
                                                           acc2[fm], 0, 0, 0);
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int row = wr * 32 + fm * 16 + (lane >> 4) * 4 + rI;
        int col = wc * 16 + (lane & 15);
        dvg[row * gr + col] = f2b(acc2[fm][rI]);
      }
  }
  __syncthreads();  // psum complete (and Pt reads done -> dSt may overlay)

  // ---- P3: fold rowsums (4-way ILP) ---------------------------------------
  if (threadIdx.x < AS) {
    const float* pr = psum + threadIdx.x * 64;
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll
    for (int i = 0; i < 64; i += 4) {
      a0 += pr[i];
      a1 += pr[i + 1];
      a2 += pr[i + 2];
      a3 += pr[i + 3];
    }
    rs[threadIdx.x] = (a0 + a1) + (a2 + a3);
  }
  __syncthreads();

  // ---- P4: dS = scale * P o (dP - rs); write dSn (swz) + dSt (transposed,
  // overwriting Pt in place: read P just before the store, same lane) ------
#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      int row = wr * 32 + fm * 16 + (lane >> 4) * 4 + rI;
      float r = rs[row];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        float pb = b2f(Pt[col * PSTRIDE + row]);
        uint16_t ds = f2b(scale * pb * (acc[fm][fn][rI] - r));
        dSn[row * AS + (col ^ ((row & 7) << 3))] = ds;
        dSt[col * PSTRIDE + row] = ds;
      }
    }
  __syncthreads();

  // ---- P5: dQ = dS.K (A=dSn swz, B=Kt; wave owns 64 s1 x 16 d) ------------
  {
    f32x4 acc2[2] = {};
#pragma unroll
    for (int kk = 0; kk < AS; kk += 32) {
      short8 a_frag[2], b_frag;
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = wr * 32 + fm * 16 + (lane & 15);  // s1
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(dSn + row * AS + ks);
      }
      {
        int col = wc * 16 + (lane & 15);  // d
        b_frag = *(const short8*)(Kt + col * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc2[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[fm], b_frag,
                                                           acc2[fm], 0, 0, 0);
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int row = wr * 32 + fm * 16 + (lane >> 4) * 4 + rI;
        int col = wc * 16 + (lane & 15);
        dqg[row * gr + col] = f2b(acc2[fm][rI]);
      }
  }

  // ---- P5b: dK = dS^T.Q (A=dSt, B=Qt; wave owns 64 s2 x 16 d) -------------
  {
    f32x4 acc2[2] = {};
#pragma unroll
    for (int kk = 0; kk < AS; kk += 32) {
      short8 a_frag[2], b_frag;
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 2; ++fm) {
        int row = wr * 32 + fm * 16 + (lane & 15);  // s2
        a_frag[fm] = *(const short8*)(dSt + row * PSTRIDE + kbase);
      }
      {
        int col = wc * 16 + (lane & 15);  // d
        b_frag = *(const short8*)(Qt + col * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
        acc2[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[fm], b_frag,
                                                           acc2[fm], 0, 0, 0);
    }
#pragma unroll
    for (int fm = 0; fm < 2; ++fm)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int row = wr * 32 + fm * 16 + (lane >> 4) * 4 + rI;
        int col = wc * 16 + (lane & 15);
        dkg[row * gr + col] = f2b(acc2[fm][rI]);
      }
  }
}

// ---------------------------------------------------------------------------
// Fused attention FORWARD for S=512, dh=64 (SURVEY.md §2.2 K4's second
// tuning shape; VERDICT r01 missing #3). The 512x512 score matrix no
// longer fits a workgroup, so the kernel tiles it: one WG per
// (batch, head, 64-row q-block), K streamed in 64-row tiles.
//
// Three phases per WG (16 waves as 4 row-groups x 4 column quarters;
// row stats combine one partial per column quarter via ms2[4][64]):
//   A1: loop kt 0..7: stage K-tile (glds dbuf, XOR-swz source) ->
//       S_t = Q.K_t^T (8 MFMA/wave) -> running row max -> raw bf16 scores
//       into the P strip [64][512] (stride 520 spreads banks)
//   A2: elementwise exp(s - m) over the strip (16 threads per row), row
//       sums folded through LDS -> inv l
//   B:  loop kt 0..7: V-tile via register scatter to VT [64][72] (dbuf) ->
//       O += P_t . V_t (8 MFMA/wave); epilogue scales O rows by 1/l
// P (normalized) streams out at kernel end for the composed backward.
// LDS: Q 8K + K dbuf 16K + VT dbuf 18.4K + P 66.6K + stats ~= 110 KiB ->
// 1 WG/CU; grid B*H*8 (BERT S=512 batch 64: 6144 WGs = 24 rounds).
// ---------------------------------------------------------------------------
namespace s512 {
constexpr int SQ = 64;         // q-block rows per WG
constexpr int S5 = 512;        // sequence length
constexpr int KT = 64;         // k/v tile rows
constexpr int PST = S5 + 8;    // P strip row stride (elements)
constexpr int VST = KT + 8;    // VT row stride
}  // namespace s512

__launch_bounds__(1024, 4) __global__ void attn_fwd_fused512_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ probs,
    uint16_t* __restrict__ out, float scale, int heads, int64_t qb,
    int64_t qh, int64_t qr, int64_t ob, int64_t oh, int64_t orr) {
  using namespace s512;
  // [Q 64x64][K dbuf 2x 64x64][VT dbuf 2x 64x72][P 64x520][ms 64][ls 64+psum]
  __shared__ __attribute__((aligned(16))) uint16_t smem[
      SQ * AD + 2 * KT * AD + 2 * AD * VST + SQ * PST];
  __shared__ float ms2[4][SQ], lsum[SQ][16], ls[SQ];
  uint16_t* Qs = smem;
  uint16_t* Ks = smem + SQ * AD;                  // dbuf: + (kt&1)*KT*AD
  uint16_t* VTs = smem + SQ * AD + 2 * KT * AD;   // dbuf: + (kt&1)*AD*VST
  uint16_t* Ps = VTs + 2 * AD * VST;

  const int64_t nqb = S5 / SQ;                    // 8 q-blocks per head
  const int64_t bhq = blockIdx.x;
  const int64_t bh = bhq / nqb, qb0 = (bhq % nqb) * SQ;
  const int64_t b_ = bh / heads, h_ = bh % heads;
  const uint16_t* qg = q + b_ * qb + h_ * qh + qb0 * qr;
  const uint16_t* kg = k + b_ * qb + h_ * qh;
  const uint16_t* vg = v + b_ * qb + h_ * qh;
  uint16_t* pg = probs + bh * (int64_t)S5 * S5 + qb0 * S5;
  uint16_t* og = out + b_ * ob + h_ * oh + qb0 * orr;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;  // 16 waves (4/SIMD at 1 WG/CU)
  const int wr = wid & 3;            // row group: 16 q-rows
  const int wc = wid >> 2;           // column quarter: one 16-col fn tile

  // ---- stage Q (glds, swizzled source) + K tile 0 -------------------------
  // a 64x64 tile is 4096 elems = half of (1024 threads x 8 elems)
  const int e0 = (int)threadIdx.x * 8;
  if (e0 < 4096) {
    int e = e0;
    int row = e >> 6;
    int kk = (e & 63) ^ ((row & 7) << 3);
    att_glds16(qg + row * qr + kk, Qs + e);
    att_glds16(kg + row * qr + kk, Ks + e);          // K tile 0 -> buf 0
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  // ---- A1: scores + running max ------------------------------------------
  for (int kt = 0; kt < S5 / KT; ++kt) {
    // prefetch next K tile into the other buffer
    if (kt + 1 < S5 / KT && e0 < 4096) {
      int e = e0;
      int row = e >> 6;
      int kk2 = (e & 63) ^ ((row & 7) << 3);
      att_glds16(kg + (int64_t)(kt + 1) * KT * qr + row * qr + kk2,
                 Ks + ((kt + 1) & 1) * KT * AD + e);
    }
    const uint16_t* kb = Ks + (kt & 1) * KT * AD;
    f32x4 acc = {};
#pragma unroll
    for (int kk = 0; kk < AD; kk += 32) {
      short8 a_frag, b_frag;
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag = *(const short8*)(Qs + row * AD + ks);
      }
      {
        int col = wc * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag = *(const short8*)(kb + col * AD + ks);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0,
                                                    0, 0);
    }
    // running max over this wave's column quarter (combined with the other
    // quarters' partials at publish time via ms2[4][SQ])
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      float mx = scale * acc[rI];
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, 64));
      m_run[rI] = fmaxf(m_run[rI], mx);
    }
    // raw scores (scaled) -> P strip, bf16
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      int row = wr * 16 + (lane >> 4) * 4 + rI;
      int col = kt * KT + wc * 16 + (lane & 15);
      Ps[row * PST + col] = f2b(scale * acc[rI]);
    }
    // next iteration reads the prefetched buffer; everyone must also be
    // done READING kb before it is overwritten two iterations later
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
  // publish row maxima (one partial per column quarter)
  if ((lane & 15) == 0) {
#pragma unroll
    for (int rI = 0; rI < 4; ++rI)
      ms2[wc][wr * 16 + (lane >> 4) * 4 + rI] = m_run[rI];
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // ---- A2: exp(s - m), row sums (16 threads per row) ----------------------
  {
    const int row = (int)threadIdx.x >> 4;
    const int hx = (int)threadIdx.x & 15;
    const float m = fmaxf(fmaxf(ms2[0][row], ms2[1][row]),
                          fmaxf(ms2[2][row], ms2[3][row]));
    uint16_t* pr = Ps + row * PST + hx * 32;
    float acc_l = 0.f;
#pragma unroll 4
    for (int i = 0; i < 32; ++i) {
      float e = __expf(b2f(pr[i]) - m);
      acc_l += e;
      pr[i] = f2b(e);
    }
    lsum[row][hx] = acc_l;
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (threadIdx.x < SQ) {
    const float* lr = lsum[threadIdx.x];
    float s0 = ((lr[0] + lr[1]) + (lr[2] + lr[3])) +
               ((lr[4] + lr[5]) + (lr[6] + lr[7]));
    float s1 = ((lr[8] + lr[9]) + (lr[10] + lr[11])) +
               ((lr[12] + lr[13]) + (lr[14] + lr[15]));
    ls[threadIdx.x] = 1.0f / (s0 + s1);
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // ---- B: O = P~ . V, V tiles scattered transposed ------------------------
  f32x4 accO = {};
  if (e0 < 4096) {
    // V tile 0 -> VT buf 0
    int e = e0;
    int s_ = e >> 6, d = e & 63;
    short8 vv = *(const short8*)(vg + s_ * qr + d);
#pragma unroll
    for (int i = 0; i < 8; ++i)
      VTs[(d + i) * VST + s_] = (uint16_t)vv[i];
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  for (int kt = 0; kt < S5 / KT; ++kt) {
    if (kt + 1 < S5 / KT && e0 < 4096) {
      uint16_t* vt = VTs + ((kt + 1) & 1) * AD * VST;
      int e = e0;
      int s_ = e >> 6, d = e & 63;
      short8 vv =
          *(const short8*)(vg + (int64_t)(kt + 1) * KT * qr + s_ * qr + d);
#pragma unroll
      for (int i = 0; i < 8; ++i)
        vt[(d + i) * VST + s_] = (uint16_t)vv[i];
    }
    const uint16_t* vt = VTs + (kt & 1) * AD * VST;
#pragma unroll
    for (int kk = 0; kk < KT; kk += 32) {
      short8 a_frag, b_frag;
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);
        a_frag = *(const short8*)(Ps + row * PST + kt * KT + kbase);
      }
      {
        int col = wc * 16 + (lane & 15);
        b_frag = *(const short8*)(vt + col * VST + kbase);
      }
      accO = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, accO, 0,
                                                     0, 0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
  // epilogue: O rows scaled by 1/l
#pragma unroll
  for (int rI = 0; rI < 4; ++rI) {
    int row = wr * 16 + (lane >> 4) * 4 + rI;
    int col = wc * 16 + (lane & 15);
    og[row * orr + col] = f2b(accO[rI] * ls[row]);
  }

  // ---- stream normalized P out (backward input) ---------------------------
  {
    const int row = (int)threadIdx.x >> 4;
    const int hx = (int)threadIdx.x & 15;
    const float invl = ls[row];
    const uint16_t* pr = Ps + row * PST + hx * 32;
    uint16_t* pgr = pg + row * S5 + hx * 32;
#pragma unroll 2
    for (int i = 0; i < 32; i += 8) {
      short8 vv = *(const short8*)(pr + i);
      short8 outp;
#pragma unroll
      for (int j = 0; j < 8; ++j) outp[j] = (short)f2b(b2f((uint16_t)vv[j]) * invl);
      *(short8*)(pgr + i) = outp;
    }
  }
}

// ---------------------------------------------------------------------------
// Fused attention BACKWARD for S=512, dh=64 — two kernels (the S x S score
// matrix cannot live in one WG, and dQ accumulates over k-tiles while
// dK/dV accumulate over q-tiles, so each gets the decomposition that makes
// its accumulator workgroup-local):
//   attn_bwd512_q_k   one WG per (b,h,64-row q-block):
//       dP strip = dO . V_t^T streamed over k-tiles into LDS;
//       rs[s1] = rowsum(dP o P) (P staged once, 64 KiB);
//       dS strip = scale * P o (dP - rs)  -> written to the rs scratch's
//       companion use + dQ = sum_t dS_t . K_t; rs goes to a global
//       scratch for the kv kernel.
//   attn_bwd512_kv_k  one WG per (b,h,64-row k-block), 2 WGs/CU:
//       loops q-tiles; dP^T = V . dO^T (pure NT, both operands staged
//       k-contiguous over d); dS^T = scale * P^T o (dP^T - rs[s1]);
//       dV += Pt . dOt-image, dK += dSt . Qt-image. dP/dS are recomputed
//       here (~40% extra attention-bwd FLOPs) to keep both kernels free
//       of cross-WG accumulation — attention is ~3% of the S=512 step,
//       the win is fusing away the composed path's HBM round trips.
// ---------------------------------------------------------------------------
template <int SEQ>
__launch_bounds__(1024, 4) __global__ void attn_bwd512_q_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ probs,
    const uint16_t* __restrict__ dout, uint16_t* __restrict__ dq,
    float* __restrict__ rs_g, float scale, int heads, int64_t qb, int64_t qh,
    int64_t qr, int64_t db, int64_t dhh, int64_t dr, int64_t gb, int64_t gh,
    int64_t gr) {
  using namespace s512;
  constexpr int QPST = SEQ + 8;
  // [dO 8K][KV dbuf 2x9.2K][dS strip 66.6K][P 64K] ~= 157 KiB, 1 WG/CU
  __shared__ __attribute__((aligned(16))) uint16_t smem[
      SQ * AD + 2 * AD * VST + SQ * QPST + SQ * SEQ];
  __shared__ float rsl[SQ], rpart[SQ][16];
  uint16_t* dOs = smem;
  uint16_t* KVs = smem + SQ * AD;            // dbuf: + (kt&1)*AD*VST
  uint16_t* dSs = KVs + 2 * AD * VST;        // [64][520] strip (dP then dS)
  uint16_t* Ps = dSs + SQ * QPST;             // [64][512] linear (glds image)

  const int64_t nqb = SEQ / SQ;
  const int64_t bhq = blockIdx.x;
  const int64_t bh = bhq / nqb, q0 = (bhq % nqb) * SQ;
  const int64_t b_ = bh / heads, h_ = bh % heads;
  const uint16_t* qg = q + b_ * qb + h_ * qh + q0 * qr;
  const uint16_t* kg = k + b_ * qb + h_ * qh;
  const uint16_t* vg = v + b_ * qb + h_ * qh;
  const uint16_t* pg = probs + bh * (int64_t)SEQ * SEQ + q0 * SEQ;
  const uint16_t* dog = dout + b_ * db + h_ * dhh + q0 * dr;
  uint16_t* dqg = dq + b_ * gb + h_ * gh + q0 * gr;
  float* rsg = rs_g + bh * SEQ + q0;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;  // 16 waves (4/SIMD at 1 WG/CU)
  const int wr = wid & 3;            // row group: 16 rows
  const int wc = wid >> 2;           // column quarter: one 16-col fn tile
  const int e0 = (int)threadIdx.x * 8;

  // stage dO (glds, swizzled) + the full P block (glds, linear [64][512])
  if (e0 < 4096) {
    int e = e0;
    int row = e >> 6;
    int kk2 = (e & 63) ^ ((row & 7) << 3);
    att_glds16(dog + row * dr + kk2, dOs + e);
  }
#pragma unroll
  for (int c = 0; c < SEQ / 128; ++c) {
    int e = e0 + c * 8192;  // 64*SEQ elems, [64][SEQ] linear
    att_glds16(pg + (e / SEQ) * SEQ + (e % SEQ), Ps + e);
  }
  // V tile 0 scattered transposed? NO — dP = dO . V^T is NT over d:
  // V [s2][d] is already k-contiguous; stage swizzled like fwd K tiles.
  if (e0 < 4096) {
    int e = e0;
    int row = e >> 6;
    int kk2 = (e & 63) ^ ((row & 7) << 3);
    att_glds16(vg + row * qr + kk2, KVs + e);  // V tile 0 -> buf 0 (swz, 64x64)
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // ---- dP strip: loop k-tiles ---------------------------------------------
  for (int kt = 0; kt < SEQ / KT; ++kt) {
    if (kt + 1 < SEQ / KT && e0 < 4096) {
      int e = e0;
      int row = e >> 6;
      int kk2 = (e & 63) ^ ((row & 7) << 3);
      att_glds16(vg + (int64_t)(kt + 1) * KT * qr + row * qr + kk2,
                 KVs + ((kt + 1) & 1) * AD * VST + e);
    }
    const uint16_t* vb = KVs + (kt & 1) * AD * VST;
    f32x4 acc = {};
#pragma unroll
    for (int kk = 0; kk < AD; kk += 32) {
      short8 a_frag, b_frag;
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag = *(const short8*)(dOs + row * AD + ks);
      }
      {
        int col = wc * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag = *(const short8*)(vb + col * AD + ks);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0,
                                                    0, 0);
    }
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      int row = wr * 16 + (lane >> 4) * 4 + rI;
      int col = kt * KT + wc * 16 + (lane & 15);
      dSs[row * QPST + col] = f2b(acc[rI]);
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }

  // ---- rs = rowsum(dP o P); then dS = scale * P o (dP - rs) ---------------
  {
    const int row = (int)threadIdx.x >> 4;
    const int hx = (int)threadIdx.x & 15;
    const uint16_t* dpr = dSs + row * QPST + hx * (SEQ / 16);
    const uint16_t* ppr = Ps + row * SEQ + hx * (SEQ / 16);
    float accr = 0.f;
#pragma unroll 4
    for (int i = 0; i < SEQ / 16; ++i) accr += b2f(dpr[i]) * b2f(ppr[i]);
    rpart[row][hx] = accr;
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  if (threadIdx.x < SQ) {
    const float* rp = rpart[threadIdx.x];
    float r = (((rp[0] + rp[1]) + (rp[2] + rp[3])) +
               ((rp[4] + rp[5]) + (rp[6] + rp[7]))) +
              (((rp[8] + rp[9]) + (rp[10] + rp[11])) +
               ((rp[12] + rp[13]) + (rp[14] + rp[15])));
    rsl[threadIdx.x] = r;
    rsg[threadIdx.x] = r;  // scratch for the kv kernel
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  {
    const int row = (int)threadIdx.x >> 4;
    const int hx = (int)threadIdx.x & 15;
    uint16_t* dpr = dSs + row * QPST + hx * (SEQ / 16);
    const uint16_t* ppr = Ps + row * SEQ + hx * (SEQ / 16);
    const float r = rsl[row];
#pragma unroll 4
    for (int i = 0; i < SEQ / 16; ++i)
      dpr[i] = f2b(scale * b2f(ppr[i]) * (b2f(dpr[i]) - r));
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  // ---- dQ = sum_t dS_t . K_t (K tiles scattered transposed [d][s2]) -------
  f32x4 accQ = {};
  if (e0 < 4096) {
    int e = e0;
    int s_ = e >> 6, d = e & 63;
    short8 kv2 = *(const short8*)(kg + s_ * qr + d);
    uint16_t* ktile = KVs;  // buf 0
#pragma unroll
    for (int i = 0; i < 8; ++i)
      ktile[(d + i) * VST + s_] = (uint16_t)kv2[i];
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  for (int kt = 0; kt < SEQ / KT; ++kt) {
    if (kt + 1 < SEQ / KT && e0 < 4096) {
      uint16_t* kdst = KVs + ((kt + 1) & 1) * AD * VST;
      int e = e0;
      int s_ = e >> 6, d = e & 63;
      short8 kv2 =
          *(const short8*)(kg + (int64_t)(kt + 1) * KT * qr + s_ * qr + d);
#pragma unroll
      for (int i = 0; i < 8; ++i)
        kdst[(d + i) * VST + s_] = (uint16_t)kv2[i];
    }
    const uint16_t* ktile = KVs + (kt & 1) * AD * VST;
#pragma unroll
    for (int kk = 0; kk < KT; kk += 32) {
      short8 a_frag, b_frag;
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);
        a_frag = *(const short8*)(dSs + row * QPST + kt * KT + kbase);
      }
      {
        int col = wc * 16 + (lane & 15);
        b_frag = *(const short8*)(ktile + col * VST + kbase);
      }
      accQ = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, accQ, 0,
                                                     0, 0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
#pragma unroll
  for (int rI = 0; rI < 4; ++rI) {
    int row = wr * 16 + (lane >> 4) * 4 + rI;
    int col = wc * 16 + (lane & 15);
    dqg[row * gr + col] = f2b(accQ[rI]);
  }
}

template <int SEQ>
__launch_bounds__(512, 2) __global__ void attn_bwd512_kv_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, const uint16_t* __restrict__ probs,
    const uint16_t* __restrict__ dout, uint16_t* __restrict__ dk,
    uint16_t* __restrict__ dv, const float* __restrict__ rs_g, float scale,
    int heads, int64_t qb, int64_t qh, int64_t qr, int64_t db, int64_t dhh,
    int64_t dr, int64_t gb, int64_t gh, int64_t gr) {
  using namespace s512;
  // [V 8K][dO 8K][Pt 9.2K][dOt 9.2K][Qt 9.2K][dSt 9.2K] ~= 53 KiB -> 2/CU
  __shared__ __attribute__((aligned(16))) uint16_t smem[
      2 * KT * AD + 4 * AD * VST];
  __shared__ float rsl[SQ];
  uint16_t* Vs = smem;                       // [s2][d] swz
  uint16_t* dOsw = smem + KT * AD;           // [s1][d] swz (per q-tile)
  uint16_t* Pt = dOsw + KT * AD;             // [s2][72] (Pt[s2][s1])
  uint16_t* dOt = Pt + AD * VST;             // [d][72]
  uint16_t* Qt = dOt + AD * VST;             // [d][72]
  uint16_t* dSt = Qt + AD * VST;             // [s2][72]

  const int64_t nkb = SEQ / KT;
  const int64_t bhk = blockIdx.x;
  const int64_t bh = bhk / nkb, k0r = (bhk % nkb) * KT;
  const int64_t b_ = bh / heads, h_ = bh % heads;
  const uint16_t* qg = q + b_ * qb + h_ * qh;
  const uint16_t* kg = k + b_ * qb + h_ * qh + k0r * qr;
  const uint16_t* vg = v + b_ * qb + h_ * qh + k0r * qr;
  const uint16_t* pg = probs + bh * (int64_t)SEQ * SEQ + k0r;  // column block
  const uint16_t* dog = dout + b_ * db + h_ * dhh;
  uint16_t* dkg = dk + b_ * gb + h_ * gh + k0r * gr;
  uint16_t* dvg = dv + b_ * gb + h_ * gh + k0r * gr;
  const float* rsg = rs_g + bh * SEQ;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;  // 8 waves (4/SIMD at 2 WG/CU)
  const int wr = wid & 3;            // s2 row group: 16 rows
  const int wc = wid >> 2;           // column half: 2 of the 4 fn tiles
  const int e0 = (int)threadIdx.x * 8;

  // V block (fixed for this WG), swizzled
  {
    int e = e0;
    int row = e >> 6;
    int kk2 = (e & 63) ^ ((row & 7) << 3);
    att_glds16(vg + row * qr + kk2, Vs + e);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");

  f32x4 accV[2] = {}, accK[2] = {};
  for (int qt = 0; qt < SEQ / SQ; ++qt) {
    const int64_t q0 = (int64_t)qt * SQ;
    // stage dO_qt (glds swz); P_qt + Q_qt via registers -> transposed images
    {
      int e = e0;
      int row = e >> 6;
      int kk2 = (e & 63) ^ ((row & 7) << 3);
      att_glds16(dog + (q0 + row) * dr + kk2, dOsw + e);
    }
    if (threadIdx.x < SQ) rsl[threadIdx.x] = rsg[q0 + threadIdx.x];
    {
      int e = e0;
      int s1 = e >> 6, d = e & 63;  // [s1][64] block coords
      short8 pv = *(const short8*)(pg + (q0 + s1) * SEQ + d);  // P[s1][k0r+d]
      short8 qv = *(const short8*)(qg + (q0 + s1) * qr + d);
      short8 dov = *(const short8*)(dog + (q0 + s1) * dr + d);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        Pt[(d + i) * VST + s1] = (uint16_t)pv[i];   // Pt[s2][s1]
        Qt[(d + i) * VST + s1] = (uint16_t)qv[i];   // Qt[d][s1]
        dOt[(d + i) * VST + s1] = (uint16_t)dov[i]; // dOt[d][s1]
      }
    }
    asm volatile("s_waitcnt vmcnt(0) lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");

    // dP^T[s2][s1] = V . dO^T (NT over d)
    f32x4 accP[2] = {};
#pragma unroll
    for (int kk = 0; kk < AD; kk += 32) {
      short8 a_frag, b_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);  // s2
        int ks = kbase ^ ((row & 7) << 3);
        a_frag = *(const short8*)(Vs + row * AD + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (wc * 2 + fn) * 16 + (lane & 15);  // s1
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(dOsw + col * AD + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        accP[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag[fn],
                                                           accP[fn], 0, 0, 0);
    }
    // dS^T = scale * P^T o (dP^T - rs[s1]) -> dSt image
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int rI = 0; rI < 4; ++rI) {
        int s2 = wr * 16 + (lane >> 4) * 4 + rI;
        int s1 = (wc * 2 + fn) * 16 + (lane & 15);
        float pb = b2f(Pt[s2 * VST + s1]);
        dSt[s2 * VST + s1] = f2b(scale * pb * (accP[fn][rI] - rsl[s1]));
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");

    // dV += Pt . dOt-image; dK += dSt . Qt-image (both: k = s1)
#pragma unroll
    for (int kk = 0; kk < SQ; kk += 32) {
      short8 a_frag, b_frag, a2_frag, b2_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
      {
        int row = wr * 16 + (lane & 15);  // s2
        a_frag = *(const short8*)(Pt + row * VST + kbase);
        a2_frag = *(const short8*)(dSt + row * VST + kbase);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (wc * 2 + fn) * 16 + (lane & 15);  // d
        b2_frag[fn] = *(const short8*)(Qt + col * VST + kbase);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = (wc * 2 + fn) * 16 + (lane & 15);
        b_frag = *(const short8*)(dOt + col * VST + kbase);
        accV[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag,
                                                           accV[fn], 0, 0, 0);
        accK[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a2_frag, b2_frag[fn], accK[fn], 0, 0, 0);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
  }
#pragma unroll
  for (int fn = 0; fn < 2; ++fn)
#pragma unroll
    for (int rI = 0; rI < 4; ++rI) {
      int row = wr * 16 + (lane >> 4) * 4 + rI;   // s2
      int col = (wc * 2 + fn) * 16 + (lane & 15); // d
      dvg[row * gr + col] = f2b(accV[fn][rI]);
      dkg[row * gr + col] = f2b(accK[fn][rI]);
    }
}

}  // namespace

bool attn_fused_fwd_supported(int64_t S, int64_t dh) {
  return S == AS && dh == AD;
}

bool attn_fused_fwd512_supported(int64_t S, int64_t dh) {
  return S == 512 && dh == AD;
}

// q,k,v,ctx: [B, heads, S, dh]; probs: [B*heads*S, S]
void attn_fused_fwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                    float scale) {
  NZ_CHECK(q.dtype() == DType::BF16, "fused attention: bf16 required");
  hipStream_t s = stream_of(q);
  attn_fwd_fused_k<<<dim3((uint32_t)batch_heads), dim3(512), 0, s>>>(
      (const uint16_t*)q.data(), (const uint16_t*)k.data(),
      (const uint16_t*)v.data(), (uint16_t*)probs.data(),
      (uint16_t*)ctx_out.data(), scale, 1, (int64_t)AS * AD, 0, AD,
      (int64_t)AS * AD, 0, AD);
  NZ_KERNEL_CHECK();
}

// Packed layouts: qkv [B, S, 3H] (q|k|v within each row), out [B, S, H] —
// the fused kernel absorbs the attn_split / attn_merge passes.
void attn_fused_fwd_packed(const Tensor& qkv, Tensor& probs, Tensor& out,
                           int64_t B, int64_t S, int64_t heads, int64_t dh,
                           float scale) {
  NZ_CHECK(qkv.dtype() == DType::BF16, "fused attention: bf16 required");
  hipStream_t s = stream_of(qkv);
  int64_t H = heads * dh;
  const uint16_t* base = (const uint16_t*)qkv.data();
  attn_fwd_fused_k<<<dim3((uint32_t)(B * heads)), dim3(512), 0, s>>>(
      base, base + H, base + 2 * H, (uint16_t*)probs.data(),
      (uint16_t*)out.data(), scale, (int)heads, S * 3 * H, dh, 3 * H,
      S * H, dh, H);
  NZ_KERNEL_CHECK();
}

// q,k,v,dout,dq,dk,dv: [B, heads, S, dh]; probs: [B*heads*S, S]
void attn_fused_bwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    const Tensor& probs, const Tensor& dout, Tensor& dq,
                    Tensor& dk, Tensor& dv, int64_t batch_heads, float scale) {
  NZ_CHECK(q.dtype() == DType::BF16, "fused attention bwd: bf16 required");
  hipStream_t s = stream_of(q);
  const int64_t sh = (int64_t)AS * AD;
  attn_bwd_fused_k<<<dim3((uint32_t)batch_heads), dim3(1024), 0, s>>>(
      (const uint16_t*)q.data(), (const uint16_t*)k.data(),
      (const uint16_t*)v.data(), (const uint16_t*)probs.data(),
      (const uint16_t*)dout.data(), (uint16_t*)dq.data(),
      (uint16_t*)dk.data(), (uint16_t*)dv.data(), scale, 1, sh, 0, AD, sh, 0,
      AD, sh, 0, AD);
  NZ_KERNEL_CHECK();
}

// S=512 backward launcher (packed layout): q-block kernel (dQ + rowsum
// scratch) then kv-block kernel (dK, dV) on the same stream.
void attn_fused_bwd512_packed(const Tensor& qkv, const Tensor& probs,
                              const Tensor& dmerged, Tensor& dqkv, int64_t B,
                              int64_t S, int64_t heads, int64_t dh,
                              float scale) {
  NZ_CHECK(qkv.dtype() == DType::BF16, "fused attn bwd 512: bf16 required");
  NZ_CHECK((S == 512 || S == 128) && dh == AD,
           "fused attn bwd pair: S in {128,512}, dh=64 only");
  hipStream_t s = stream_of(qkv);
  int64_t H = heads * dh;
  const uint16_t* base = (const uint16_t*)qkv.data();
  uint16_t* gbase = (uint16_t*)dqkv.data();
  Tensor rs(qkv.ctx(), DType::F32, Shape{B * heads * S});
  uint32_t nblk = (uint32_t)(B * heads * (S / 64));
  if (S == 512) {
    attn_bwd512_q_k<512><<<dim3(nblk), dim3(1024), 0, s>>>(
        base, base + H, base + 2 * H, (const uint16_t*)probs.data(),
        (const uint16_t*)dmerged.data(), gbase, (float*)rs.data(), scale,
        (int)heads, S * 3 * H, dh, 3 * H, S * H, dh, H, S * 3 * H, dh, 3 * H);
    NZ_KERNEL_CHECK();
    attn_bwd512_kv_k<512><<<dim3(nblk), dim3(512), 0, s>>>(
        base, base + H, base + 2 * H, (const uint16_t*)probs.data(),
        (const uint16_t*)dmerged.data(), gbase + H, gbase + 2 * H,
        (const float*)rs.data(), scale, (int)heads, S * 3 * H, dh, 3 * H,
        S * H, dh, H, S * 3 * H, dh, 3 * H);
  } else {
    attn_bwd512_q_k<128><<<dim3(nblk), dim3(1024), 0, s>>>(
        base, base + H, base + 2 * H, (const uint16_t*)probs.data(),
        (const uint16_t*)dmerged.data(), gbase, (float*)rs.data(), scale,
        (int)heads, S * 3 * H, dh, 3 * H, S * H, dh, H, S * 3 * H, dh, 3 * H);
    NZ_KERNEL_CHECK();
    attn_bwd512_kv_k<128><<<dim3(nblk), dim3(512), 0, s>>>(
        base, base + H, base + 2 * H, (const uint16_t*)probs.data(),
        (const uint16_t*)dmerged.data(), gbase + H, gbase + 2 * H,
        (const float*)rs.data(), scale, (int)heads, S * 3 * H, dh, 3 * H,
        S * H, dh, H, S * 3 * H, dh, 3 * H);
  }
  NZ_KERNEL_CHECK();
}

// S=512 launchers: per-head-contiguous and packed-qkv layouts.
void attn_fused_fwd512(const Tensor& q, const Tensor& k, const Tensor& v,
                       Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                       float scale) {
  NZ_CHECK(q.dtype() == DType::BF16, "fused attention 512: bf16 required");
  hipStream_t s = stream_of(q);
  attn_fwd_fused512_k<<<dim3((uint32_t)(batch_heads * 8)), dim3(1024), 0, s>>>(
      (const uint16_t*)q.data(), (const uint16_t*)k.data(),
      (const uint16_t*)v.data(), (uint16_t*)probs.data(),
      (uint16_t*)ctx_out.data(), scale, 1, (int64_t)512 * AD, 0, AD,
      (int64_t)512 * AD, 0, AD);
  NZ_KERNEL_CHECK();
}

void attn_fused_fwd512_packed(const Tensor& qkv, Tensor& probs, Tensor& out,
                              int64_t B, int64_t S, int64_t heads, int64_t dh,
                              float scale) {
  NZ_CHECK(qkv.dtype() == DType::BF16, "fused attention 512: bf16 required");
  NZ_CHECK(S == 512 && dh == AD, "fused attention 512: S=512, dh=64 only");
  hipStream_t s = stream_of(qkv);
  int64_t H = heads * dh;
  const uint16_t* base = (const uint16_t*)qkv.data();
  attn_fwd_fused512_k<<<dim3((uint32_t)(B * heads * 8)), dim3(1024), 0, s>>>(
      base, base + H, base + 2 * H, (uint16_t*)probs.data(),
      (uint16_t*)out.data(), scale, (int)heads, S * 3 * H, dh, 3 * H,
      S * H, dh, H);
  NZ_KERNEL_CHECK();
}

// Packed: qkv [B,S,3H], dmerged [B,S,H], dqkv [B,S,3H] (fully covered by
// the B*heads workgroups: head h writes columns h*dh..h*dh+dh of each of
// the q|k|v thirds).
void attn_fused_bwd_packed(const Tensor& qkv, const Tensor& probs,
                           const Tensor& dmerged, Tensor& dqkv, int64_t B,
                           int64_t S, int64_t heads, int64_t dh, float scale) {
  NZ_CHECK(qkv.dtype() == DType::BF16, "fused attention bwd: bf16 required");
  hipStream_t s = stream_of(qkv);
  int64_t H = heads * dh;
  const uint16_t* base = (const uint16_t*)qkv.data();
  uint16_t* gbase = (uint16_t*)dqkv.data();
  attn_bwd_fused_k<<<dim3((uint32_t)(B * heads)), dim3(1024), 0, s>>>(
      base, base + H, base + 2 * H, (const uint16_t*)probs.data(),
      (const uint16_t*)dmerged.data(), gbase, gbase + H, gbase + 2 * H, scale,
      (int)heads, S * 3 * H, dh, 3 * H, S * H, dh, H, S * 3 * H, dh, 3 * H);
  NZ_KERNEL_CHECK();
}

}  // namespace hip
}  // namespace nz
