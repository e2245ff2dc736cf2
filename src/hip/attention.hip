// Fused attention forward (K5 upgraded from "composed" to a single-kernel
// fusion for the BERT contract shape S=128, dh=64): one workgroup per
// (batch, head) computes QK^T -> scaled softmax -> P.V entirely on-chip.
// The S x S score tile never touches HBM; P is written once (backward
// consumes it — the backward pass stays composed from K1/K4).
//
// LDS plan (49.5 KB -> 3 blocks/CU):
//   region Q  [128][64]  16 KB  (glds lane-linear, XOR-swizzled)  \ reused as
//   region K  [128][64]  16 KB  (glds)                            / P [128][136]
//   region VT [64][136]  17 KB  (V transposed via register scatter; +8 pad
//                                gives row stride 272 B = 4 banks mod 64 ->
//                                conflict-free column-of-rows reads)
// Waves: 4 (256 threads). QK^T: 2x2 wave grid, 64x64 each (NT, 32 MFMA).
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

__launch_bounds__(256) __global__ void attn_fwd_fused_k(
    const uint16_t* __restrict__ q, const uint16_t* __restrict__ k,
    const uint16_t* __restrict__ v, uint16_t* __restrict__ probs,
    uint16_t* __restrict__ out, float scale) {
  // LDS: [P 128*136 | VT 64*136]; Q and K overlay P's first 32 KB (the P
  // region is 34 KB — VT must live BEYOND it, clobbering it was a bug).
  __shared__ __attribute__((aligned(16))) uint16_t smem[(AS + AD) * PSTRIDE];
  uint16_t* Qs = smem;
  uint16_t* Ks = smem + AS * AD;
  uint16_t* Ps = smem;                      // overlays Q..K after QK^T
  uint16_t* VTs = smem + AS * PSTRIDE;

  const int64_t bh = blockIdx.x;            // batch*heads index
  const uint16_t* qg = q + bh * AS * AD;
  const uint16_t* kg = k + bh * AS * AD;
  const uint16_t* vg = v + bh * AS * AD;
  uint16_t* pg = probs + bh * AS * AS;
  uint16_t* og = out + bh * AS * AD;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;

  // ---- stage Q,K by glds (swizzled source), V to registers ---------------
  const int e0 = (int)threadIdx.x * 8;      // 2 chunks x 8 elems per thread
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = e0 + c * 2048;
    int row = e >> 6;
    int kk = (e & 63) ^ ((row & 7) << 3);
    att_glds16(qg + row * AD + kk, Qs + e);
    att_glds16(kg + row * AD + kk, Ks + e);
  }
  short8 vreg[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = e0 + c * 2048;                  // element in [s][d]
    vreg[c] = *(const short8*)(vg + e);
  }
  // scatter V^T while the DMA flies: (s,d..d+7) -> VT[d+i][s]
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int e = e0 + c * 2048;
    int s = e >> 6, d = e & 63;
#pragma unroll
    for (int i = 0; i < 8; ++i)
      VTs[(d + i) * PSTRIDE + s] = (uint16_t)vreg[c][i];
  }
  __syncthreads();  // drains glds (vmcnt0) + V scatter visible

  // ---- QK^T: wave (wr,wc) of 2x2 computes rows wr*64.., cols wc*64.. -----
  {
    const int wr = wid >> 1, wc = wid & 1;
    f32x4 acc[4][4] = {};
#pragma unroll
    for (int kk = 0; kk < AD; kk += 32) {
      short8 a_frag[4], b_frag[4];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        int ks = kbase ^ ((row & 7) << 3);
        a_frag[fm] = *(const short8*)(Qs + row * AD + ks);
      }
#pragma unroll
      for (int fn = 0; fn < 4; ++fn) {
        int col = wc * 64 + fn * 16 + (lane & 15);
        int ks = kbase ^ ((col & 7) << 3);
        b_frag[fn] = *(const short8*)(Ks + col * AD + ks);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 4; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();  // everyone done READING Qs/Ks before P overlays them
    // write scores (bf16) into the P overlay
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 4; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int row = wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
          int col = wc * 64 + fn * 16 + (lane & 15);
          Ps[row * PSTRIDE + col] = f2b(acc[fm][fn][rI]);
        }
  }
  __syncthreads();

  // ---- softmax: wave w handles rows w, w+4, ... (2 elems per lane) -------
  for (int row = wid; row < AS; row += 4) {
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
    // global P (backward input), coalesced ushort2
    *(ushort2*)(pg + row * AS + lane * 2) = make_ushort2(o0, o1);
  }
  __syncthreads();

  // ---- P.V: output 128x64; wave (wr,wc) of 2x2 computes 64x32 ------------
  {
    const int wr = wid >> 1, wc = wid & 1;
    f32x4 acc[4][2] = {};
#pragma unroll
    for (int kk = 0; kk < AS; kk += 32) {
      short8 a_frag[4], b_frag[2];
      const int kbase = kk + (lane >> 4) * 8;
#pragma unroll
      for (int fm = 0; fm < 4; ++fm) {
        int row = wr * 64 + fm * 16 + (lane & 15);
        a_frag[fm] = *(const short8*)(Ps + row * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fn = 0; fn < 2; ++fn) {
        int col = wc * 32 + fn * 16 + (lane & 15);
        b_frag[fn] = *(const short8*)(VTs + col * PSTRIDE + kbase);
      }
#pragma unroll
      for (int fm = 0; fm < 4; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[fm], b_frag[fn], acc[fm][fn], 0, 0, 0);
    }
#pragma unroll
    for (int fm = 0; fm < 4; ++fm)
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
#pragma unroll
        for (int rI = 0; rI < 4; ++rI) {
          int row = wr * 64 + fm * 16 + (lane >> 4) * 4 + rI;
          int col = wc * 32 + fn * 16 + (lane & 15);
          og[row * AD + col] = f2b(acc[fm][fn][rI]);
        }
  }
}

}  // namespace

bool attn_fused_fwd_supported(int64_t S, int64_t dh) {
  return S == AS && dh == AD;
}

// q,k,v,ctx: [B, heads, S, dh]; probs: [B*heads*S, S]
void attn_fused_fwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                    float scale) {
  NZ_CHECK(q.dtype() == DType::BF16, "fused attention: bf16 required");
  hipStream_t s = stream_of(q);
  attn_fwd_fused_k<<<dim3((uint32_t)batch_heads), dim3(256), 0, s>>>(
      (const uint16_t*)q.data(), (const uint16_t*)k.data(),
      (const uint16_t*)v.data(), (uint16_t*)probs.data(),
      (uint16_t*)ctx_out.data(), scale);
  NZ_KERNEL_CHECK();
}

}  // namespace hip
}  // namespace nz
