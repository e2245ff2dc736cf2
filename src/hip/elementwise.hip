// CDNA4 kernel set, part 1: K2 (bias+GELU), K3 (LayerNorm), K4 (softmax),
// K6/K7 (Adam/SGD), K8 (elementwise family), K9 (losses), K10 (reductions),
// attention layout ops. Design notes per /opt/skills/guides/cdna_hip_programming.md:
// 64-wide waves (no 32-lane warp idioms), bf16 vectorized as ushort4/short8
// where hot (Guideline 13), two-stage deterministic reductions (no atomics
// on the loss scalar), block-per-row norm/softmax with wave shuffles.
#include <algorithm>

#include "src/hip/hip_common.h"
#include "src/hip/hip_ops.h"

namespace nz {
namespace hip {

using ushort8_t = __attribute__((ext_vector_type(8))) unsigned short;

// ------------------------------------------------------------------ fill ---
template <typename T>
__global__ void fill_k(T* p, int64_t n, float v) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) st(p, i, v);
}

void fill(Tensor& t, float value) {
  hipStream_t s = stream_of(t);
  int64_t n = t.numel();
  dim3 g(grid_1d(n, 256)), b(256);
  if (t.dtype() == DType::F32)
    fill_k<float><<<g, b, 0, s>>>((float*)t.data(), n, value);
  else
    fill_k<uint16_t><<<g, b, 0, s>>>((uint16_t*)t.data(), n, value);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------------------- rng ---
template <typename T, bool NORMAL>
__global__ void rand_k(T* p, int64_t n, uint64_t seed, float a, float b) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float v = NORMAL ? (a + b * rng_normal(seed, i))
                     : (a + (b - a) * rng_uniform01(seed, i));
    st(p, i, v);
  }
}

void rand_uniform(Tensor& t, uint64_t seed, float lo, float hi) {
  hipStream_t s = stream_of(t);
  int64_t n = t.numel();
  dim3 g(grid_1d(n, 256)), b(256);
  if (t.dtype() == DType::F32)
    rand_k<float, false><<<g, b, 0, s>>>((float*)t.data(), n, seed, lo, hi);
  else
    rand_k<uint16_t, false><<<g, b, 0, s>>>((uint16_t*)t.data(), n, seed, lo, hi);
  NZ_KERNEL_CHECK();
}
void rand_normal(Tensor& t, uint64_t seed, float mean, float std) {
  hipStream_t s = stream_of(t);
  int64_t n = t.numel();
  dim3 g(grid_1d(n, 256)), b(256);
  if (t.dtype() == DType::F32)
    rand_k<float, true><<<g, b, 0, s>>>((float*)t.data(), n, seed, mean, std);
  else
    rand_k<uint16_t, true><<<g, b, 0, s>>>((uint16_t*)t.data(), n, seed, mean, std);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------------- cast/copy ---
__global__ void cast_f2b_k(uint16_t* d, const float* src, int64_t n) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] = f2b(src[i]);
}
__global__ void cast_b2f_k(float* d, const uint16_t* src, int64_t n) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] = b2f(src[i]);
}

void cast_copy(Tensor& dst, const Tensor& src) {
  NZ_CHECK(dst.numel() == src.numel(), "cast_copy numel mismatch");
  hipStream_t s = stream_of(dst);
  int64_t n = src.numel();
  if (dst.dtype() == src.dtype()) {
    NZ_HIP_CHECK_K(hipMemcpyAsync(dst.data(), src.data(), src.bytes(),
                                  hipMemcpyDeviceToDevice, s));
    return;
  }
  dim3 g(grid_1d(n, 256)), b(256);
  if (src.dtype() == DType::F32)
    cast_f2b_k<<<g, b, 0, s>>>((uint16_t*)dst.data(), (const float*)src.data(), n);
  else
    cast_b2f_k<<<g, b, 0, s>>>((float*)dst.data(), (const uint16_t*)src.data(), n);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------------- add/scale ---
// bf16 += bf16 vectorized: 4 bf16 per thread (ushort4 = 8 B/lane).
__global__ void add_b16_k(uint16_t* d, const uint16_t* src, int64_t n, float a) {
  int64_t i4 = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (; i4 + 3 < n; i4 += stride) {
    ushort4 dv = *(const ushort4*)(d + i4);
    ushort4 sv = *(const ushort4*)(src + i4);
    dv.x = f2b(b2f(dv.x) + a * b2f(sv.x));
    dv.y = f2b(b2f(dv.y) + a * b2f(sv.y));
    dv.z = f2b(b2f(dv.z) + a * b2f(sv.z));
    dv.w = f2b(b2f(dv.w) + a * b2f(sv.w));
    *(ushort4*)(d + i4) = dv;
  }
  // tail (n % 4 elements), done once by the single global thread 0
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n & ~int64_t(3); i < n; ++i)
      d[i] = f2b(b2f(d[i]) + a * b2f(src[i]));
}
__global__ void add_f32_k(float* d, const float* src, int64_t n, float a) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] += a * src[i];
}
__global__ void add_mixed_k(uint16_t* d, const float* src, int64_t n, float a) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) d[i] = f2b(b2f(d[i]) + a * src[i]);
}

void add_inplace(Tensor& dst, const Tensor& src, float alpha) {
  NZ_CHECK(dst.numel() == src.numel(), "add numel mismatch");
  hipStream_t s = stream_of(dst);
  int64_t n = dst.numel();
  dim3 b(256);
  if (dst.dtype() == DType::BF16 && src.dtype() == DType::BF16) {
    dim3 g(grid_1d((n + 3) / 4, 256));
    add_b16_k<<<g, b, 0, s>>>((uint16_t*)dst.data(), (const uint16_t*)src.data(), n, alpha);
  } else if (dst.dtype() == DType::F32 && src.dtype() == DType::F32) {
    add_f32_k<<<dim3(grid_1d(n, 256)), b, 0, s>>>((float*)dst.data(),
                                                  (const float*)src.data(), n, alpha);
  } else if (dst.dtype() == DType::BF16 && src.dtype() == DType::F32) {
    add_mixed_k<<<dim3(grid_1d(n, 256)), b, 0, s>>>((uint16_t*)dst.data(),
                                                    (const float*)src.data(), n, alpha);
  } else {
    NZ_CHECK(false, "unsupported add dtype combo");
  }
  NZ_KERNEL_CHECK();
}

__global__ void add_out_b16_k(uint16_t* o, const uint16_t* x,
                              const uint16_t* y, int64_t n) {
  int64_t i4 = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 4;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (; i4 + 3 < n; i4 += stride) {
    ushort4 xv = *(const ushort4*)(x + i4);
    ushort4 yv = *(const ushort4*)(y + i4);
    ushort4 ov;
    ov.x = f2b(b2f(xv.x) + b2f(yv.x));
    ov.y = f2b(b2f(xv.y) + b2f(yv.y));
    ov.z = f2b(b2f(xv.z) + b2f(yv.z));
    ov.w = f2b(b2f(xv.w) + b2f(yv.w));
    *(ushort4*)(o + i4) = ov;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n & ~int64_t(3); i < n; ++i)
      o[i] = f2b(b2f(x[i]) + b2f(y[i]));
}
template <typename T>
__global__ void add_out_k(T* o, const T* x, const T* y, int64_t n) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) st(o, i, ld(x, i) + ld(y, i));
}

// out = x + y (residual forward: the copy+add pair cost an extra full pass)
void add_out(Tensor& out, const Tensor& x, const Tensor& y) {
  NZ_CHECK(out.numel() == x.numel() && x.numel() == y.numel(),
           "add_out numel mismatch");
  NZ_CHECK(out.dtype() == x.dtype() && x.dtype() == y.dtype(),
           "add_out dtype mismatch");
  hipStream_t s = stream_of(out);
  int64_t n = out.numel();
  dim3 b(256);
  if (out.dtype() == DType::BF16) {
    add_out_b16_k<<<dim3(grid_1d((n + 3) / 4, 256)), b, 0, s>>>(
        (uint16_t*)out.data(), (const uint16_t*)x.data(),
        (const uint16_t*)y.data(), n);
  } else {
    add_out_k<float><<<dim3(grid_1d(n, 256)), b, 0, s>>>(
        (float*)out.data(), (const float*)x.data(), (const float*)y.data(), n);
  }
  NZ_KERNEL_CHECK();
}

template <typename T>
__global__ void scale_k(T* d, int64_t n, float a) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) st(d, i, ld(d, i) * a);
}
void scale_inplace(Tensor& dst, float alpha) {
  hipStream_t s = stream_of(dst);
  int64_t n = dst.numel();
  dim3 g(grid_1d(n, 256)), b(256);
  if (dst.dtype() == DType::F32)
    scale_k<float><<<g, b, 0, s>>>((float*)dst.data(), n, alpha);
  else
    scale_k<uint16_t><<<g, b, 0, s>>>((uint16_t*)dst.data(), n, alpha);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------------- transpose ---
// Fast path (shapes % 64): 64x64 LDS tile, 16-B vector global loads AND
// stores (Guideline 13 — scalar bf16 is 2-2.5x slower); rows padded one
// vector to break bank conflicts on the gather pass. Also runs batched
// (grid.z) to serve the attention/PV lowering.
template <typename T, int VEC>
__global__ void transpose_vec_k(const T* __restrict__ src, T* __restrict__ dst,
                                int64_t R, int64_t C, int64_t sSrc, int64_t sDst) {
  using vec_t = __attribute__((ext_vector_type(VEC))) T;
  static_assert(sizeof(T) * VEC == 16, "vec width must be 16 B");
  __shared__ T tile[64][64 + VEC];
  const T* sp = src + blockIdx.z * sSrc;
  T* dp = dst + blockIdx.z * sDst;
  int64_t c0 = (int64_t)blockIdx.x * 64, r0 = (int64_t)blockIdx.y * 64;
  constexpr int CHUNKS_PER_ROW = 64 / VEC;
  // column-group XOR swizzle: without it the gather below reads 8 rows a
  // VEC-group apart whose addresses land on ONE bank (row stride x 8 is
  // 0 mod 32 dwords for any 16B-aligned stride) — an 8-way conflict on
  // every gathered vector. Permuting the stored group per row spreads the
  // gather to ~2-way while keeping the stores 16 B vectors.
#pragma unroll
  for (int rep = 0; rep < 64 * CHUNKS_PER_ROW / 256; ++rep) {
    int cidx = rep * 256 + threadIdx.x;
    int row = cidx / CHUNKS_PER_ROW, cg = cidx % CHUNKS_PER_ROW;
    *(vec_t*)&tile[row][(cg ^ (row & (CHUNKS_PER_ROW - 1))) * VEC] =
        *(const vec_t*)(sp + (r0 + row) * C + c0 + cg * VEC);
  }
  __syncthreads();
#pragma unroll
  for (int rep = 0; rep < 64 * CHUNKS_PER_ROW / 256; ++rep) {
    int cidx = rep * 256 + threadIdx.x;
    int orow = cidx / CHUNKS_PER_ROW, og = cidx % CHUNKS_PER_ROW;
    const int lg = orow / VEC, off = orow % VEC;
    T vals[VEC];
#pragma unroll
    for (int i = 0; i < VEC; ++i) {
      int r = og * VEC + i;
      vals[i] = tile[r][(lg ^ (r & (CHUNKS_PER_ROW - 1))) * VEC + off];
    }
    *(vec_t*)(dp + (c0 + orow) * R + r0 + og * VEC) = *(vec_t*)vals;
  }
}

// generic fallback for ragged shapes (scalar, guarded)
template <typename T>
__global__ void transpose_k(const T* __restrict__ src, T* __restrict__ dst,
                            int64_t R, int64_t C) {
  __shared__ T tile[64][65];
  int64_t c0 = blockIdx.x * 64, r0 = blockIdx.y * 64;
  int tx = threadIdx.x & 63, ty = threadIdx.x >> 6;  // 256 thr: 64 x 4
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    int64_t r = r0 + ty + i * 4, c = c0 + tx;
    if (r < R && c < C) tile[ty + i * 4][tx] = src[r * C + c];
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 16; ++i) {
    int64_t r = c0 + ty + i * 4, c = r0 + tx;  // transposed coords
    if (r < C && c < R) dst[r * R + c] = tile[tx][ty + i * 4];
  }
}

void transpose2d_batched(const Tensor& src, Tensor& dst, int64_t batch,
                         int64_t R, int64_t C, int64_t sSrc, int64_t sDst) {
  hipStream_t s = stream_of(src);
  NZ_CHECK(R % 64 == 0 && C % 64 == 0, "batched transpose needs %64 shapes");
  dim3 g(C / 64, R / 64, batch), b(256);
  if (src.dtype() == DType::F32)
    transpose_vec_k<float, 4><<<g, b, 0, s>>>((const float*)src.data(),
                                              (float*)dst.data(), R, C, sSrc, sDst);
  else
    transpose_vec_k<uint16_t, 8><<<g, b, 0, s>>>((const uint16_t*)src.data(),
                                                 (uint16_t*)dst.data(), R, C,
                                                 sSrc, sDst);
  NZ_KERNEL_CHECK();
}

void transpose2d(const Tensor& src, Tensor& dst) {
  int64_t R = src.rows(), C = src.cols();
  NZ_CHECK(dst.rows() == C && dst.cols() == R, "transpose shape mismatch");
  if (R % 64 == 0 && C % 64 == 0) {
    transpose2d_batched(src, dst, 1, R, C, 0, 0);
    return;
  }
  hipStream_t s = stream_of(src);
  dim3 g((C + 63) / 64, (R + 63) / 64), b(256);
  if (src.dtype() == DType::F32)
    transpose_k<float><<<g, b, 0, s>>>((const float*)src.data(), (float*)dst.data(), R, C);
  else
    transpose_k<uint16_t><<<g, b, 0, s>>>((const uint16_t*)src.data(),
                                          (uint16_t*)dst.data(), R, C);
  NZ_KERNEL_CHECK();
}

// ---------------------------------------------------------------- colsum ---
// dst[j] = beta*dst[j] + sum_m src[m,j]. Two-stage: stage 1 splits the rows
// into CHUNK groups so the grid fills all 256 CUs (a single thread-per-
// column pass left 15/16 of the chip idle and ran 160x slower), each block
// accumulating its 256 columns over its row range (coalesced 512-B row
// sweeps); stage 2 folds the f32 partials.
template <typename T>
__global__ void colsum_stage1_k(const T* __restrict__ src,
                                float* __restrict__ partials, int64_t M,
                                int64_t N, int64_t rows_per_chunk) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (j >= N) return;
  int64_t r0 = blockIdx.y * rows_per_chunk;
  int64_t r1 = min(M, r0 + rows_per_chunk);
  float acc = 0.f;
  for (int64_t i = r0; i < r1; ++i) acc += ld(src, i * N + j);
  partials[blockIdx.y * N + j] = acc;
}

// bf16 fast path. One 256-thread WG covers 128 columns (16 ushort8 lanes)
// x 16 row-subchunks: thread (jj, sub) sweeps rows sub, sub+16, ... of the
// chunk, then a fixed-order LDS tree folds the 16 sub-partials. The earlier
// one-thread-per-column-strip version was LATENCY bound on the narrow BERT
// bias grads (N=2304 -> 288 active lanes per chunk, one dependent load in
// flight, 1.5 TB/s measured); this layout gives 8x the wave count and
// fully-coalesced 256 B quarter-wave segments. Deterministic: serial row
// order and a fixed 16-way fold order per column.
__global__ void colsum_stage1_vec_k(const ushort8_t* __restrict__ src8,
                                    float* __restrict__ partials, int64_t M,
                                    int64_t N8, int64_t rows_per_chunk) {
  __shared__ float sm[16 * 128];
  const int jj = threadIdx.x & 15;         // ushort8 column within the block
  const int sub = threadIdx.x >> 4;        // row subchunk 0..15
  const int64_t j8 = (int64_t)blockIdx.x * 16 + jj;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(M, r0 + rows_per_chunk);
  float acc[8] = {};
  if (j8 < N8) {
    for (int64_t i = r0 + sub; i < r1; i += 16) {
      ushort8_t v = src8[i * N8 + j8];
#pragma unroll
      for (int u = 0; u < 8; ++u) acc[u] += b2f(((const uint16_t*)&v)[u]);
    }
  }
#pragma unroll
  for (int u = 0; u < 8; ++u) sm[sub * 128 + jj * 8 + u] = acc[u];
  __syncthreads();
  if (threadIdx.x < 128) {
    const int col = threadIdx.x;
    float v = 0.f;
#pragma unroll
    for (int s_ = 0; s_ < 16; ++s_) v += sm[s_ * 128 + col];
    int64_t gcol = (int64_t)blockIdx.x * 128 + col;
    if (gcol < N8 * 8) partials[blockIdx.y * N8 * 8 + gcol] = v;
  }
}
template <typename TD>
__global__ void colsum_stage2_k(const float* __restrict__ partials,
                                TD* __restrict__ dst, int64_t N, int chunks,
                                float beta) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (j >= N) return;
  // 4 independent accumulators: the fold is latency-bound at high chunk
  // counts (few waves, serial dependent adds); fixed order -> deterministic
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  int c = 0;
  for (; c + 4 <= chunks; c += 4) {
    a0 += partials[(int64_t)c * N + j];
    a1 += partials[(int64_t)(c + 1) * N + j];
    a2 += partials[(int64_t)(c + 2) * N + j];
    a3 += partials[(int64_t)(c + 3) * N + j];
  }
  for (; c < chunks; ++c) a0 += partials[(int64_t)c * N + j];
  float acc = (a0 + a1) + (a2 + a3);
  float prev = beta != 0.f ? beta * ld(dst, j) : 0.f;
  st(dst, j, prev + acc);
}
void colsum(const Tensor& src, Tensor& dst, float beta) {
  int64_t M = src.rows(), N = src.cols();
  NZ_CHECK(dst.numel() == N, "colsum dst size");
  hipStream_t s = stream_of(src);
  // enough chunks that colblocks*chunks fills the chip, but capped so the
  // stage-2 fold stays short (BERT: 512 chunks made stage2 the hot kernel)
  int64_t colblocks = (N + 255) / 256;
  int chunks = (int)std::min<int64_t>(
      (M + 63) / 64,
      std::min<int64_t>(64, std::max<int64_t>(1, 512 / colblocks)));
  int64_t rows_per_chunk = (M + chunks - 1) / chunks;
  if (src.dtype() == DType::BF16 && N % 8 == 0) {
    // vectorized: WG covers 128 columns x 16 row-subchunks; pick the chunk
    // count so jblocks*chunks ~ 1024 WGs (4 waves/SIMD)
    int64_t cb8 = (N + 127) / 128;
    int ch8 = (int)std::min<int64_t>(
        (M + 15) / 16,
        std::min<int64_t>(256, std::max<int64_t>(1, 1024 / cb8)));
    int64_t rpc8 = (M + ch8 - 1) / ch8;
    Tensor partials8(src.ctx(), DType::F32, Shape{(int64_t)ch8, N});
    colsum_stage1_vec_k<<<dim3(cb8, ch8), dim3(256), 0, s>>>(
        (const ushort8_t*)src.data(), (float*)partials8.data(), M, N / 8, rpc8);
    dim3 g2v((N + 255) / 256);
    if (dst.dtype() == DType::BF16)
      colsum_stage2_k<uint16_t><<<g2v, dim3(256), 0, s>>>(
          (const float*)partials8.data(), (uint16_t*)dst.data(), N, ch8, beta);
    else
      colsum_stage2_k<float><<<g2v, dim3(256), 0, s>>>(
          (const float*)partials8.data(), (float*)dst.data(), N, ch8, beta);
    NZ_KERNEL_CHECK();
    return;
  }
  Tensor partials(src.ctx(), DType::F32, Shape{(int64_t)chunks, N});
  dim3 g1(colblocks, chunks), b(256);
  if (src.dtype() == DType::BF16)
    colsum_stage1_k<uint16_t><<<g1, b, 0, s>>>((const uint16_t*)src.data(),
                                               (float*)partials.data(), M, N,
                                               rows_per_chunk);
  else
    colsum_stage1_k<float><<<g1, b, 0, s>>>((const float*)src.data(),
                                            (float*)partials.data(), M, N,
                                            rows_per_chunk);
  dim3 g2(colblocks);
  if (dst.dtype() == DType::BF16)
    colsum_stage2_k<uint16_t><<<g2, b, 0, s>>>((const float*)partials.data(),
                                               (uint16_t*)dst.data(), N, chunks,
                                               beta);
  else
    colsum_stage2_k<float><<<g2, b, 0, s>>>((const float*)partials.data(),
                                            (float*)dst.data(), N, chunks, beta);
  NZ_KERNEL_CHECK();
}

// --------------------------------------------------------- K2: bias+GELU ---
template <typename T>
__global__ void bias_gelu_fwd_k(const T* __restrict__ x, const T* __restrict__ bias,
                                T* __restrict__ y, T* __restrict__ z, int64_t n,
                                int64_t N) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float v = ld(x, i) + (bias ? ld(bias, i % N) : 0.f);
    if (z) st(z, i, v);
    st(y, i, gelu_fwd_scalar(v));
  }
}
void bias_gelu_fwd(const Tensor& x, const Tensor* bias, Tensor& y, Tensor* save_pre) {
  hipStream_t s = stream_of(x);
  int64_t n = x.numel(), N = x.cols();
  dim3 g(grid_1d(n, 256)), b(256);
  if (x.dtype() == DType::BF16)
    bias_gelu_fwd_k<uint16_t><<<g, b, 0, s>>>(
        (const uint16_t*)x.data(), bias ? (const uint16_t*)bias->data() : nullptr,
        (uint16_t*)y.data(), save_pre ? (uint16_t*)save_pre->data() : nullptr, n, N);
  else
    bias_gelu_fwd_k<float><<<g, b, 0, s>>>(
        (const float*)x.data(), bias ? (const float*)bias->data() : nullptr,
        (float*)y.data(), save_pre ? (float*)save_pre->data() : nullptr, n, N);
  NZ_KERNEL_CHECK();
}

// dz = dy * gelu'(z), vectorized 4-wide for bf16.
__global__ void gelu_bwd_b16_k(const uint16_t* __restrict__ dy,
                               const uint16_t* __restrict__ z,
                               uint16_t* __restrict__ dz, int64_t n4) {
  // two independent 8-byte vectors per iteration: load ILP + 16 B stores
  // (the single-ushort4 version measured ~60% of the traffic bound)
  int64_t i8 = (blockIdx.x * (int64_t)blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; i8 + 7 < n4; i8 += stride) {
    ushort4 d0 = *(const ushort4*)(dy + i8);
    ushort4 d1 = *(const ushort4*)(dy + i8 + 4);
    ushort4 z0 = *(const ushort4*)(z + i8);
    ushort4 z1 = *(const ushort4*)(z + i8 + 4);
    ushort4 o0, o1;
    o0.x = f2b(b2f(d0.x) * gelu_bwd_scalar(b2f(z0.x)));
    o0.y = f2b(b2f(d0.y) * gelu_bwd_scalar(b2f(z0.y)));
    o0.z = f2b(b2f(d0.z) * gelu_bwd_scalar(b2f(z0.z)));
    o0.w = f2b(b2f(d0.w) * gelu_bwd_scalar(b2f(z0.w)));
    o1.x = f2b(b2f(d1.x) * gelu_bwd_scalar(b2f(z1.x)));
    o1.y = f2b(b2f(d1.y) * gelu_bwd_scalar(b2f(z1.y)));
    o1.z = f2b(b2f(d1.z) * gelu_bwd_scalar(b2f(z1.z)));
    o1.w = f2b(b2f(d1.w) * gelu_bwd_scalar(b2f(z1.w)));
    *(ushort4*)(dz + i8) = o0;
    *(ushort4*)(dz + i8 + 4) = o1;
  }
  if (blockIdx.x == 0 && threadIdx.x == 0)
    for (int64_t i = n4 & ~int64_t(7); i < n4; ++i)
      dz[i] = f2b(b2f(dy[i]) * gelu_bwd_scalar(b2f(z[i])));
}
template <typename T>
__global__ void gelu_bwd_k(const T* __restrict__ dy, const T* __restrict__ z,
                           T* __restrict__ dz, int64_t n) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) st(dz, i, ld(dy, i) * gelu_bwd_scalar(ld(z, i)));
}
void bias_gelu_bwd(const Tensor& dy, const Tensor& z, Tensor& dz, Tensor* dbias) {
  hipStream_t s = stream_of(dy);
  int64_t n = dy.numel();
  dim3 b(256);
  if (dy.dtype() == DType::BF16 && (n & 3) == 0) {
    gelu_bwd_b16_k<<<dim3(grid_1d((n + 7) / 8, 256)), b, 0, s>>>(
        (const uint16_t*)dy.data(), (const uint16_t*)z.data(),
        (uint16_t*)dz.data(), n);
  } else if (dy.dtype() == DType::BF16) {
    gelu_bwd_k<uint16_t><<<dim3(grid_1d(n, 256)), b, 0, s>>>(
        (const uint16_t*)dy.data(), (const uint16_t*)z.data(), (uint16_t*)dz.data(), n);
  } else {
    gelu_bwd_k<float><<<dim3(grid_1d(n, 256)), b, 0, s>>>(
        (const float*)dy.data(), (const float*)z.data(), (float*)dz.data(), n);
  }
  NZ_KERNEL_CHECK();
  if (dbias) colsum(dz, *dbias, 0.f);
}

// --------------------------------------------------------- K3: LayerNorm ---
// One 256-thread block per row; f32 accumulation; mean/rstd cached for bwd.
template <typename T>
__global__ void layernorm_fwd_k(const T* __restrict__ x, const T* __restrict__ g,
                                const T* __restrict__ be, T* __restrict__ y,
                                float* __restrict__ mean, float* __restrict__ rstd,
                                int64_t R, int64_t N, float eps) {
  __shared__ float lds4[4];
  for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
    const T* xr = x + row * N;
    float s = 0.f;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x) s += ld(xr, j);
    float mu = block_sum<256>(s, lds4) / N;
    __syncthreads();
    float v = 0.f;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x) {
      float d = ld(xr, j) - mu;
      v += d * d;
    }
    float rs = rsqrtf(block_sum<256>(v, lds4) / N + eps);
    if (threadIdx.x == 0) { mean[row] = mu; rstd[row] = rs; }
    T* yr = y + row * N;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      st(yr, j, (ld(xr, j) - mu) * rs * ld(g, j) + ld(be, j));
    __syncthreads();
  }
}
// Wave-per-row LayerNorm for bf16 rows that fit in registers (BERT N=768:
// 12 elems/lane). The block-per-row version re-reads x three times with
// scalar 2-byte loads and runs ~5x off the traffic bound; here the row is
// loaded once (vectorized), mean/var come from two wave shuffles, and y is
// written straight from registers.
template <int NE>  // elements per lane; N == 64*NE, NE % 4 == 0
__global__ void layernorm_fwd_wave_b16_k(
    const uint16_t* __restrict__ x, const uint16_t* __restrict__ g,
    const uint16_t* __restrict__ be, uint16_t* __restrict__ y,
    float* __restrict__ mean, float* __restrict__ rstd, int64_t R, float eps) {
  const int lane = threadIdx.x & 63;
  const int64_t N = 64 * NE;
  const float invn = 1.0f / (float)N;
  float gv[NE], bv[NE];
#pragma unroll
  for (int c = 0; c < NE / 4; ++c) {
    ushort4 gq = *(const ushort4*)(g + lane * NE + c * 4);
    ushort4 bq = *(const ushort4*)(be + lane * NE + c * 4);
    gv[c * 4] = b2f(gq.x); gv[c * 4 + 1] = b2f(gq.y);
    gv[c * 4 + 2] = b2f(gq.z); gv[c * 4 + 3] = b2f(gq.w);
    bv[c * 4] = b2f(bq.x); bv[c * 4 + 1] = b2f(bq.y);
    bv[c * 4 + 2] = b2f(bq.z); bv[c * 4 + 3] = b2f(bq.w);
  }
  int64_t row = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (; row < R; row += rstride) {
    const uint16_t* xr = x + row * N + lane * NE;
    float xv[NE], s = 0.f;
#pragma unroll
    for (int c = 0; c < NE / 4; ++c) {
      ushort4 q = *(const ushort4*)(xr + c * 4);
      xv[c * 4] = b2f(q.x); xv[c * 4 + 1] = b2f(q.y);
      xv[c * 4 + 2] = b2f(q.z); xv[c * 4 + 3] = b2f(q.w);
      s += xv[c * 4] + xv[c * 4 + 1] + xv[c * 4 + 2] + xv[c * 4 + 3];
    }
    float mu = wave_sum(s) * invn;
    float v = 0.f;
#pragma unroll
    for (int i = 0; i < NE; ++i) {
      float d = xv[i] - mu;
      v += d * d;
    }
    float rs = rsqrtf(wave_sum(v) * invn + eps);
    if (lane == 0) { mean[row] = mu; rstd[row] = rs; }
    uint16_t* yr = y + row * N + lane * NE;
#pragma unroll
    for (int c = 0; c < NE / 4; ++c) {
      ushort4 q;
      q.x = f2b((xv[c * 4] - mu) * rs * gv[c * 4] + bv[c * 4]);
      q.y = f2b((xv[c * 4 + 1] - mu) * rs * gv[c * 4 + 1] + bv[c * 4 + 1]);
      q.z = f2b((xv[c * 4 + 2] - mu) * rs * gv[c * 4 + 2] + bv[c * 4 + 2]);
      q.w = f2b((xv[c * 4 + 3] - mu) * rs * gv[c * 4 + 3] + bv[c * 4 + 3]);
      *(ushort4*)(yr + c * 4) = q;
    }
  }
}

template <int NE>
__global__ void layernorm_dx_wave_b16_k(
    const uint16_t* __restrict__ dy, const uint16_t* __restrict__ x,
    const uint16_t* __restrict__ g, const float* __restrict__ mean,
    const float* __restrict__ rstd, uint16_t* __restrict__ dx, int64_t R,
    float beta_dx) {
  const int lane = threadIdx.x & 63;
  const int64_t N = 64 * NE;
  const float invn = 1.0f / (float)N;
  float gv[NE];
#pragma unroll
  for (int c = 0; c < NE / 4; ++c) {
    ushort4 gq = *(const ushort4*)(g + lane * NE + c * 4);
    gv[c * 4] = b2f(gq.x); gv[c * 4 + 1] = b2f(gq.y);
    gv[c * 4 + 2] = b2f(gq.z); gv[c * 4 + 3] = b2f(gq.w);
  }
  int64_t row = blockIdx.x * 4 + (threadIdx.x >> 6);
  const int64_t rstride = (int64_t)gridDim.x * 4;
  for (; row < R; row += rstride) {
    const uint16_t* xr = x + row * N + lane * NE;
    const uint16_t* dyr = dy + row * N + lane * NE;
    const float mu = mean[row], rs = rstd[row];
    float xh[NE], dyg[NE], s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int c = 0; c < NE / 4; ++c) {
      ushort4 xq = *(const ushort4*)(xr + c * 4);
      ushort4 dq = *(const ushort4*)(dyr + c * 4);
      const uint16_t* xe = (const uint16_t*)&xq;
      const uint16_t* de = (const uint16_t*)&dq;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int e = c * 4 + i;
        xh[e] = (b2f(xe[i]) - mu) * rs;
        dyg[e] = b2f(de[i]) * gv[e];
        s1 += dyg[e];
        s2 += dyg[e] * xh[e];
      }
    }
    const float m1 = wave_sum(s1) * invn;
    const float m2 = wave_sum(s2) * invn;
    uint16_t* dxr = dx + row * N + lane * NE;
#pragma unroll
    for (int c = 0; c < NE / 4; ++c) {
      ushort4 q;
      uint16_t* qe = (uint16_t*)&q;
      if (beta_dx != 0.f) {
        ushort4 pq = *(const ushort4*)(dxr + c * 4);
        const uint16_t* pe = (const uint16_t*)&pq;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int e = c * 4 + i;
          qe[i] = f2b(beta_dx * b2f(pe[i]) +
                      rs * (dyg[e] - m1 - xh[e] * m2));
        }
      } else {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int e = c * 4 + i;
          qe[i] = f2b(rs * (dyg[e] - m1 - xh[e] * m2));
        }
      }
      *(ushort4*)(dxr + c * 4) = q;
    }
  }
}

void layernorm_fwd(const Tensor& x, const Tensor& gamma, const Tensor& beta,
                   Tensor& y, Tensor& mean, Tensor& rstd, float eps) {
  hipStream_t s = stream_of(x);
  int64_t R = x.rows(), N = x.cols();
  dim3 g(grid_1d(R, 1, 2048)), b(256);
  if (x.dtype() == DType::BF16 && N == 768) {
    dim3 gw(grid_1d((R + 3) / 4, 1, 4096));
    layernorm_fwd_wave_b16_k<12><<<gw, b, 0, s>>>(
        (const uint16_t*)x.data(), (const uint16_t*)gamma.data(),
        (const uint16_t*)beta.data(), (uint16_t*)y.data(), (float*)mean.data(),
        (float*)rstd.data(), R, eps);
    NZ_KERNEL_CHECK();
    return;
  }
  if (x.dtype() == DType::BF16 && N == 1024) {
    dim3 gw(grid_1d((R + 3) / 4, 1, 4096));
    layernorm_fwd_wave_b16_k<16><<<gw, b, 0, s>>>(
        (const uint16_t*)x.data(), (const uint16_t*)gamma.data(),
        (const uint16_t*)beta.data(), (uint16_t*)y.data(), (float*)mean.data(),
        (float*)rstd.data(), R, eps);
    NZ_KERNEL_CHECK();
    return;
  }
  if (x.dtype() == DType::BF16)
    layernorm_fwd_k<uint16_t><<<g, b, 0, s>>>(
        (const uint16_t*)x.data(), (const uint16_t*)gamma.data(),
        (const uint16_t*)beta.data(), (uint16_t*)y.data(), (float*)mean.data(),
        (float*)rstd.data(), R, N, eps);
  else
    layernorm_fwd_k<float><<<g, b, 0, s>>>(
        (const float*)x.data(), (const float*)gamma.data(),
        (const float*)beta.data(), (float*)y.data(), (float*)mean.data(),
        (float*)rstd.data(), R, N, eps);
  NZ_KERNEL_CHECK();
}

template <typename T>
__global__ void layernorm_dx_k(const T* __restrict__ dy, const T* __restrict__ x,
                               const T* __restrict__ g, const float* __restrict__ mean,
                               const float* __restrict__ rstd, T* __restrict__ dx,
                               int64_t R, int64_t N, float beta_dx) {
  __shared__ float lds4[4];
  for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
    const T* xr = x + row * N;
    const T* dyr = dy + row * N;
    float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x) {
      float xh = (ld(xr, j) - mu) * rs;
      float dyg = ld(dyr, j) * ld(g, j);
      s1 += dyg;
      s2 += dyg * xh;
    }
    float sum_dy = block_sum<256>(s1, lds4);
    __syncthreads();
    float sum_dyxh = block_sum<256>(s2, lds4);
    T* dxr = dx + row * N;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x) {
      float xh = (ld(xr, j) - mu) * rs;
      float dyg = ld(dyr, j) * ld(g, j);
      float v = rs * (dyg - sum_dy / N - xh * sum_dyxh / N);
      float prev = beta_dx != 0.f ? beta_dx * ld(dxr, j) : 0.f;
      st(dxr, j, prev + v);
    }
    __syncthreads();
  }
}
// dgamma[j] = sum_i dy[i,j]*xhat[i,j]; dbeta[j] = sum_i dy[i,j].
// Two-stage like colsum (row chunks fill the chip), partials [chunk][2][N].
template <typename T>
__global__ void layernorm_dgb1_k(const T* __restrict__ dy, const T* __restrict__ x,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ rstd,
                                 float* __restrict__ partials, int64_t R,
                                 int64_t N, int64_t rows_per_chunk) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (j >= N) return;
  int64_t r0 = blockIdx.y * rows_per_chunk;
  int64_t r1 = min(R, r0 + rows_per_chunk);
  float sg = 0.f, sb = 0.f;
  for (int64_t i = r0; i < r1; ++i) {
    float dyv = ld(dy, i * N + j);
    float xh = (ld(x, i * N + j) - mean[i]) * rstd[i];
    sg += dyv * xh;
    sb += dyv;
  }
  partials[(blockIdx.y * 2 + 0) * N + j] = sg;
  partials[(blockIdx.y * 2 + 1) * N + j] = sb;
}
// bf16 fast path: 8 columns per thread, ushort8 sweeps over dy and x
// Same WG layout as colsum_stage1_vec_k (128 cols x 16 row-subchunks with a
// fixed-order LDS fold): the one-thread-per-column-strip version was
// latency-bound on BERT's N=768 (96 active lanes per chunk).
__global__ void layernorm_dgb1_vec_k(const ushort8_t* __restrict__ dy8,
                                     const ushort8_t* __restrict__ x8,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ partials, int64_t R,
                                     int64_t N8, int64_t rows_per_chunk) {
  __shared__ float smg[16 * 128], smb[16 * 128];
  const int jj = threadIdx.x & 15;
  const int sub = threadIdx.x >> 4;
  const int64_t j8 = (int64_t)blockIdx.x * 16 + jj;
  const int64_t r0 = blockIdx.y * rows_per_chunk;
  const int64_t r1 = min(R, r0 + rows_per_chunk);
  float sg[8] = {}, sb[8] = {};
  if (j8 < N8) {
    for (int64_t i = r0 + sub; i < r1; i += 16) {
      ushort8_t dv = dy8[i * N8 + j8];
      ushort8_t xv = x8[i * N8 + j8];
      float mu = mean[i], rs = rstd[i];
#pragma unroll
      for (int u = 0; u < 8; ++u) {
        float dyv = b2f(((const uint16_t*)&dv)[u]);
        float xh = (b2f(((const uint16_t*)&xv)[u]) - mu) * rs;
        sg[u] += dyv * xh;
        sb[u] += dyv;
      }
    }
  }
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    smg[sub * 128 + jj * 8 + u] = sg[u];
    smb[sub * 128 + jj * 8 + u] = sb[u];
  }
  __syncthreads();
  if (threadIdx.x < 128) {
    const int col = threadIdx.x;
    float vg = 0.f, vb = 0.f;
#pragma unroll
    for (int s_ = 0; s_ < 16; ++s_) {
      vg += smg[s_ * 128 + col];
      vb += smb[s_ * 128 + col];
    }
    int64_t gcol = (int64_t)blockIdx.x * 128 + col;
    if (gcol < N8 * 8) {
      partials[(blockIdx.y * 2 + 0) * N8 * 8 + gcol] = vg;
      partials[(blockIdx.y * 2 + 1) * N8 * 8 + gcol] = vb;
    }
  }
}

template <typename T>
__global__ void layernorm_dgb2_k(const float* __restrict__ partials,
                                 T* __restrict__ dgamma, T* __restrict__ dbeta,
                                 int64_t N, int chunks) {
  int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  if (j >= N) return;
  // paired accumulators for load ILP (fold is latency-bound; fixed order)
  float g0 = 0.f, g1 = 0.f, b0 = 0.f, b1 = 0.f;
  int c = 0;
  for (; c + 2 <= chunks; c += 2) {
    g0 += partials[((int64_t)c * 2 + 0) * N + j];
    b0 += partials[((int64_t)c * 2 + 1) * N + j];
    g1 += partials[((int64_t)(c + 1) * 2 + 0) * N + j];
    b1 += partials[((int64_t)(c + 1) * 2 + 1) * N + j];
  }
  for (; c < chunks; ++c) {
    g0 += partials[((int64_t)c * 2 + 0) * N + j];
    b0 += partials[((int64_t)c * 2 + 1) * N + j];
  }
  st(dgamma, j, g0 + g1);
  st(dbeta, j, b0 + b1);
}
void layernorm_bwd(const Tensor& dy, const Tensor& x, const Tensor& gamma,
                   const Tensor& mean, const Tensor& rstd, Tensor& dx,
                   Tensor& dgamma, Tensor& dbeta, float beta_dx) {
  hipStream_t s = stream_of(dy);
  int64_t R = x.rows(), N = x.cols();
  dim3 b(256);
  int64_t colblocks = (N + 255) / 256;
  int chunks = (int)std::min<int64_t>(
      (R + 63) / 64,
      std::min<int64_t>(64, std::max<int64_t>(1, 512 / colblocks)));
  int64_t rows_per_chunk = (R + chunks - 1) / chunks;
  Tensor partials(x.ctx(), DType::F32, Shape{(int64_t)chunks * 2, N});
  dim3 g1(colblocks, chunks), g2(colblocks);
  if (x.dtype() == DType::BF16) {
    if (N == 768) {
      layernorm_dx_wave_b16_k<12><<<dim3(grid_1d((R + 3) / 4, 1, 4096)), b, 0,
                                    s>>>(
          (const uint16_t*)dy.data(), (const uint16_t*)x.data(),
          (const uint16_t*)gamma.data(), (const float*)mean.data(),
          (const float*)rstd.data(), (uint16_t*)dx.data(), R, beta_dx);
    } else if (N == 1024) {
      layernorm_dx_wave_b16_k<16><<<dim3(grid_1d((R + 3) / 4, 1, 4096)), b, 0,
                                    s>>>(
          (const uint16_t*)dy.data(), (const uint16_t*)x.data(),
          (const uint16_t*)gamma.data(), (const float*)mean.data(),
          (const float*)rstd.data(), (uint16_t*)dx.data(), R, beta_dx);
    } else
    layernorm_dx_k<uint16_t><<<dim3(grid_1d(R, 1, 2048)), b, 0, s>>>(
        (const uint16_t*)dy.data(), (const uint16_t*)x.data(),
        (const uint16_t*)gamma.data(), (const float*)mean.data(),
        (const float*)rstd.data(), (uint16_t*)dx.data(), R, N, beta_dx);
    if (N % 8 == 0) {
      int64_t cb8 = (N + 127) / 128;
      int ch8 = (int)std::min<int64_t>(
          (R + 15) / 16,
          std::min<int64_t>(256, std::max<int64_t>(1, 1024 / cb8)));
      int64_t rpc8 = (R + ch8 - 1) / ch8;
      Tensor p8(x.ctx(), DType::F32, Shape{(int64_t)ch8 * 2, N});
      layernorm_dgb1_vec_k<<<dim3(cb8, ch8), b, 0, s>>>(
          (const ushort8_t*)dy.data(), (const ushort8_t*)x.data(),
          (const float*)mean.data(), (const float*)rstd.data(),
          (float*)p8.data(), R, N / 8, rpc8);
      layernorm_dgb2_k<uint16_t><<<g2, b, 0, s>>>(
          (const float*)p8.data(), (uint16_t*)dgamma.data(),
          (uint16_t*)dbeta.data(), N, ch8);
    } else {
      layernorm_dgb1_k<uint16_t><<<g1, b, 0, s>>>(
          (const uint16_t*)dy.data(), (const uint16_t*)x.data(),
          (const float*)mean.data(), (const float*)rstd.data(),
          (float*)partials.data(), R, N, rows_per_chunk);
      layernorm_dgb2_k<uint16_t><<<g2, b, 0, s>>>(
          (const float*)partials.data(), (uint16_t*)dgamma.data(),
          (uint16_t*)dbeta.data(), N, chunks);
    }
  } else {
    layernorm_dx_k<float><<<dim3(grid_1d(R, 1, 2048)), b, 0, s>>>(
        (const float*)dy.data(), (const float*)x.data(), (const float*)gamma.data(),
        (const float*)mean.data(), (const float*)rstd.data(), (float*)dx.data(),
        R, N, beta_dx);
    layernorm_dgb1_k<float><<<g1, b, 0, s>>>(
        (const float*)dy.data(), (const float*)x.data(), (const float*)mean.data(),
        (const float*)rstd.data(), (float*)partials.data(), R, N, rows_per_chunk);
    layernorm_dgb2_k<float><<<g2, b, 0, s>>>(
        (const float*)partials.data(), (float*)dgamma.data(),
        (float*)dbeta.data(), N, chunks);
  }
  NZ_KERNEL_CHECK();
}

// ----------------------------------------------------------- K4: softmax ---
// Block per row, online-style: pass1 max, pass2 exp+sum (f32), pass3 scale.
template <typename T>
__global__ void softmax_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                              int64_t R, int64_t N, float scale) {
  __shared__ float lds4[4];
  for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
    const T* xr = x + row * N;
    float m = -INFINITY;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      m = fmaxf(m, scale * ld(xr, j));
    m = block_max<256>(m, lds4);
    __syncthreads();
    float sum = 0.f;
    T* yr = y + row * N;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x) {
      float e = __expf(scale * ld(xr, j) - m);
      st(yr, j, e);
      sum += e;
    }
    float inv = 1.0f / block_sum<256>(sum, lds4);
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      st(yr, j, ld(yr, j) * inv);
    __syncthreads();
  }
}
// One wave per row for narrow rows (attention scores S<=256): the row
// lives in registers (<=4 elems/lane), one pass, wave shuffles only.
template <typename T, int EPL>  // elements per lane (N = EPL*64)
__global__ void softmax_fwd_wave_k(const T* __restrict__ x, T* __restrict__ y,
                                   int64_t R, int64_t N, float scale) {
  int64_t row = blockIdx.x * (int64_t)(blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (row >= R) return;
  const T* xr = x + row * N;
  float vals[EPL];
  float m = -INFINITY;
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int64_t j = lane * EPL + i;  // contiguous per lane -> 2*EPL B/lane loads
    vals[i] = j < N ? scale * ld(xr, j) : -INFINITY;
    m = fmaxf(m, vals[i]);
  }
  m = wave_max(m);
  float sum = 0.f;
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    vals[i] = __expf(vals[i] - m);
    sum += vals[i];
  }
  sum = wave_sum(sum);
  float inv = 1.0f / sum;
  T* yr = y + row * N;
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int64_t j = lane * EPL + i;
    if (j < N) st(yr, j, vals[i] * inv);
  }
}

template <typename T, int EPL>
__global__ void softmax_bwd_wave_k(const T* __restrict__ dy,
                                   const T* __restrict__ y, T* __restrict__ dx,
                                   int64_t R, int64_t N, float scale) {
  int64_t row = blockIdx.x * (int64_t)(blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (row >= R) return;
  const T* dyr = dy + row * N;
  const T* yr = y + row * N;
  float dv[EPL], yv[EPL], dot = 0.f;
  T* dxr = dx + row * N;
  if constexpr (sizeof(T) == 2) {
    if (N == 64 * EPL) {
      // packed dword loads/stores: the scalar path issued 2-byte strided
      // transactions and ran 3.5x off the traffic bound on the S=128 probs
      const uint32_t* dw = (const uint32_t*)dyr + lane * (EPL / 2);
      const uint32_t* yw = (const uint32_t*)yr + lane * (EPL / 2);
      uint32_t dpk[EPL / 2], ypk[EPL / 2];
#pragma unroll
      for (int w = 0; w < EPL / 2; ++w) {
        dpk[w] = dw[w];
        ypk[w] = yw[w];
      }
#pragma unroll
      for (int i = 0; i < EPL; ++i) {
        uint16_t db = (uint16_t)(dpk[i / 2] >> (16 * (i & 1)));
        uint16_t yb = (uint16_t)(ypk[i / 2] >> (16 * (i & 1)));
        dv[i] = b2f(db);
        yv[i] = b2f(yb);
        dot += dv[i] * yv[i];
      }
      dot = wave_sum(dot);
      uint32_t* ow = (uint32_t*)dxr + lane * (EPL / 2);
#pragma unroll
      for (int w = 0; w < EPL / 2; ++w) {
        uint32_t lo = f2b(scale * yv[2 * w] * (dv[2 * w] - dot));
        uint32_t hi = f2b(scale * yv[2 * w + 1] * (dv[2 * w + 1] - dot));
        ow[w] = lo | (hi << 16);
      }
      return;
    }
  }
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int64_t j = lane * EPL + i;
    dv[i] = j < N ? ld(dyr, j) : 0.f;
    yv[i] = j < N ? ld(yr, j) : 0.f;
    dot += dv[i] * yv[i];
  }
  dot = wave_sum(dot);
#pragma unroll
  for (int i = 0; i < EPL; ++i) {
    int64_t j = lane * EPL + i;
    if (j < N) st(dxr, j, scale * yv[i] * (dv[i] - dot));
  }
}

void softmax_fwd(const Tensor& x, Tensor& y, float scale) {
  hipStream_t s = stream_of(x);
  int64_t R = x.rows(), N = x.cols();
  dim3 g(grid_1d(R, 1, 4096)), b(256);
  if (N <= 128 && x.dtype() == DType::BF16) {
    dim3 gw(grid_1d((R + 3) / 4, 1, (int64_t)1 << 30));
    softmax_fwd_wave_k<uint16_t, 2><<<gw, b, 0, s>>>(
        (const uint16_t*)x.data(), (uint16_t*)y.data(), R, N, scale);
    NZ_KERNEL_CHECK();
    return;
  }
  if (N <= 256 && x.dtype() == DType::BF16) {
    dim3 gw(grid_1d((R + 3) / 4, 1, (int64_t)1 << 30));
    softmax_fwd_wave_k<uint16_t, 4><<<gw, b, 0, s>>>(
        (const uint16_t*)x.data(), (uint16_t*)y.data(), R, N, scale);
    NZ_KERNEL_CHECK();
    return;
  }
  if (x.dtype() == DType::BF16)
    softmax_fwd_k<uint16_t><<<g, b, 0, s>>>((const uint16_t*)x.data(),
                                            (uint16_t*)y.data(), R, N, scale);
  else
    softmax_fwd_k<float><<<g, b, 0, s>>>((const float*)x.data(), (float*)y.data(),
                                         R, N, scale);
  NZ_KERNEL_CHECK();
}

template <typename T>
__global__ void softmax_bwd_k(const T* __restrict__ dy, const T* __restrict__ y,
                              T* __restrict__ dx, int64_t R, int64_t N, float scale) {
  __shared__ float lds4[4];
  for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
    const T* dyr = dy + row * N;
    const T* yr = y + row * N;
    float dot = 0.f;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      dot += ld(dyr, j) * ld(yr, j);
    dot = block_sum<256>(dot, lds4);
    T* dxr = dx + row * N;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      st(dxr, j, scale * ld(yr, j) * (ld(dyr, j) - dot));
    __syncthreads();
  }
}
void softmax_bwd(const Tensor& dy, const Tensor& y, Tensor& dx, float scale) {
  hipStream_t s = stream_of(dy);
  int64_t R = y.rows(), N = y.cols();
  dim3 g(grid_1d(R, 1, 4096)), b(256);
  if (N <= 128 && y.dtype() == DType::BF16) {
    dim3 gw(grid_1d((R + 3) / 4, 1, (int64_t)1 << 30));
    softmax_bwd_wave_k<uint16_t, 2><<<gw, b, 0, s>>>(
        (const uint16_t*)dy.data(), (const uint16_t*)y.data(),
        (uint16_t*)dx.data(), R, N, scale);
    NZ_KERNEL_CHECK();
    return;
  }
  if (N <= 256 && y.dtype() == DType::BF16) {
    dim3 gw(grid_1d((R + 3) / 4, 1, (int64_t)1 << 30));
    softmax_bwd_wave_k<uint16_t, 4><<<gw, b, 0, s>>>(
        (const uint16_t*)dy.data(), (const uint16_t*)y.data(),
        (uint16_t*)dx.data(), R, N, scale);
    NZ_KERNEL_CHECK();
    return;
  }
  if (y.dtype() == DType::BF16)
    softmax_bwd_k<uint16_t><<<g, b, 0, s>>>((const uint16_t*)dy.data(),
                                            (const uint16_t*)y.data(),
                                            (uint16_t*)dx.data(), R, N, scale);
  else
    softmax_bwd_k<float><<<g, b, 0, s>>>((const float*)dy.data(),
                                         (const float*)y.data(), (float*)dx.data(),
                                         R, N, scale);
  NZ_KERNEL_CHECK();
}

// --------------------------------------------------------- K6/K7: optim ---
// HBM-bound (C3 "one kernel launch per bucket"): fully vectorized — 16-B
// float4 loads/stores on the f32 state, 8-B ushort4 on bf16 param/grad.
__global__ void adam_vec_k(ushort4* __restrict__ param,
                           const ushort4* __restrict__ grad,
                           float4* __restrict__ master, float4* __restrict__ m,
                           float4* __restrict__ v, int64_t n4, float lr,
                           float b1, float b2, float eps, float bc1, float bc2,
                           float gs, float wd, const float* __restrict__ bc_dev) {
  if (bc_dev) { bc1 = bc_dev[0]; bc2 = bc_dev[1]; }  // graph-replay path
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    ushort4 gv = grad[i];
    float4 ma = master[i], mv = m[i], vv = v[i];
    float gf[4] = {b2f(gv.x) * gs, b2f(gv.y) * gs, b2f(gv.z) * gs,
                   b2f(gv.w) * gs};
    float* map = &ma.x;
    float* mvp = &mv.x;
    float* vvp = &vv.x;
    ushort4 pv;
    uint16_t* pvp = &pv.x;
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      float g = gf[u];
      if (wd != 0.f) g += wd * map[u];
      float mi = b1 * mvp[u] + (1 - b1) * g;
      float vi = b2 * vvp[u] + (1 - b2) * g * g;
      mvp[u] = mi;
      vvp[u] = vi;
      map[u] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
      pvp[u] = f2b(map[u]);
    }
    master[i] = ma;
    m[i] = mv;
    v[i] = vv;
    param[i] = pv;
  }
}

template <typename TP, typename TG>
__global__ void adam_k(TP* __restrict__ param, const TG* __restrict__ grad,
                       float* __restrict__ master, float* __restrict__ m,
                       float* __restrict__ v, int64_t n, float lr, float b1,
                       float b2, float eps, float bc1, float bc2, float gs,
                       float wd, const float* __restrict__ bc_dev) {
  if (bc_dev) { bc1 = bc_dev[0]; bc2 = bc_dev[1]; }  // graph-replay path
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = ld(grad, i) * gs;
    float ma = master[i];
    if (wd != 0.f) g += wd * ma;
    float mi = b1 * m[i] + (1 - b1) * g;
    float vi = b2 * v[i] + (1 - b2) * g * g;
    m[i] = mi;
    v[i] = vi;
    ma -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    master[i] = ma;
    st(param, i, ma);
  }
}
void adam_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor& m,
               Tensor& v, float lr, float b1, float b2, float eps, int step,
               float gs, float wd, const Tensor* bc_dev) {
  hipStream_t s = stream_of(param);
  int64_t n = param.numel();
  float bc1 = 1.f - powf(b1, (float)step), bc2 = 1.f - powf(b2, (float)step);
  dim3 g(grid_1d((n + 3) / 4, 256)), b(256);
  if (param.dtype() == DType::BF16 && (n & 3) == 0)
    adam_vec_k<<<g, b, 0, s>>>(
        (ushort4*)param.data(), (const ushort4*)grad.data(),
        (float4*)master.data(), (float4*)m.data(), (float4*)v.data(), n / 4, lr,
        b1, b2, eps, bc1, bc2, gs, wd,
        bc_dev ? (const float*)bc_dev->data() : nullptr);
  else if (param.dtype() == DType::BF16)
    adam_k<uint16_t, uint16_t><<<g, b, 0, s>>>(
        (uint16_t*)param.data(), (const uint16_t*)grad.data(),
        (float*)master.data(), (float*)m.data(), (float*)v.data(), n, lr, b1, b2,
        eps, bc1, bc2, gs, wd,
        bc_dev ? (const float*)bc_dev->data() : nullptr);
  else
    adam_k<float, float><<<g, b, 0, s>>>(
        (float*)param.data(), (const float*)grad.data(), (float*)master.data(),
        (float*)m.data(), (float*)v.data(), n, lr, b1, b2, eps, bc1, bc2, gs, wd,
        bc_dev ? (const float*)bc_dev->data() : nullptr);
  NZ_KERNEL_CHECK();
}

template <typename TP>
__global__ void sgd_k(TP* __restrict__ param, const TP* __restrict__ grad,
                      float* __restrict__ master, float* __restrict__ mom,
                      int64_t n, float lr, float mu, float gs) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = ld(grad, i) * gs;
    if (mom) {
      float mi = mu * mom[i] + g;
      mom[i] = mi;
      g = mi;
    }
    float ma = master[i] - lr * g;
    master[i] = ma;
    st(param, i, ma);
  }
}
void sgd_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor* momentum,
              float lr, float mu, float gs) {
  hipStream_t s = stream_of(param);
  int64_t n = param.numel();
  dim3 g(grid_1d(n, 256)), b(256);
  if (param.dtype() == DType::BF16)
    sgd_k<uint16_t><<<g, b, 0, s>>>((uint16_t*)param.data(),
                                    (const uint16_t*)grad.data(),
                                    (float*)master.data(),
                                    momentum ? (float*)momentum->data() : nullptr,
                                    n, lr, mu, gs);
  else
    sgd_k<float><<<g, b, 0, s>>>((float*)param.data(), (const float*)grad.data(),
                                 (float*)master.data(),
                                 momentum ? (float*)momentum->data() : nullptr, n,
                                 lr, mu, gs);
  NZ_KERNEL_CHECK();
}

// ----------------------------------------------------------- K8: dropout ---
template <typename T>
__global__ void dropout_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                              T* __restrict__ mask, int64_t n, float p,
                              float inv, uint64_t seed) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float m = rng_uniform01(seed, (uint64_t)i) >= p ? 1.0f : 0.0f;
    st(mask, i, m);
    st(y, i, ld(x, i) * m * inv);
  }
}
template <typename T>
__global__ void dropout_bwd_k(const T* __restrict__ dy, const T* __restrict__ mask,
                              T* __restrict__ dx, int64_t n, float inv) {
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) st(dx, i, ld(dy, i) * ld(mask, i) * inv);
}
void dropout_fwd(const Tensor& x, Tensor& y, Tensor& mask, float p, uint64_t seed) {
  hipStream_t s = stream_of(x);
  int64_t n = x.numel();
  float inv = p < 1.f ? 1.0f / (1.0f - p) : 0.f;
  dim3 g(grid_1d(n, 256)), b(256);
  if (x.dtype() == DType::BF16)
    dropout_fwd_k<uint16_t><<<g, b, 0, s>>>((const uint16_t*)x.data(),
                                            (uint16_t*)y.data(),
                                            (uint16_t*)mask.data(), n, p, inv, seed);
  else
    dropout_fwd_k<float><<<g, b, 0, s>>>((const float*)x.data(), (float*)y.data(),
                                         (float*)mask.data(), n, p, inv, seed);
  NZ_KERNEL_CHECK();
}
void dropout_bwd(const Tensor& dy, const Tensor& mask, Tensor& dx, float p) {
  hipStream_t s = stream_of(dy);
  int64_t n = dy.numel();
  float inv = p < 1.f ? 1.0f / (1.0f - p) : 0.f;
  dim3 g(grid_1d(n, 256)), b(256);
  if (dy.dtype() == DType::BF16)
    dropout_bwd_k<uint16_t><<<g, b, 0, s>>>((const uint16_t*)dy.data(),
                                            (const uint16_t*)mask.data(),
                                            (uint16_t*)dx.data(), n, inv);
  else
    dropout_bwd_k<float><<<g, b, 0, s>>>((const float*)dy.data(),
                                         (const float*)mask.data(),
                                         (float*)dx.data(), n, inv);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------------ K9: losses ---
// Two-stage deterministic reduction: per-block partials, then single-block
// combine (no float atomics -> bit-stable run to run).
template <typename T>
__global__ void mse_partial_k(const T* __restrict__ y, const T* __restrict__ t,
                              T* __restrict__ dy, float* __restrict__ partials,
                              int64_t n, float gs) {
  __shared__ float lds4[4];
  float acc = 0.f;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float inv_n = 1.0f / float(n);
  for (; i < n; i += stride) {
    float d = ld(y, i) - ld(t, i);
    acc += d * d;
    if (dy) st(dy, i, 2.0f * d * inv_n * gs);
  }
  float bs = block_sum<256>(acc, lds4);
  if (threadIdx.x == 0) partials[blockIdx.x] = bs;
}

// bf16 fast path: ushort4 vector loads/stores (HBM-bound op, Guideline 13).
__global__ void mse_partial_b16_k(const ushort4* __restrict__ y,
                                  const ushort4* __restrict__ t,
                                  ushort4* __restrict__ dy,
                                  float* __restrict__ partials, int64_t n4,
                                  float gs_over_n) {
  __shared__ float lds4[4];
  float acc = 0.f;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    ushort4 yv = y[i], tv = t[i];
    float d0 = b2f(yv.x) - b2f(tv.x), d1 = b2f(yv.y) - b2f(tv.y);
    float d2 = b2f(yv.z) - b2f(tv.z), d3 = b2f(yv.w) - b2f(tv.w);
    acc += d0 * d0 + d1 * d1 + d2 * d2 + d3 * d3;
    if (dy) {
      ushort4 o;
      o.x = f2b(2.0f * d0 * gs_over_n);
      o.y = f2b(2.0f * d1 * gs_over_n);
      o.z = f2b(2.0f * d2 * gs_over_n);
      o.w = f2b(2.0f * d3 * gs_over_n);
      dy[i] = o;
    }
  }
  float bs = block_sum<256>(acc, lds4);
  if (threadIdx.x == 0) partials[blockIdx.x] = bs;
}
__global__ void combine_partials_k(const float* __restrict__ partials, int nb,
                                   float* __restrict__ out, float scale) {
  __shared__ float lds4[4];
  float acc = 0.f;
  for (int i = threadIdx.x; i < nb; i += blockDim.x) acc += partials[i];
  float s = block_sum<256>(acc, lds4);
  if (threadIdx.x == 0) out[0] = s * scale;
}

void mse_loss(const Tensor& y, const Tensor& target, Tensor& loss_out, Tensor* dy,
              float gs) {
  hipStream_t s = stream_of(y);
  int64_t n = y.numel();
  int nb = (int)grid_1d((n + 3) / 4, 256, 1024);
  Tensor partials(y.ctx(), DType::F32, Shape{nb});
  dim3 g(nb), b(256);
  if (y.dtype() == DType::BF16 && (n & 3) == 0)
    mse_partial_b16_k<<<g, b, 0, s>>>(
        (const ushort4*)y.data(), (const ushort4*)target.data(),
        dy ? (ushort4*)dy->data() : nullptr, (float*)partials.data(), n / 4,
        gs / float(n));
  else if (y.dtype() == DType::BF16)
    mse_partial_k<uint16_t><<<g, b, 0, s>>>(
        (const uint16_t*)y.data(), (const uint16_t*)target.data(),
        dy ? (uint16_t*)dy->data() : nullptr, (float*)partials.data(), n, gs);
  else
    mse_partial_k<float><<<g, b, 0, s>>>(
        (const float*)y.data(), (const float*)target.data(),
        dy ? (float*)dy->data() : nullptr, (float*)partials.data(), n, gs);
  combine_partials_k<<<dim3(1), b, 0, s>>>((const float*)partials.data(), nb,
                                           (float*)loss_out.data(), 1.0f / n);
  NZ_KERNEL_CHECK();
}

// CE over rows (f32 or bf16 logits; labels hold integral class ids in f32).
template <typename T>
__global__ void ce_loss_k(const T* __restrict__ logits,
                          const float* __restrict__ labels,
                          T* __restrict__ dlogits,
                          float* __restrict__ partials, int64_t R, int64_t N,
                          float gs) {
  __shared__ float lds4[4];
  float row_losses = 0.f;
  for (int64_t row = blockIdx.x; row < R; row += gridDim.x) {
    const T* lr = logits + row * N;
    float m = -INFINITY;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      m = fmaxf(m, ld(lr, j));
    m = block_max<256>(m, lds4);
    __syncthreads();
    float z = 0.f;
    for (int64_t j = threadIdx.x; j < N; j += blockDim.x)
      z += __expf(ld(lr, j) - m);
    z = block_sum<256>(z, lds4);
    int64_t y = (int64_t)labels[row];
    if (threadIdx.x == 0) row_losses += -(ld(lr, y) - m - __logf(z));
    if (dlogits) {
      T* dlr = dlogits + row * N;
      for (int64_t j = threadIdx.x; j < N; j += blockDim.x) {
        float p = __expf(ld(lr, j) - m) / z;
        st(dlr, j, (p - (j == y ? 1.f : 0.f)) / float(R) * gs);
      }
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) partials[blockIdx.x] = row_losses;
}
void ce_loss(const Tensor& logits, const Tensor& labels, Tensor& loss_out,
             Tensor* dlogits, float gs) {
  hipStream_t s = stream_of(logits);
  int64_t R = logits.rows(), N = logits.cols();
  int nb = (int)grid_1d(R, 1, 1024);
  Tensor partials(logits.ctx(), DType::F32, Shape{nb});
  if (logits.dtype() == DType::BF16)
    ce_loss_k<uint16_t><<<dim3(nb), dim3(256), 0, s>>>(
        (const uint16_t*)logits.data(), (const float*)labels.data(),
        dlogits ? (uint16_t*)dlogits->data() : nullptr,
        (float*)partials.data(), R, N, gs);
  else
    ce_loss_k<float><<<dim3(nb), dim3(256), 0, s>>>(
        (const float*)logits.data(), (const float*)labels.data(),
        dlogits ? (float*)dlogits->data() : nullptr, (float*)partials.data(),
        R, N, gs);
  combine_partials_k<<<dim3(1), dim3(256), 0, s>>>(
      (const float*)partials.data(), nb, (float*)loss_out.data(), 1.0f / R);
  NZ_KERNEL_CHECK();
}

// ------------------------------------------------------- K10: reductions ---
template <typename T, bool SQ>
__global__ void reduce_partial_k(const T* __restrict__ src,
                                 float* __restrict__ partials, int64_t n) {
  __shared__ float lds4[4];
  float acc = 0.f;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float v = ld(src, i);
    acc += SQ ? v * v : v;
  }
  float bs = block_sum<256>(acc, lds4);
  if (threadIdx.x == 0) partials[blockIdx.x] = bs;
}
template <bool SQ>
static void reduce_impl(const Tensor& src, Tensor& out) {
  hipStream_t s = stream_of(src);
  int64_t n = src.numel();
  int nb = (int)grid_1d((n + 3) / 4, 256, 1024);
  Tensor partials(src.ctx(), DType::F32, Shape{nb});
  if (src.dtype() == DType::BF16)
    reduce_partial_k<uint16_t, SQ><<<dim3(nb), dim3(256), 0, s>>>(
        (const uint16_t*)src.data(), (float*)partials.data(), n);
  else
    reduce_partial_k<float, SQ><<<dim3(nb), dim3(256), 0, s>>>(
        (const float*)src.data(), (float*)partials.data(), n);
  combine_partials_k<<<dim3(1), dim3(256), 0, s>>>((const float*)partials.data(),
                                                   nb, (float*)out.data(), 1.0f);
  NZ_KERNEL_CHECK();
}
void reduce_sum(const Tensor& src, Tensor& out) { reduce_impl<false>(src, out); }
void l2norm_sq(const Tensor& src, Tensor& out) { reduce_impl<true>(src, out); }

// -------------------------------------------- attention layout (K8 fam) ---
// One thread per dh-element group of 2 (ushort2/float2 moves); dh >= 2.
template <typename T, bool FORWARD>
__global__ void attn_split_k(const T* __restrict__ a, T* __restrict__ q,
                             T* __restrict__ k, T* __restrict__ v, int64_t total,
                             int64_t S, int64_t heads, int64_t dh) {
  int64_t H = heads * dh;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {
    int64_t d = i % dh, h = (i / dh) % heads, which = (i / (dh * heads)) % 3;
    int64_t s_ = (i / (3 * H)) % S, b = i / (3 * H * S);
    int64_t bhsd = ((b * heads + h) * S + s_) * dh + d;
    T* dst3 = which == 0 ? q : (which == 1 ? k : v);
    if (FORWARD)
      dst3[bhsd] = a[i];
    else
      const_cast<T*>(a)[i] = dst3[bhsd];
  }
}

// bf16 fast path: one thread moves 8 contiguous elements along dh
// (dh % 8 == 0) with one index computation per 16-B chunk.
template <bool FORWARD>
__global__ void attn_split_vec_k(uint16_t* __restrict__ a,
                                 uint16_t* __restrict__ q,
                                 uint16_t* __restrict__ k,
                                 uint16_t* __restrict__ v, int64_t chunks,
                                 int64_t S, int64_t heads, int64_t dh) {
  using v8 = __attribute__((ext_vector_type(8))) short;
  int64_t dh8 = dh / 8;
  int64_t H8 = heads * dh8;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < chunks; i += stride) {
    int64_t d = i % dh8, h = (i / dh8) % heads, which = (i / H8) % 3;
    int64_t s_ = (i / (3 * H8)) % S, b = i / (3 * H8 * S);
    int64_t bhsd = (((b * heads + h) * S + s_) * dh8 + d) * 8;
    uint16_t* dst3 = which == 0 ? q : (which == 1 ? k : v);
    if (FORWARD)
      *(v8*)(dst3 + bhsd) = *(const v8*)(a + i * 8);
    else
      *(v8*)(a + i * 8) = *(const v8*)(dst3 + bhsd);
  }
}
void attn_split_qkv(const Tensor& qkv, Tensor& q, Tensor& k, Tensor& v,
                    int64_t B, int64_t S, int64_t heads, int64_t dh) {
  hipStream_t s = stream_of(qkv);
  int64_t total = qkv.numel();
  dim3 g(grid_1d(total, 256, 4096)), b(256);
  if (qkv.dtype() == DType::BF16 && dh % 8 == 0)
    attn_split_vec_k<true><<<dim3(grid_1d(total / 8, 256, 4096)), b, 0, s>>>(
        (uint16_t*)qkv.data(), (uint16_t*)q.data(), (uint16_t*)k.data(),
        (uint16_t*)v.data(), total / 8, S, heads, dh);
  else if (qkv.dtype() == DType::BF16)
    attn_split_k<uint16_t, true><<<g, b, 0, s>>>(
        (const uint16_t*)qkv.data(), (uint16_t*)q.data(), (uint16_t*)k.data(),
        (uint16_t*)v.data(), total, S, heads, dh);
  else
    attn_split_k<float, true><<<g, b, 0, s>>>(
        (const float*)qkv.data(), (float*)q.data(), (float*)k.data(),
        (float*)v.data(), total, S, heads, dh);
  NZ_KERNEL_CHECK();
}
void attn_split_qkv_bwd(const Tensor& dq, const Tensor& dk, const Tensor& dv,
                        Tensor& dqkv, int64_t B, int64_t S, int64_t heads,
                        int64_t dh) {
  hipStream_t s = stream_of(dqkv);
  int64_t total = dqkv.numel();
  dim3 g(grid_1d(total, 256, 4096)), b(256);
  if (dqkv.dtype() == DType::BF16 && dh % 8 == 0)
    attn_split_vec_k<false><<<dim3(grid_1d(total / 8, 256, 4096)), b, 0, s>>>(
        (uint16_t*)dqkv.data(), (uint16_t*)dq.data(), (uint16_t*)dk.data(),
        (uint16_t*)dv.data(), total / 8, S, heads, dh);
  else if (dqkv.dtype() == DType::BF16)
    attn_split_k<uint16_t, false><<<g, b, 0, s>>>(
        (const uint16_t*)dqkv.data(), (uint16_t*)dq.data(), (uint16_t*)dk.data(),
        (uint16_t*)dv.data(), total, S, heads, dh);
  else
    attn_split_k<float, false><<<g, b, 0, s>>>(
        (const float*)dqkv.data(), (float*)dq.data(), (float*)dk.data(),
        (float*)dv.data(), total, S, heads, dh);
  NZ_KERNEL_CHECK();
}

template <typename T, bool FORWARD>
__global__ void attn_merge_k(const T* __restrict__ ctx, T* __restrict__ out,
                             int64_t total, int64_t S, int64_t heads, int64_t dh) {
  int64_t H = heads * dh;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < total; i += stride) {  // i indexes [B,S,H]
    int64_t d = i % dh, h = (i / dh) % heads, s_ = (i / H) % S, b = i / (H * S);
    int64_t bhsd = ((b * heads + h) * S + s_) * dh + d;
    if (FORWARD)
      out[i] = ctx[bhsd];
    else
      const_cast<T*>(ctx)[bhsd] = out[i];
  }
}

template <bool FORWARD>
__global__ void attn_merge_vec_k(uint16_t* __restrict__ ctx,
                                 uint16_t* __restrict__ out, int64_t chunks,
                                 int64_t S, int64_t heads, int64_t dh) {
  using v8 = __attribute__((ext_vector_type(8))) short;
  int64_t dh8 = dh / 8;
  int64_t H8 = heads * dh8;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < chunks; i += stride) {  // i indexes chunks of [B,S,H]
    int64_t d = i % dh8, h = (i / dh8) % heads, s_ = (i / H8) % S,
            b = i / (H8 * S);
    int64_t bhsd = (((b * heads + h) * S + s_) * dh8 + d) * 8;
    if (FORWARD)
      *(v8*)(out + i * 8) = *(const v8*)(ctx + bhsd);
    else
      *(v8*)(ctx + bhsd) = *(const v8*)(out + i * 8);
  }
}
void attn_merge(const Tensor& ctx, Tensor& out, int64_t B, int64_t S,
                int64_t heads, int64_t dh) {
  hipStream_t s = stream_of(ctx);
  int64_t total = out.numel();
  dim3 g(grid_1d(total, 256, 4096)), b(256);
  if (ctx.dtype() == DType::BF16 && dh % 8 == 0)
    attn_merge_vec_k<true><<<dim3(grid_1d(total / 8, 256, 4096)), b, 0, s>>>(
        (uint16_t*)ctx.data(), (uint16_t*)out.data(), total / 8, S, heads, dh);
  else if (ctx.dtype() == DType::BF16)
    attn_merge_k<uint16_t, true><<<g, b, 0, s>>>((const uint16_t*)ctx.data(),
                                                 (uint16_t*)out.data(), total, S,
                                                 heads, dh);
  else
    attn_merge_k<float, true><<<g, b, 0, s>>>((const float*)ctx.data(),
                                              (float*)out.data(), total, S, heads, dh);
  NZ_KERNEL_CHECK();
}
void attn_merge_bwd(const Tensor& dout, Tensor& dctx, int64_t B, int64_t S,
                    int64_t heads, int64_t dh) {
  hipStream_t s = stream_of(dout);
  int64_t total = dout.numel();
  dim3 g(grid_1d(total, 256, 4096)), b(256);
  if (dout.dtype() == DType::BF16 && dh % 8 == 0)
    attn_merge_vec_k<false><<<dim3(grid_1d(total / 8, 256, 4096)), b, 0, s>>>(
        (uint16_t*)dctx.data(), (uint16_t*)dout.data(), total / 8, S, heads, dh);
  else if (dout.dtype() == DType::BF16)
    attn_merge_k<uint16_t, false><<<g, b, 0, s>>>((const uint16_t*)dctx.data(),
                                                  (uint16_t*)dout.data(), total, S,
                                                  heads, dh);
  else
    attn_merge_k<float, false><<<g, b, 0, s>>>((const float*)dctx.data(),
                                               (float*)dout.data(), total, S,
                                               heads, dh);
  NZ_KERNEL_CHECK();
}

}  // namespace hip
}  // namespace nz
