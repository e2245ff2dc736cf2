// Shared device-side helpers for the CDNA4 kernel set.
#pragma once

#include <hip/hip_runtime.h>

#include "src/core/act.h"
#include "src/core/common.h"
#include "src/core/tensor.h"

namespace nz {
namespace hip {

#define NZ_HIP_CHECK_K(expr)                                                  \
  do {                                                                        \
    hipError_t nz_hip_err_ = (expr);                                                   \
    if (nz_hip_err_ != hipSuccess)                                                     \
      ::nz::fail(std::string("HIP error: ") + hipGetErrorString(nz_hip_err_) +         \
                 " in " #expr " at " + __FILE__ + ":" +                       \
                 std::to_string(__LINE__));                                   \
  } while (0)

#define NZ_KERNEL_CHECK()                                                     \
  do {                                                                        \
    hipError_t nz_hip_err_ = hipGetLastError();                                        \
    if (nz_hip_err_ != hipSuccess)                                                     \
      ::nz::fail(std::string("kernel launch failed: ") + hipGetErrorString(nz_hip_err_)); \
  } while (0)

// wave width is 64 on CDNA4 — hard-coded per the programming guide.
constexpr int kWave = 64;

__device__ __forceinline__ float b2f(uint16_t h) {
  union { uint32_t u; float f; } cv;
  cv.u = uint32_t(h) << 16;
  return cv.f;
}
__device__ __forceinline__ uint16_t f2b(float f) {
  union { float f; uint32_t u; } cv;
  cv.f = f;
  if ((cv.u & 0x7fffffffu) > 0x7f800000u) return uint16_t((cv.u >> 16) | 0x40u);
  uint32_t rounding = 0x7fffu + ((cv.u >> 16) & 1u);
  return uint16_t((cv.u + rounding) >> 16);
}

// Generic element accessors for templated f32/bf16 kernels.
template <typename T>
__device__ __forceinline__ float ld(const T* p, int64_t i);
template <>
__device__ __forceinline__ float ld<float>(const float* p, int64_t i) { return p[i]; }
template <>
__device__ __forceinline__ float ld<uint16_t>(const uint16_t* p, int64_t i) {
  return b2f(p[i]);
}
template <typename T>
__device__ __forceinline__ void st(T* p, int64_t i, float v);
template <>
__device__ __forceinline__ void st<float>(float* p, int64_t i, float v) { p[i] = v; }
template <>
__device__ __forceinline__ void st<uint16_t>(uint16_t* p, int64_t i, float v) {
  p[i] = f2b(v);
}

// Wave-wide (64-lane) reductions via xor shuffles.
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block reduction (256 threads = 4 waves) through LDS.
template <int BLOCK>
__device__ __forceinline__ float block_sum(float v, float* lds4) {
  v = wave_sum(v);
  int wid = threadIdx.x / kWave;
  if ((threadIdx.x & (kWave - 1)) == 0) lds4[wid] = v;
  __syncthreads();
  float r = 0.f;
#pragma unroll
  for (int i = 0; i < BLOCK / kWave; ++i) r += lds4[i];
  return r;
}
template <int BLOCK>
__device__ __forceinline__ float block_max(float v, float* lds4) {
  v = wave_max(v);
  int wid = threadIdx.x / kWave;
  if ((threadIdx.x & (kWave - 1)) == 0) lds4[wid] = v;
  __syncthreads();
  float r = -INFINITY;
#pragma unroll
  for (int i = 0; i < BLOCK / kWave; ++i) r = fmaxf(r, lds4[i]);
  return r;
}

// Launch helper: stream of a tensor's context.
inline hipStream_t stream_of(const Tensor& t) {
  t.ctx()->make_current();
  return t.ctx()->compute_stream()->handle();
}

inline int64_t grid_1d(int64_t work, int block, int64_t cap = 2048) {
  int64_t g = (work + block - 1) / block;
  return g < cap ? (g > 0 ? g : 1) : cap;
}

}  // namespace hip
}  // namespace nz
