// nezha — MI355X-native tensor + dense-NN runtime.
// Core common definitions: status/error handling, dtypes, bf16 conversion,
// deterministic counter-based RNG shared by CPU and HIP backends.
//
// Capability contract: SURVEY.md §0.1 (C1..C15); reference checkout is empty
// (SURVEY.md §0), spec artifact is BASELINE.json.
#pragma once

#include <cstdint>
#include <cstddef>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <stdexcept>

namespace nz {

// ---------------------------------------------------------------- errors ---
// All public entry points catch nz::Error and stash the message in a
// thread-local buffer surfaced through nz_last_error() (include/nezha.h).
struct Error : std::runtime_error {
  explicit Error(const std::string& m) : std::runtime_error(m) {}
};

[[noreturn]] inline void fail(const std::string& msg) { throw Error(msg); }

#define NZ_CHECK(cond, msg)                                                   \
  do {                                                                        \
    if (!(cond)) ::nz::fail(std::string("NZ_CHECK failed: ") + (msg) +        \
                            " at " + __FILE__ + ":" + std::to_string(__LINE__)); \
  } while (0)

// ---------------------------------------------------------------- dtypes ---
enum class DType : int32_t { F32 = 0, BF16 = 1 };
enum class Device : int32_t { CPU = 0, HIP = 1 };

inline size_t dtype_size(DType d) { return d == DType::F32 ? 4 : 2; }
inline const char* dtype_name(DType d) { return d == DType::F32 ? "f32" : "bf16"; }
inline const char* device_name(Device d) { return d == Device::CPU ? "cpu" : "hip"; }

// ------------------------------------------------------------------ bf16 ---
// Host-side bf16 <-> f32. bf16 = top 16 bits of the f32 pattern,
// round-to-nearest-even (matches hardware v_cvt_pk_bf16_f32 semantics).
inline uint16_t f32_to_bf16(float f) {
  uint32_t u;
  std::memcpy(&u, &f, 4);
  if ((u & 0x7fffffffu) > 0x7f800000u) return uint16_t((u >> 16) | 0x0040u);  // quiet NaN
  uint32_t rounding = 0x7fffu + ((u >> 16) & 1u);
  return uint16_t((u + rounding) >> 16);
}
inline float bf16_to_f32(uint16_t h) {
  uint32_t u = uint32_t(h) << 16;
  float f;
  std::memcpy(&f, &u, 4);
  return f;
}

// ------------------------------------------------------------------- rng ---
// Stateless counter-based RNG: splitmix64 of (seed, index). Identical on
// CPU and GPU so golden tests can compare initializations bit-for-bit.
#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define NZ_HD __host__ __device__
#else
#define NZ_HD
#endif

NZ_HD inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97f4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// Uniform in [0,1): use the top 24 bits.
NZ_HD inline float rng_uniform01(uint64_t seed, uint64_t idx) {
  uint64_t h = splitmix64(seed ^ splitmix64(idx));
  return float(h >> 40) * (1.0f / 16777216.0f);
}

// Standard normal via Box-Muller on two decorrelated uniforms.
NZ_HD inline float rng_normal(uint64_t seed, uint64_t idx) {
  uint64_t h1 = splitmix64(seed ^ splitmix64(idx * 2 + 0));
  uint64_t h2 = splitmix64(seed ^ splitmix64(idx * 2 + 1));
  float u1 = (float(h1 >> 40) + 0.5f) * (1.0f / 16777216.0f);  // (0,1)
  float u2 = float(h2 >> 40) * (1.0f / 16777216.0f);
#if defined(__HIP_DEVICE_COMPILE__)
  return __builtin_sqrtf(-2.0f * __builtin_logf(u1)) *
         __builtin_cosf(6.28318530717958647692f * u2);
#else
  return std::sqrt(-2.0f * std::log(u1)) * std::cos(6.28318530717958647692f * u2);
#endif
}

}  // namespace nz
