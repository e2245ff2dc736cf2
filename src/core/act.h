// Shared scalar activation math (host + device): tanh-approx GELU.
#pragma once
#include "src/core/common.h"

namespace nz {

// tanh: libm tanhf on host; on device the exp form with the fast-math
// hardware exp (libm tanhf measured ~40 us/call extra on the BERT ffn
// gelu backward). exp(+inf) -> inf -> 2/inf = 0 gives the correct +-1
// saturation at the extremes.
NZ_HD inline float nz_tanh(float x) {
#if defined(__HIP_DEVICE_COMPILE__)
  return 1.0f - 2.0f / (__expf(2.0f * x) + 1.0f);
#else
  return tanhf(x);
#endif
}

// gelu(x) = 0.5 x (1 + tanh(sqrt(2/pi) (x + 0.044715 x^3)))
NZ_HD inline float gelu_fwd_scalar(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float x3 = x * x * x;
  float t = nz_tanh(c * (x + 0.044715f * x3));
  return 0.5f * x * (1.0f + t);
}

// d gelu / dx at x (same tanh approximation, exact derivative of it)
NZ_HD inline float gelu_bwd_scalar(float x) {
  const float c = 0.7978845608028654f;
  float x2 = x * x;
  float inner = c * (x + 0.044715f * x2 * x);
  float t = nz_tanh(inner);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * c * (1.0f + 3.0f * 0.044715f * x2);
}

}  // namespace nz
