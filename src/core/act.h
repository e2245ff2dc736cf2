// Shared scalar activation math (host + device): tanh-approx GELU.
#pragma once
#include "src/core/common.h"

namespace nz {

// gelu(x) = 0.5 x (1 + tanh(sqrt(2/pi) (x + 0.044715 x^3)))
NZ_HD inline float gelu_fwd_scalar(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float x3 = x * x * x;
  float t = tanhf(c * (x + 0.044715f * x3));
  return 0.5f * x * (1.0f + t);
}

// d gelu / dx at x (same tanh approximation, exact derivative of it)
NZ_HD inline float gelu_bwd_scalar(float x) {
  const float c = 0.7978845608028654f;
  float x2 = x * x;
  float inner = c * (x + 0.044715f * x2 * x);
  float t = tanhf(inner);
  float sech2 = 1.0f - t * t;
  return 0.5f * (1.0f + t) + 0.5f * x * sech2 * c * (1.0f + 3.0f * 0.044715f * x2);
}

}  // namespace nz
