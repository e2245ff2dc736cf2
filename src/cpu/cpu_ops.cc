#include "src/cpu/cpu_ops.h"

#include <algorithm>
#include <atomic>
#include <cmath>
#include <thread>
#include <vector>

#include "src/core/act.h"

namespace nz {
namespace cpu {

static float* f32p(const Tensor& t) {
  NZ_CHECK(t.dtype() == DType::F32, "cpu op expects f32 tensor");
  return (float*)t.data();
}

void parallel_for(int64_t n, int64_t grain, const std::function<void(int64_t, int64_t)>& fn) {
  if (n <= 0) return;
  unsigned hw = std::thread::hardware_concurrency();
  int64_t nthreads = std::min<int64_t>(hw ? hw : 4, (n + grain - 1) / grain);
  if (nthreads <= 1) { fn(0, n); return; }
  std::vector<std::thread> ts;
  int64_t chunk = (n + nthreads - 1) / nthreads;
  for (int64_t t = 0; t < nthreads; ++t) {
    int64_t lo = t * chunk, hi = std::min(n, lo + chunk);
    if (lo >= hi) break;
    ts.emplace_back(fn, lo, hi);
  }
  for (auto& th : ts) th.join();
}

// ------------------------------------------------------------------ gemm ---
// Row-major blocked GEMM, f32. op(A)=[M,K], op(B)=[K,N], C=[M,N].
void gemm(const Tensor& A, const Tensor& B, Tensor& C, bool transA, bool transB,
          float alpha, float beta, const ops::GemmEpilogue* epi) {
  const float* a = f32p(A);
  const float* b = f32p(B);
  float* c = f32p(C);
  int64_t M = C.rows(), N = C.cols();
  int64_t K = transA ? A.rows() : A.cols();
  NZ_CHECK((transA ? A.cols() : A.rows()) == M, "gemm M mismatch");
  NZ_CHECK((transB ? B.rows() : B.cols()) == N, "gemm N mismatch");
  NZ_CHECK((transB ? B.cols() : B.rows()) == K, "gemm K mismatch");
  int64_t lda = A.cols(), ldb = B.cols();
  const float* bias = epi && epi->bias ? f32p(*epi->bias) : nullptr;
  float* pre = epi && epi->save_pre ? f32p(*epi->save_pre) : nullptr;
  bool gelu = epi && epi->gelu;

  parallel_for(M, 16, [&](int64_t m0, int64_t m1) {
    std::vector<float> acc(N);
    for (int64_t i = m0; i < m1; ++i) {
      std::fill(acc.begin(), acc.end(), 0.f);
      if (!transA && !transB) {
        for (int64_t k = 0; k < K; ++k) {
          float av = a[i * lda + k];
          const float* brow = b + k * ldb;
          for (int64_t j = 0; j < N; ++j) acc[j] += av * brow[j];
        }
      } else if (!transA && transB) {
        for (int64_t j = 0; j < N; ++j) {
          float s = 0.f;
          const float* ar = a + i * lda;
          const float* br = b + j * ldb;
          for (int64_t k = 0; k < K; ++k) s += ar[k] * br[k];
          acc[j] = s;
        }
      } else if (transA && !transB) {
        for (int64_t k = 0; k < K; ++k) {
          float av = a[k * lda + i];
          const float* brow = b + k * ldb;
          for (int64_t j = 0; j < N; ++j) acc[j] += av * brow[j];
        }
      } else {
        for (int64_t j = 0; j < N; ++j) {
          float s = 0.f;
          for (int64_t k = 0; k < K; ++k) s += a[k * lda + i] * b[j * ldb + k];
          acc[j] = s;
        }
      }
      for (int64_t j = 0; j < N; ++j) {
        float v = alpha * acc[j] + (beta != 0.f ? beta * c[i * N + j] : 0.f);
        if (bias) v += bias[j];
        if (pre) pre[i * N + j] = v;
        if (gelu) v = gelu_fwd_scalar(v);
        c[i * N + j] = v;
      }
    }
  });
}

void gemm_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                  bool transB, float alpha, int64_t batch, int64_t M, int64_t N,
                  int64_t K, int64_t sA, int64_t sB, int64_t sC) {
  const float* a0 = f32p(A);
  const float* b0 = f32p(B);
  float* c0 = f32p(C);
  int64_t lda = transA ? M : K, ldb = transB ? K : N;
  parallel_for(batch, 1, [&](int64_t g0, int64_t g1) {
    for (int64_t g = g0; g < g1; ++g) {
      const float* a = a0 + g * sA;
      const float* b = b0 + g * sB;
      float* c = c0 + g * sC;
      for (int64_t i = 0; i < M; ++i)
        for (int64_t j = 0; j < N; ++j) {
          float s = 0.f;
          for (int64_t k = 0; k < K; ++k) {
            float av = transA ? a[k * lda + i] : a[i * lda + k];
            float bv = transB ? b[j * ldb + k] : b[k * ldb + j];
            s += av * bv;
          }
          c[i * N + j] = alpha * s;
        }
    }
  });
}

// --------------------------------------------------------------- k2 gelu ---
void bias_gelu_fwd(const Tensor& x, const Tensor* bias, Tensor& y, Tensor* save_pre) {
  const float* xp = f32p(x);
  float* yp = f32p(y);
  const float* bp = bias ? f32p(*bias) : nullptr;
  float* zp = save_pre ? f32p(*save_pre) : nullptr;
  int64_t N = x.cols(), total = x.numel();
  parallel_for(total, 4096, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) {
      float v = xp[i] + (bp ? bp[i % N] : 0.f);
      if (zp) zp[i] = v;
      yp[i] = gelu_fwd_scalar(v);
    }
  });
}

void bias_gelu_bwd(const Tensor& dy, const Tensor& z, Tensor& dz, Tensor* dbias) {
  const float* dyp = f32p(dy);
  const float* zp = f32p(z);
  float* dzp = f32p(dz);
  int64_t M = dy.rows(), N = dy.cols();
  parallel_for(M * N, 4096, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) dzp[i] = dyp[i] * gelu_bwd_scalar(zp[i]);
  });
  if (dbias) {
    float* dbp = f32p(*dbias);
    for (int64_t j = 0; j < N; ++j) dbp[j] = 0.f;
    for (int64_t i = 0; i < M; ++i)
      for (int64_t j = 0; j < N; ++j) dbp[j] += dzp[i * N + j];
  }
}

// ----------------------------------------------------------- k3 layernorm ---
void layernorm_fwd(const Tensor& x, const Tensor& gamma, const Tensor& beta,
                   Tensor& y, Tensor& mean, Tensor& rstd, float eps) {
  const float* xp = f32p(x);
  const float* g = f32p(gamma);
  const float* b = f32p(beta);
  float* yp = f32p(y);
  float* mp = f32p(mean);
  float* rp = f32p(rstd);
  int64_t R = x.rows(), N = x.cols();
  parallel_for(R, 8, [&](int64_t r0, int64_t r1) {
    for (int64_t i = r0; i < r1; ++i) {
      const float* row = xp + i * N;
      double s = 0;
      for (int64_t j = 0; j < N; ++j) s += row[j];
      float mu = float(s / N);
      double v = 0;
      for (int64_t j = 0; j < N; ++j) { float d = row[j] - mu; v += double(d) * d; }
      float rs = 1.0f / std::sqrt(float(v / N) + eps);
      mp[i] = mu; rp[i] = rs;
      for (int64_t j = 0; j < N; ++j)
        yp[i * N + j] = (row[j] - mu) * rs * g[j] + b[j];
    }
  });
}

void layernorm_bwd(const Tensor& dy, const Tensor& x, const Tensor& gamma,
                   const Tensor& mean, const Tensor& rstd, Tensor& dx,
                   Tensor& dgamma, Tensor& dbeta, float beta_dx) {
  const float* dyp = f32p(dy);
  const float* xp = f32p(x);
  const float* g = f32p(gamma);
  const float* mp = f32p(mean);
  const float* rp = f32p(rstd);
  float* dxp = f32p(dx);
  float* dgp = f32p(dgamma);
  float* dbp = f32p(dbeta);
  int64_t R = x.rows(), N = x.cols();
  for (int64_t j = 0; j < N; ++j) { dgp[j] = 0.f; dbp[j] = 0.f; }
  for (int64_t i = 0; i < R; ++i) {
    const float* row = xp + i * N;
    const float* dyr = dyp + i * N;
    float mu = mp[i], rs = rp[i];
    float sum_dy = 0.f, sum_dyxh = 0.f;
    for (int64_t j = 0; j < N; ++j) {
      float xh = (row[j] - mu) * rs;
      float dyg = dyr[j] * g[j];
      sum_dy += dyg;
      sum_dyxh += dyg * xh;
      dgp[j] += dyr[j] * xh;
      dbp[j] += dyr[j];
    }
    for (int64_t j = 0; j < N; ++j) {
      float xh = (row[j] - mu) * rs;
      float dyg = dyr[j] * g[j];
      float v = rs * (dyg - sum_dy / N - xh * sum_dyxh / N);
      dxp[i * N + j] = beta_dx * (beta_dx != 0.f ? dxp[i * N + j] : 0.f) + v;
    }
  }
}

// ------------------------------------------------------------- k4 softmax ---
void softmax_fwd(const Tensor& x, Tensor& y, float scale) {
  const float* xp = f32p(x);
  float* yp = f32p(y);
  int64_t R = x.rows(), N = x.cols();
  parallel_for(R, 8, [&](int64_t r0, int64_t r1) {
    for (int64_t i = r0; i < r1; ++i) {
      const float* row = xp + i * N;
      float m = -INFINITY;
      for (int64_t j = 0; j < N; ++j) m = std::max(m, scale * row[j]);
      double s = 0;
      for (int64_t j = 0; j < N; ++j) {
        float e = std::exp(scale * row[j] - m);
        yp[i * N + j] = e;
        s += e;
      }
      float inv = float(1.0 / s);
      for (int64_t j = 0; j < N; ++j) yp[i * N + j] *= inv;
    }
  });
}

void softmax_bwd(const Tensor& dy, const Tensor& y, Tensor& dx, float scale) {
  const float* dyp = f32p(dy);
  const float* yp = f32p(y);
  float* dxp = f32p(dx);
  int64_t R = y.rows(), N = y.cols();
  parallel_for(R, 8, [&](int64_t r0, int64_t r1) {
    for (int64_t i = r0; i < r1; ++i) {
      double dot = 0;
      for (int64_t j = 0; j < N; ++j) dot += double(dyp[i * N + j]) * yp[i * N + j];
      for (int64_t j = 0; j < N; ++j)
        dxp[i * N + j] = scale * yp[i * N + j] * (dyp[i * N + j] - float(dot));
    }
  });
}

// ----------------------------------------------------------- k6/k7 optim ---
void adam_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor& m,
               Tensor& v, float lr, float b1, float b2, float eps, int step,
               float grad_scale, float wd, const Tensor* bc_dev) {
  (void)bc_dev;  // CPU path has no graph capture; `step` is authoritative
  float* mast = f32p(master);
  float* mp = f32p(m);
  float* vp = f32p(v);
  int64_t n = param.numel();
  float bc1 = 1.0f - std::pow(b1, (float)step);
  float bc2 = 1.0f - std::pow(b2, (float)step);
  bool pbf16 = param.dtype() == DType::BF16;
  bool gbf16 = grad.dtype() == DType::BF16;
  uint16_t* pb = (uint16_t*)param.data();
  float* pf = pbf16 ? nullptr : (float*)param.data();
  const uint16_t* gb = (const uint16_t*)grad.data();
  const float* gf = gbf16 ? nullptr : (const float*)grad.data();
  parallel_for(n, 4096, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) {
      float g = (gbf16 ? bf16_to_f32(gb[i]) : gf[i]) * grad_scale;
      if (wd != 0.f) g += wd * mast[i];
      mp[i] = b1 * mp[i] + (1 - b1) * g;
      vp[i] = b2 * vp[i] + (1 - b2) * g * g;
      float mhat = mp[i] / bc1, vhat = vp[i] / bc2;
      mast[i] -= lr * mhat / (std::sqrt(vhat) + eps);
      if (pbf16) pb[i] = f32_to_bf16(mast[i]); else pf[i] = mast[i];
    }
  });
}

void sgd_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor* momentum,
              float lr, float mu, float grad_scale) {
  float* mast = f32p(master);
  float* mom = momentum ? f32p(*momentum) : nullptr;
  int64_t n = param.numel();
  bool pbf16 = param.dtype() == DType::BF16;
  bool gbf16 = grad.dtype() == DType::BF16;
  uint16_t* pb = (uint16_t*)param.data();
  float* pf = pbf16 ? nullptr : (float*)param.data();
  const uint16_t* gb = (const uint16_t*)grad.data();
  const float* gf = gbf16 ? nullptr : (const float*)grad.data();
  parallel_for(n, 4096, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) {
      float g = (gbf16 ? bf16_to_f32(gb[i]) : gf[i]) * grad_scale;
      if (mom) { mom[i] = mu * mom[i] + g; g = mom[i]; }
      mast[i] -= lr * g;
      if (pbf16) pb[i] = f32_to_bf16(mast[i]); else pf[i] = mast[i];
    }
  });
}

// -------------------------------------------------------- k8 elementwise ---
void fill(Tensor& t, float value) {
  int64_t n = t.numel();
  if (t.dtype() == DType::F32) {
    float* p = (float*)t.data();
    for (int64_t i = 0; i < n; ++i) p[i] = value;
  } else {
    uint16_t v = f32_to_bf16(value);
    uint16_t* p = (uint16_t*)t.data();
    for (int64_t i = 0; i < n; ++i) p[i] = v;
  }
}

template <typename F>
static void fill_rng(Tensor& t, F gen) {
  int64_t n = t.numel();
  if (t.dtype() == DType::F32) {
    float* p = (float*)t.data();
    parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) p[i] = gen(i);
    });
  } else {
    uint16_t* p = (uint16_t*)t.data();
    parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) p[i] = f32_to_bf16(gen(i));
    });
  }
}

void rand_uniform(Tensor& t, uint64_t seed, float lo, float hi) {
  fill_rng(t, [=](int64_t i) { return lo + (hi - lo) * rng_uniform01(seed, i); });
}
void rand_normal(Tensor& t, uint64_t seed, float mean, float std) {
  fill_rng(t, [=](int64_t i) { return mean + std * rng_normal(seed, i); });
}

void cast_copy(Tensor& dst, const Tensor& src) {
  NZ_CHECK(dst.numel() == src.numel(), "cast_copy numel mismatch");
  int64_t n = src.numel();
  if (dst.dtype() == src.dtype()) {
    std::memcpy(dst.data(), src.data(), src.bytes());
  } else if (src.dtype() == DType::F32) {
    const float* s = (const float*)src.data();
    uint16_t* d = (uint16_t*)dst.data();
    for (int64_t i = 0; i < n; ++i) d[i] = f32_to_bf16(s[i]);
  } else {
    const uint16_t* s = (const uint16_t*)src.data();
    float* d = (float*)dst.data();
    for (int64_t i = 0; i < n; ++i) d[i] = bf16_to_f32(s[i]);
  }
}

void add_out(Tensor& out, const Tensor& x, const Tensor& y) {
  NZ_CHECK(out.numel() == x.numel() && x.numel() == y.numel(),
           "add_out numel mismatch");
  int64_t n = out.numel();
  if (out.dtype() == DType::F32) {
    float* o = (float*)out.data();
    const float* a = (const float*)x.data();
    const float* b = (const float*)y.data();
    parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) o[i] = a[i] + b[i];
    });
  } else {
    uint16_t* o = (uint16_t*)out.data();
    const uint16_t* a = (const uint16_t*)x.data();
    const uint16_t* b = (const uint16_t*)y.data();
    parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i)
        o[i] = f32_to_bf16(bf16_to_f32(a[i]) + bf16_to_f32(b[i]));
    });
  }
}

void add_inplace(Tensor& dst, const Tensor& src, float alpha) {
  NZ_CHECK(dst.numel() == src.numel(), "add numel mismatch");
  int64_t n = dst.numel();
  if (dst.dtype() == DType::F32 && src.dtype() == DType::F32) {
    float* d = (float*)dst.data();
    const float* s = (const float*)src.data();
    parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
      for (int64_t i = lo; i < hi; ++i) d[i] += alpha * s[i];
    });
  } else {
    for (int64_t i = 0; i < n; ++i) {
      auto get = [&](const Tensor& t) {
        return t.dtype() == DType::F32 ? ((const float*)t.data())[i]
                                       : bf16_to_f32(((const uint16_t*)t.data())[i]);
      };
      float v = get(dst) + alpha * get(src);
      if (dst.dtype() == DType::F32) ((float*)dst.data())[i] = v;
      else ((uint16_t*)dst.data())[i] = f32_to_bf16(v);
    }
  }
}

void scale_inplace(Tensor& dst, float alpha) {
  int64_t n = dst.numel();
  if (dst.dtype() == DType::F32) {
    float* d = (float*)dst.data();
    for (int64_t i = 0; i < n; ++i) d[i] *= alpha;
  } else {
    uint16_t* d = (uint16_t*)dst.data();
    for (int64_t i = 0; i < n; ++i) d[i] = f32_to_bf16(bf16_to_f32(d[i]) * alpha);
  }
}

void transpose2d(const Tensor& src, Tensor& dst) {
  int64_t R = src.rows(), C = src.cols();
  NZ_CHECK(dst.rows() == C && dst.cols() == R, "transpose shape mismatch");
  size_t es = dtype_size(src.dtype());
  const char* s = (const char*)src.data();
  char* d = (char*)dst.data();
  for (int64_t i = 0; i < R; ++i)
    for (int64_t j = 0; j < C; ++j)
      std::memcpy(d + (j * R + i) * es, s + (i * C + j) * es, es);
}

void dropout_fwd(const Tensor& x, Tensor& y, Tensor& mask, float p, uint64_t seed) {
  int64_t n = x.numel();
  float keep = 1.0f - p, inv = keep > 0 ? 1.0f / keep : 0.f;
  auto get = [&](const Tensor& t, int64_t i) {
    return t.dtype() == DType::F32 ? ((const float*)t.data())[i]
                                   : bf16_to_f32(((const uint16_t*)t.data())[i]);
  };
  auto put = [&](Tensor& t, int64_t i, float v) {
    if (t.dtype() == DType::F32) ((float*)t.data())[i] = v;
    else ((uint16_t*)t.data())[i] = f32_to_bf16(v);
  };
  for (int64_t i = 0; i < n; ++i) {
    float m = rng_uniform01(seed, (uint64_t)i) >= p ? 1.0f : 0.0f;
    put(mask, i, m);
    put(y, i, get(x, i) * m * inv);
  }
}
void dropout_bwd(const Tensor& dy, const Tensor& mask, Tensor& dx, float p) {
  int64_t n = dy.numel();
  float keep = 1.0f - p, inv = keep > 0 ? 1.0f / keep : 0.f;
  auto get = [&](const Tensor& t, int64_t i) {
    return t.dtype() == DType::F32 ? ((const float*)t.data())[i]
                                   : bf16_to_f32(((const uint16_t*)t.data())[i]);
  };
  for (int64_t i = 0; i < n; ++i) {
    float v = get(dy, i) * get(mask, i) * inv;
    if (dx.dtype() == DType::F32) ((float*)dx.data())[i] = v;
    else ((uint16_t*)dx.data())[i] = f32_to_bf16(v);
  }
}

// ------------------------------------------------------------- k9 losses ---
void mse_loss(const Tensor& y, const Tensor& target, Tensor& loss_out, Tensor* dy,
              float grad_scale) {
  int64_t n = y.numel();
  auto get = [&](const Tensor& t, int64_t i) {
    return t.dtype() == DType::F32 ? ((const float*)t.data())[i]
                                   : bf16_to_f32(((const uint16_t*)t.data())[i]);
  };
  double s = 0;
  for (int64_t i = 0; i < n; ++i) {
    float d = get(y, i) - get(target, i);
    s += double(d) * d;
    if (dy) {
      float g = 2.0f * d / float(n) * grad_scale;
      if (dy->dtype() == DType::F32) ((float*)dy->data())[i] = g;
      else ((uint16_t*)dy->data())[i] = f32_to_bf16(g);
    }
  }
  ((float*)loss_out.data())[0] = float(s / n);
}

void ce_loss(const Tensor& logits, const Tensor& labels, Tensor& loss_out,
             Tensor* dlogits, float grad_scale) {
  const float* lp = f32p(logits);
  const float* lab = f32p(labels);
  float* dl = dlogits ? f32p(*dlogits) : nullptr;
  int64_t R = logits.rows(), N = logits.cols();
  double total = 0;
  for (int64_t i = 0; i < R; ++i) {
    const float* row = lp + i * N;
    float m = -INFINITY;
    for (int64_t j = 0; j < N; ++j) m = std::max(m, row[j]);
    double z = 0;
    for (int64_t j = 0; j < N; ++j) z += std::exp(row[j] - m);
    int64_t y = (int64_t)lab[i];
    total += -(row[y] - m - std::log(z));
    if (dl) {
      for (int64_t j = 0; j < N; ++j) {
        float p = float(std::exp(row[j] - m) / z);
        dl[i * N + j] = (p - (j == y ? 1.f : 0.f)) / float(R) * grad_scale;
      }
    }
  }
  ((float*)loss_out.data())[0] = float(total / R);
}

// --------------------------------------------------------- k10 reductions ---
void reduce_sum(const Tensor& src, Tensor& out) {
  int64_t n = src.numel();
  double s = 0;
  if (src.dtype() == DType::F32) {
    const float* p = (const float*)src.data();
    for (int64_t i = 0; i < n; ++i) s += p[i];
  } else {
    const uint16_t* p = (const uint16_t*)src.data();
    for (int64_t i = 0; i < n; ++i) s += bf16_to_f32(p[i]);
  }
  ((float*)out.data())[0] = float(s);
}

void l2norm_sq(const Tensor& src, Tensor& out) {
  int64_t n = src.numel();
  double s = 0;
  if (src.dtype() == DType::F32) {
    const float* p = (const float*)src.data();
    for (int64_t i = 0; i < n; ++i) s += double(p[i]) * p[i];
  } else {
    const uint16_t* p = (const uint16_t*)src.data();
    for (int64_t i = 0; i < n; ++i) { double v = bf16_to_f32(p[i]); s += v * v; }
  }
  ((float*)out.data())[0] = float(s);
}

}  // namespace cpu
}  // namespace nz

namespace nz { namespace cpu {

void colsum(const Tensor& src, Tensor& dst, float beta) {
  int64_t M = src.rows(), N = src.cols();
  NZ_CHECK(dst.numel() == N, "colsum dst size");
  std::vector<double> acc(N, 0.0);
  auto get = [&](int64_t i) {
    return src.dtype() == DType::F32 ? ((const float*)src.data())[i]
                                     : bf16_to_f32(((const uint16_t*)src.data())[i]);
  };
  for (int64_t i = 0; i < M; ++i)
    for (int64_t j = 0; j < N; ++j) acc[j] += get(i * N + j);
  for (int64_t j = 0; j < N; ++j) {
    float prev = 0.f;
    if (beta != 0.f)
      prev = dst.dtype() == DType::F32 ? ((float*)dst.data())[j]
                                       : bf16_to_f32(((uint16_t*)dst.data())[j]);
    float v = beta * prev + float(acc[j]);
    if (dst.dtype() == DType::F32) ((float*)dst.data())[j] = v;
    else ((uint16_t*)dst.data())[j] = f32_to_bf16(v);
  }
}

// elementwise index remap helpers for attention layouts (dtype-agnostic memmove)
template <typename CopyFn>
static void remap(int64_t n, size_t es, CopyFn fn) {
  parallel_for(n, 8192, [&](int64_t lo, int64_t hi) {
    for (int64_t i = lo; i < hi; ++i) fn(i);
  });
  (void)es;
}

void attn_split_qkv(const Tensor& qkv, Tensor& q, Tensor& k, Tensor& v,
                    int64_t B, int64_t S, int64_t heads, int64_t dh) {
  size_t es = dtype_size(qkv.dtype());
  const char* src = (const char*)qkv.data();
  char* dsts[3] = {(char*)q.data(), (char*)k.data(), (char*)v.data()};
  int64_t H = heads * dh;
  remap(B * S * 3 * H, es, [&](int64_t i) {
    int64_t d = i % dh, h = (i / dh) % heads, which = (i / (dh * heads)) % 3;
    int64_t s = (i / (3 * H)) % S, b = i / (3 * H * S);
    int64_t dst_idx = ((b * heads + h) * S + s) * dh + d;
    std::memcpy(dsts[which] + dst_idx * es, src + i * es, es);
  });
}

void attn_split_qkv_bwd(const Tensor& dq, const Tensor& dk, const Tensor& dv,
                        Tensor& dqkv, int64_t B, int64_t S, int64_t heads, int64_t dh) {
  size_t es = dtype_size(dqkv.dtype());
  char* dst = (char*)dqkv.data();
  const char* srcs[3] = {(const char*)dq.data(), (const char*)dk.data(), (const char*)dv.data()};
  int64_t H = heads * dh;
  remap(B * S * 3 * H, es, [&](int64_t i) {
    int64_t d = i % dh, h = (i / dh) % heads, which = (i / (dh * heads)) % 3;
    int64_t s = (i / (3 * H)) % S, b = i / (3 * H * S);
    int64_t src_idx = ((b * heads + h) * S + s) * dh + d;
    std::memcpy(dst + i * es, srcs[which] + src_idx * es, es);
  });
}

void attn_merge(const Tensor& ctx, Tensor& out, int64_t B, int64_t S,
                int64_t heads, int64_t dh) {
  size_t es = dtype_size(ctx.dtype());
  const char* src = (const char*)ctx.data();
  char* dst = (char*)out.data();
  int64_t H = heads * dh;
  remap(B * S * H, es, [&](int64_t i) {  // i indexes out [B,S,H]
    int64_t d = i % dh, h = (i / dh) % heads, s = (i / H) % S, b = i / (H * S);
    int64_t src_idx = ((b * heads + h) * S + s) * dh + d;
    std::memcpy(dst + i * es, src + src_idx * es, es);
  });
}

void attn_merge_bwd(const Tensor& dout, Tensor& dctx, int64_t B, int64_t S,
                    int64_t heads, int64_t dh) {
  size_t es = dtype_size(dout.dtype());
  const char* src = (const char*)dout.data();
  char* dst = (char*)dctx.data();
  int64_t H = heads * dh;
  remap(B * S * H, es, [&](int64_t i) {  // i indexes dout [B,S,H]
    int64_t d = i % dh, h = (i / dh) % heads, s = (i / H) % S, b = i / (H * S);
    int64_t dst_idx = ((b * heads + h) * S + s) * dh + d;
    std::memcpy(dst + dst_idx * es, src + i * es, es);
  });
}

}}
