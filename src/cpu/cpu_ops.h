// CPU f32 backend: portable reference implementation.
// Runs config C10 (2-layer 128-dim MLP, no GPU) and is the golden oracle
// the HIP kernels are tested against (test tier T3).
#pragma once
#include <functional>

#include "src/core/ops.h"

namespace nz {
namespace cpu {

void gemm(const Tensor& A, const Tensor& B, Tensor& C, bool transA, bool transB,
          float alpha, float beta, const ops::GemmEpilogue* epi);
void gemm_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                  bool transB, float alpha, int64_t batch, int64_t M, int64_t N,
                  int64_t K, int64_t sA, int64_t sB, int64_t sC);
void bias_gelu_fwd(const Tensor& x, const Tensor* bias, Tensor& y, Tensor* save_pre);
void bias_gelu_bwd(const Tensor& dy, const Tensor& z, Tensor& dz, Tensor* dbias);
void layernorm_fwd(const Tensor& x, const Tensor& gamma, const Tensor& beta,
                   Tensor& y, Tensor& mean, Tensor& rstd, float eps);
void layernorm_bwd(const Tensor& dy, const Tensor& x, const Tensor& gamma,
                   const Tensor& mean, const Tensor& rstd, Tensor& dx,
                   Tensor& dgamma, Tensor& dbeta, float beta_dx);
void softmax_fwd(const Tensor& x, Tensor& y, float scale);
void softmax_bwd(const Tensor& dy, const Tensor& y, Tensor& dx, float scale);
void adam_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor& m,
               Tensor& v, float lr, float b1, float b2, float eps, int step,
               float grad_scale, float wd, const Tensor* bc_dev = nullptr);
void sgd_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor* momentum,
              float lr, float mu, float grad_scale);
void fill(Tensor& t, float value);
void rand_uniform(Tensor& t, uint64_t seed, float lo, float hi);
void rand_normal(Tensor& t, uint64_t seed, float mean, float std);
void cast_copy(Tensor& dst, const Tensor& src);
void add_inplace(Tensor& dst, const Tensor& src, float alpha);
void add_out(Tensor& out, const Tensor& x, const Tensor& y);
void scale_inplace(Tensor& dst, float alpha);
void transpose2d(const Tensor& src, Tensor& dst);
void dropout_fwd(const Tensor& x, Tensor& y, Tensor& mask, float p, uint64_t seed);
void dropout_bwd(const Tensor& dy, const Tensor& mask, Tensor& dx, float p);
void mse_loss(const Tensor& y, const Tensor& target, Tensor& loss_out, Tensor* dy,
              float grad_scale);
void ce_loss(const Tensor& logits, const Tensor& labels, Tensor& loss_out,
             Tensor* dlogits, float grad_scale);
void reduce_sum(const Tensor& src, Tensor& out);
void l2norm_sq(const Tensor& src, Tensor& out);

// Simple fork-join parallel loop over [0, n) in contiguous chunks.
void parallel_for(int64_t n, int64_t grain, const std::function<void(int64_t, int64_t)>& fn);

}  // namespace cpu
}  // namespace nz

// appended: colsum + attention layout ops (decls mirror core/ops.h)
namespace nz { namespace cpu {
void colsum(const Tensor& src, Tensor& dst, float beta);
void attn_split_qkv(const Tensor& qkv, Tensor& q, Tensor& k, Tensor& v,
                    int64_t B, int64_t S, int64_t heads, int64_t dh);
void attn_split_qkv_bwd(const Tensor& dq, const Tensor& dk, const Tensor& dv,
                        Tensor& dqkv, int64_t B, int64_t S, int64_t heads, int64_t dh);
void attn_merge(const Tensor& ctx, Tensor& out, int64_t B, int64_t S,
                int64_t heads, int64_t dh);
void attn_merge_bwd(const Tensor& dout, Tensor& dctx, int64_t B, int64_t S,
                    int64_t heads, int64_t dh);
}}
