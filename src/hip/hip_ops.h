// HIP/CDNA4 backend launchers (SURVEY.md §2.2 kernel inventory K1..K10).
// Implementations: src/hip/elementwise.hip (K2,K3,K4,K6..K10), src/hip/gemm.hip (K1).
// All launches go on the tensor's context compute stream; MFMA/LDS tilings
// are chosen for gfx950 (MI355X) per /opt/skills/guides/cdna_hip_programming.md.
#pragma once
#include "src/core/ops.h"

namespace nz {
namespace hip {

void gemm(const Tensor& A, const Tensor& B, Tensor& C, bool transA, bool transB,
          float alpha, float beta, const ops::GemmEpilogue* epi);
void gemm_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                  bool transB, float alpha, int64_t batch, int64_t M, int64_t N,
                  int64_t K, int64_t sA, int64_t sB, int64_t sC);
bool gemm_can_fuse_dgelu(int64_t M, int64_t N, int64_t K, bool transA,
                         bool transB);
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
// batched fast transpose (R,C % 64): dst[b][c][r] = src[b][r][c]
void transpose2d_batched(const Tensor& src, Tensor& dst, int64_t batch,
                         int64_t R, int64_t C, int64_t sSrc, int64_t sDst);
void dropout_fwd(const Tensor& x, Tensor& y, Tensor& mask, float p, uint64_t seed);
void dropout_bwd(const Tensor& dy, const Tensor& mask, Tensor& dx, float p);
void mse_loss(const Tensor& y, const Tensor& target, Tensor& loss_out, Tensor* dy,
              float grad_scale);
void ce_loss(const Tensor& logits, const Tensor& labels, Tensor& loss_out,
             Tensor* dlogits, float grad_scale);
void reduce_sum(const Tensor& src, Tensor& out);
void l2norm_sq(const Tensor& src, Tensor& out);

}  // namespace hip
}  // namespace nz

// f32 GEMM on the exact-f32 MFMA (src/hip/gemm_f32.hip)
namespace nz { namespace hip {
void gemm_f32(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
              bool transB, float alpha, float beta, const ops::GemmEpilogue* epi);
void gemm_f32_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                      bool transB, float alpha, int64_t batch, int64_t M,
                      int64_t N, int64_t K, int64_t sA, int64_t sB, int64_t sC);
}}

// fused attention forward (src/hip/attention.hip)
namespace nz { namespace hip {
bool attn_fused_fwd_supported(int64_t S, int64_t dh);
bool attn_fused_fwd512_supported(int64_t S, int64_t dh);
void attn_fused_fwd512_packed(const Tensor& qkv, Tensor& probs, Tensor& out,
                              int64_t B, int64_t S, int64_t heads, int64_t dh,
                              float scale);
void attn_fused_fwd512(const Tensor& q, const Tensor& k, const Tensor& v,
                       Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                       float scale);
void attn_fused_bwd512_packed(const Tensor& qkv, const Tensor& probs,
                              const Tensor& dmerged, Tensor& dqkv, int64_t B,
                              int64_t S, int64_t heads, int64_t dh,
                              float scale);
void attn_fused_fwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                    float scale);
void attn_fused_bwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    const Tensor& probs, const Tensor& dout, Tensor& dq,
                    Tensor& dk, Tensor& dv, int64_t batch_heads, float scale);
void attn_fused_fwd_packed(const Tensor& qkv, Tensor& probs, Tensor& out,
                           int64_t B, int64_t S, int64_t heads, int64_t dh,
                           float scale);
void attn_fused_bwd_packed(const Tensor& qkv, const Tensor& probs,
                           const Tensor& dmerged, Tensor& dqkv, int64_t B,
                           int64_t S, int64_t heads, int64_t dh, float scale);
}}

// appended: colsum + attention layout ops (decls mirror core/ops.h)
namespace nz { namespace hip {
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
