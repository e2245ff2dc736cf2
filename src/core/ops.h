// Op surface of the tensor engine. Single dispatch rule:
// a tensor lives on exactly one device; each op dispatches on device() to
// exactly one of two backends — CPU f32 (config C10 + golden oracle) or the
// hand-written HIP/CDNA4 kernel set. No registry,
// no multi-backend dispatch, no Triton, no CUDA path.
#pragma once

#include "src/core/tensor.h"

namespace nz {
namespace ops {

// ---- K1: GEMM. C = alpha * op(A)·op(B) + beta * C, f32 accumulate.
// Optional fused epilogue (forward path): t = acc + bias[col];
// save_pre <- t (pre-activation, needed by K2 backward); C <- gelu(t).
struct GemmEpilogue {
  const Tensor* bias = nullptr;   // [N] or null
  bool gelu = false;
  Tensor* save_pre = nullptr;     // [M,N] or null
  // backward-path fusion: C = (alpha*acc) * gelu'(dgelu_pre[idx]) [+ beta*C
  // + bias]. HIP-only, and only on routes gemm_can_fuse_dgelu approves —
  // callers must gate on that predicate (hip::gemm fails loudly otherwise).
  const Tensor* dgelu_pre = nullptr;  // [M,N] pre-activation, or null
};
void gemm(const Tensor& A, const Tensor& B, Tensor& C, bool transA, bool transB,
          float alpha, float beta, const GemmEpilogue* epi = nullptr);
// True iff the HIP dispatch for this shape lands on a kernel implementing
// the fused GELU-derivative epilogue (false on CPU).
bool gemm_can_fuse_dgelu(const Tensor& like, int64_t M, int64_t N, int64_t K,
                         bool transA, bool transB);

// Strided-batched GEMM (attention: QK^T and P·V). Strides in elements.
void gemm_batched(const Tensor& A, const Tensor& B, Tensor& C, bool transA,
                  bool transB, float alpha, int64_t batch, int64_t M, int64_t N,
                  int64_t K, int64_t strideA, int64_t strideB, int64_t strideC);

// ---- K2: bias+GELU backward. dz = dy * gelu'(z); dbias[n] = sum_m dz[m,n].
void bias_gelu_bwd(const Tensor& dy, const Tensor& z, Tensor& dz, Tensor* dbias);
// Standalone forward variant (unfused clarity path): y = gelu(x + bias).
void bias_gelu_fwd(const Tensor& x, const Tensor* bias, Tensor& y, Tensor* save_pre);

// ---- K3: LayerNorm over the last dim.
void layernorm_fwd(const Tensor& x, const Tensor& gamma, const Tensor& beta,
                   Tensor& y, Tensor& mean, Tensor& rstd, float eps);
void layernorm_bwd(const Tensor& dy, const Tensor& x, const Tensor& gamma,
                   const Tensor& mean, const Tensor& rstd, Tensor& dx,
                   Tensor& dgamma, Tensor& dbeta, float beta_dx);

// ---- K4: row-wise softmax over the last dim (online max). y = softmax(scale*x).
void softmax_fwd(const Tensor& x, Tensor& y, float scale);
// dx = scale * y * (dy - rowsum(dy*y))
void softmax_bwd(const Tensor& dy, const Tensor& y, Tensor& dx, float scale);

// ---- K6/K7: optimizer steps on flat buckets.
// param bf16 (HIP) / f32 (CPU); master/m/v f32. grad_scale multiplies the
// raw gradient (1/world_size fused here for DP averaging).
// bc_dev (optional, HIP): f32[2] device tensor holding the bias-correction
// pair {1-b1^t, 1-b2^t}; when given it overrides `step` so the launch can
// be captured in a hipGraph and replayed with fresh corrections.
void adam_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor& m,
               Tensor& v, float lr, float beta1, float beta2, float eps,
               int step, float grad_scale, float weight_decay,
               const Tensor* bc_dev = nullptr);
void sgd_step(Tensor& param, const Tensor& grad, Tensor& master, Tensor* momentum,
              float lr, float mu, float grad_scale);

// ---- K8: elementwise family.
void fill(Tensor& t, float value);
void rand_uniform(Tensor& t, uint64_t seed, float lo, float hi);
void rand_normal(Tensor& t, uint64_t seed, float mean, float std);
void cast_copy(Tensor& dst, const Tensor& src);       // dtype conversion or plain copy
void add_inplace(Tensor& dst, const Tensor& src, float alpha);  // dst += alpha*src
void add_out(Tensor& out, const Tensor& x, const Tensor& y);     // out = x + y
void scale_inplace(Tensor& dst, float alpha);
void transpose2d(const Tensor& src, Tensor& dst);     // dst[j,i] = src[i,j]

// Dropout (K8 family; off by default in the contract models). Forward
// writes y = x * mask / (1-p) and the bf16/f32 mask itself (counter-based
// philox-style RNG keyed by seed — deterministic, replayable); backward is
// dx = dy * mask / (1-p) via the same call with the saved mask.
void dropout_fwd(const Tensor& x, Tensor& y, Tensor& mask, float p, uint64_t seed);
void dropout_bwd(const Tensor& dy, const Tensor& mask, Tensor& dx, float p);

// ---- K9: losses. loss_out: f32 scalar tensor on device. dy optional.
void mse_loss(const Tensor& y, const Tensor& target, Tensor& loss_out,
              Tensor* dy, float grad_scale);
// Cross-entropy over last dim; labels: f32 tensor holding integral class ids.
void ce_loss(const Tensor& logits, const Tensor& labels, Tensor& loss_out,
             Tensor* dlogits, float grad_scale);

// ---- K10: reductions (debug/grad-norm).
void reduce_sum(const Tensor& src, Tensor& out_scalar);
void l2norm_sq(const Tensor& src, Tensor& out_scalar);

// Column sum over rows: dst[n] = beta*dst[n] + sum_m src[m,n]  (bias grads).
void colsum(const Tensor& src, Tensor& dst, float beta);

// ---- attention layout ops (K8 family, memory-movement only).
// qkv: [B, S, 3*H] with H = heads*dh -> q,k,v each [B, heads, S, dh].
void attn_split_qkv(const Tensor& qkv, Tensor& q, Tensor& k, Tensor& v,
                    int64_t B, int64_t S, int64_t heads, int64_t dh);
// inverse: dq,dk,dv [B,heads,S,dh] -> dqkv [B, S, 3*H]
void attn_split_qkv_bwd(const Tensor& dq, const Tensor& dk, const Tensor& dv,
                        Tensor& dqkv, int64_t B, int64_t S, int64_t heads, int64_t dh);
// Fused attention forward (K5): probs = softmax(scale * q.k^T) saved for
// backward, ctx_out = probs.v. Single-kernel fusion on HIP for the BERT
// contract shape (S=128, dh=64); composed from K1 batched + K4 otherwise.
void attn_fused_fwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    Tensor& probs, Tensor& ctx_out, int64_t batch_heads,
                    int64_t S, int64_t dh, float scale);

// Fused attention backward (K5): given saved probs and upstream dctx,
// produces dq/dk/dv. Single kernel on HIP for the contract shape
// (src/hip/attention.hip attn_bwd_fused_k); composed from K1 batched +
// K4 softmax_bwd otherwise (and as the CPU oracle).
void attn_fused_bwd(const Tensor& q, const Tensor& k, const Tensor& v,
                    const Tensor& probs, const Tensor& dctx, Tensor& dq,
                    Tensor& dk, Tensor& dv, int64_t batch_heads, int64_t S,
                    int64_t dh, float scale);

// Packed-layout attention (the module's hot path): consumes the qkv
// projection [B, S, 3H] and produces merged [B, S, H] directly — on HIP at
// the contract shape the fused kernels absorb the attn_split / attn_merge
// layout passes; elsewhere this composes split + fused + merge.
void attn_fwd_packed(const Tensor& qkv, Tensor& probs, Tensor& out, int64_t B,
                     int64_t S, int64_t heads, int64_t dh, float scale);
void attn_bwd_packed(const Tensor& qkv, const Tensor& probs,
                     const Tensor& dmerged, Tensor& dqkv, int64_t B, int64_t S,
                     int64_t heads, int64_t dh, float scale);

// ctx [B,heads,S,dh] -> out [B, S, H]
void attn_merge(const Tensor& ctx, Tensor& out, int64_t B, int64_t S,
                int64_t heads, int64_t dh);
// dout [B,S,H] -> dctx [B,heads,S,dh]
void attn_merge_bwd(const Tensor& dout, Tensor& dctx, int64_t B, int64_t S,
                    int64_t heads, int64_t dh);

}  // namespace ops
}  // namespace nz
