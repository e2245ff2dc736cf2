#include "src/nn/modules.h"

#include <cmath>

namespace nz {

static DType param_dtype(Context* ctx) {
  return ctx->device() == Device::HIP ? DType::BF16 : DType::F32;
}

// ---------------------------------------------------------------- Linear ---
Linear::Linear(Context* ctx, std::string name, int64_t in, int64_t out,
               bool bias, bool gelu, uint64_t seed)
    : has_bias(bias), fuse_gelu(gelu), in_dim(in), out_dim(out), ctx_(ctx) {
  DType pd = param_dtype(ctx);
  w.name = name + ".w";
  w.data = Tensor(ctx, pd, Shape{in, out});
  float std = std::sqrt(2.0f / float(in + out));  // Xavier
  ops::rand_normal(w.data, seed, 0.f, std);
  w.grad = Tensor(ctx, pd, Shape{in, out});
  if (has_bias) {
    b.name = name + ".b";
    b.data = Tensor(ctx, pd, Shape{out});
    ops::fill(b.data, 0.f);
    b.grad = Tensor(ctx, pd, Shape{out});
  }
}

void Linear::collect_params(std::vector<Param*>& out) {
  out.push_back(&w);
  if (has_bias) out.push_back(&b);
}

// NEZHA_FUSE_DGELU=0 disables folding gelu' into downstream dgrad GEMMs
// (useful for kernel-variant A/B runs where the route would not support it).
static bool fuse_dgelu_enabled() {
  static const bool on = [] {
    const char* v = getenv("NEZHA_FUSE_DGELU");
    return !v || atoi(v) != 0;
  }();
  return on;
}

VarPtr Linear::forward(Tape& tape, VarPtr x) {
  int64_t M = x->value.rows();
  DType pd = param_dtype(ctx_);
  auto y = make_var(Tensor(ctx_, pd, Shape{M, out_dim}));
  Tensor z;  // pre-activation, saved for GELU backward
  ops::GemmEpilogue epi;
  epi.bias = has_bias ? &b.data : nullptr;
  epi.gelu = fuse_gelu;
  if (fuse_gelu) {
    z = Tensor(ctx_, pd, Shape{M, out_dim});
    epi.save_pre = &z;
  }
  ops::gemm(x->value, w.data, y->value, false, false, 1.0f, 0.0f, &epi);
  if (fuse_gelu) y->dgelu_pre_t = z;  // downstream dgrad may fuse gelu' in

  tape.push([this, x, y, z]() {
    Context* ctx = ctx_;
    DType pd = param_dtype(ctx);
    Tensor& dy = y->grad_for_read();
    Tensor dz;
    if (fuse_gelu && y->dgelu_applied) {
      // the downstream consumer's dgrad GEMM already multiplied gelu'(z)
      // into dy in its epilogue; it must have been the only contributor.
      NZ_CHECK(y->grad_writes == 1,
               "dgelu fusion with multiple grad writers on " + w.name);
      dz = dy;
      if (has_bias) {
        NZ_CHECK(!b.grad_init, "param " + b.name + " used twice");
        ops::colsum(dy, b.grad, b.grad_beta());
        b.grad_init = true;
        b.mark_ready();
      }
    } else if (fuse_gelu) {
      dz = Tensor(ctx, pd, dy.shape());
      NZ_CHECK(!has_bias || !b.grad_init, "param " + b.name + " used twice");
      ops::bias_gelu_bwd(dy, z, dz, has_bias ? &b.grad : nullptr);
      if (has_bias) { b.grad_init = true; b.mark_ready(); }
    } else {
      dz = dy;
      if (has_bias) {
        ops::colsum(dy, b.grad, b.grad_beta());
        b.mark_ready();
      }
    }
    // wgrad: dW[in,out] = x^T . dz
    ops::gemm(x->value, dz, w.grad, true, false, 1.0f, w.grad_beta(), nullptr);
    w.mark_ready();
    // dgrad: dx[M,in] = dz . W^T — if x is a fused-GELU output and the
    // route supports it, fold gelu'(pre) into this GEMM's epilogue
    // (saves the separate bias_gelu_bwd read+write pass over dx).
    if (x->needs_grad) {
      x->ensure_grad();
      float bx = x->grad_beta();
      ops::GemmEpilogue epi2;
      // never fuse into an aliased (residual-trunk) grad buffer: its
      // accumulation contract is outside this Linear's view
      bool fuse_dg = fuse_dgelu_enabled() && bx == 0.0f &&
                     x->dgelu_pre_t.defined() && !x->dgelu_applied &&
                     !x->alias_guard &&
                     ops::gemm_can_fuse_dgelu(dz, dz.rows(), in_dim, out_dim,
                                              false, true);
      if (fuse_dg) epi2.dgelu_pre = &x->dgelu_pre_t;
      ops::gemm(dz, w.data, x->grad, false, true, 1.0f, bx,
                fuse_dg ? &epi2 : nullptr);
      if (fuse_dg) x->dgelu_applied = true;
    }
  });
  return y;
}

// ------------------------------------------------------------- LayerNorm ---
LayerNorm::LayerNorm(Context* ctx, std::string name, int64_t d)
    : dim(d), ctx_(ctx) {
  DType pd = param_dtype(ctx);
  gamma.name = name + ".gamma";
  gamma.data = Tensor(ctx, pd, Shape{d});
  ops::fill(gamma.data, 1.f);
  gamma.grad = Tensor(ctx, pd, Shape{d});
  beta.name = name + ".beta";
  beta.data = Tensor(ctx, pd, Shape{d});
  ops::fill(beta.data, 0.f);
  beta.grad = Tensor(ctx, pd, Shape{d});
}

void LayerNorm::collect_params(std::vector<Param*>& out) {
  out.push_back(&gamma);
  out.push_back(&beta);
}

VarPtr LayerNorm::forward(Tape& tape, VarPtr x) {
  int64_t R = x->value.rows();
  auto y = make_var(Tensor(ctx_, x->value.dtype(), x->value.shape()));
  Tensor mean(ctx_, DType::F32, Shape{R});
  Tensor rstd(ctx_, DType::F32, Shape{R});
  ops::layernorm_fwd(x->value, gamma.data, beta.data, y->value, mean, rstd, 1e-5f);

  tape.push([this, x, y, mean, rstd]() {
    Tensor& dy = y->grad_for_read();
    NZ_CHECK(!gamma.grad_init && !beta.grad_init, "LN params used twice");
    if (x->needs_grad) {
      x->ensure_grad();
      ops::layernorm_bwd(dy, x->value, gamma.data, mean, rstd, x->grad,
                         gamma.grad, beta.grad, x->grad_beta());
    } else {
      Tensor scratch_dx(ctx_, x->value.dtype(), x->value.shape());
      ops::layernorm_bwd(dy, x->value, gamma.data, mean, rstd, scratch_dx,
                         gamma.grad, beta.grad, 0.f);
    }
    gamma.grad_init = beta.grad_init = true;
    gamma.mark_ready();
    beta.mark_ready();
  });
  return y;
}

// --------------------------------------------------- MultiHeadAttention ---
MultiHeadAttention::MultiHeadAttention(Context* ctx, std::string name,
                                       int64_t hidden_, int64_t heads_,
                                       int64_t seq_, int64_t batch_, uint64_t seed)
    : qkv_proj(ctx, name + ".qkv", hidden_, 3 * hidden_, true, false, seed),
      out_proj(ctx, name + ".out", hidden_, hidden_, true, false, seed + 1),
      hidden(hidden_), heads(heads_), dh(hidden_ / heads_), seq(seq_),
      batch(batch_), ctx_(ctx) {
  NZ_CHECK(hidden % heads == 0, "hidden not divisible by heads");
}

void MultiHeadAttention::collect_params(std::vector<Param*>& out) {
  qkv_proj.collect_params(out);
  out_proj.collect_params(out);
}

VarPtr MultiHeadAttention::forward(Tape& tape, VarPtr x) {
  // x: [B*S, H]
  DType pd = x->value.dtype();
  int64_t B = batch, S = seq, Hh = heads, D = dh;
  float scale = 1.0f / std::sqrt(float(D));
  VarPtr qkv = qkv_proj.forward(tape, x);  // [B*S, 3H]

  Tensor probs(ctx_, pd, Shape{B * Hh * S, S});
  auto merged = make_var(Tensor(ctx_, pd, Shape{B * S, hidden}));
  ops::attn_fwd_packed(qkv->value, probs, merged->value, B, S, Hh, D, scale);

  tape.push([qkv, merged, probs, B, S, Hh, D, scale]() {
    Tensor& dmerged = merged->grad_for_read();
    qkv->ensure_grad();
    NZ_CHECK(!qkv->grad_init, "qkv grad written twice");
    ops::attn_bwd_packed(qkv->value, probs, dmerged, qkv->grad, B, S, Hh, D,
                         scale);
    qkv->grad_init = true;
  });
  return out_proj.forward(tape, merged);
}

// ------------------------------------------------- standalone activations ---
VarPtr Gelu::forward(Tape& tape, VarPtr x) {
  auto y = make_var(Tensor(ctx_, x->value.dtype(), x->value.shape()));
  Tensor z = x->value;  // pre-activation IS the input (no bias)
  ops::bias_gelu_fwd(x->value, nullptr, y->value, nullptr);
  tape.push([this, x, y, z]() {
    Tensor& dy = y->grad_for_read();
    if (!x->needs_grad) return;
    x->ensure_grad();
    if (!x->grad_init) {
      ops::bias_gelu_bwd(dy, z, x->grad, nullptr);
      x->grad_init = true;
    } else {
      Tensor dz(ctx_, z.dtype(), z.shape());
      ops::bias_gelu_bwd(dy, z, dz, nullptr);
      x->note_grad_mutation();
      ops::add_inplace(x->grad, dz, 1.0f);
    }
  });
  return y;
}

VarPtr Softmax::forward(Tape& tape, VarPtr x) {
  auto y = make_var(Tensor(ctx_, x->value.dtype(), x->value.shape()));
  ops::softmax_fwd(x->value, y->value, scale_);
  float sc = scale_;
  Context* ctx = ctx_;
  tape.push([ctx, sc, x, y]() {
    Tensor& dy = y->grad_for_read();
    if (!x->needs_grad) return;
    x->ensure_grad();
    if (!x->grad_init) {
      ops::softmax_bwd(dy, y->value, x->grad, sc);
      x->grad_init = true;
    } else {
      Tensor dx(ctx, x->value.dtype(), x->value.shape());
      ops::softmax_bwd(dy, y->value, dx, sc);
      x->note_grad_mutation();
      ops::add_inplace(x->grad, dx, 1.0f);
    }
  });
  return y;
}

// ---------------------------------------------------------- residual add ---
VarPtr residual_add(Tape& tape, VarPtr a, VarPtr b) {
  Context* ctx = a->value.ctx();
  auto out = make_var(Tensor(ctx, a->value.dtype(), a->value.shape()));
  ops::add_out(out->value, a->value, b->value);
  tape.push([a, b, out]() {
    Tensor& dout = out->grad_for_read();
    // ALIAS out's grad into uninitialized inputs instead of copying: out is
    // dead after this entry, and the replay order guarantees b's grad is
    // fully consumed (start of the sublayer's reverse chain) before any
    // accumulation into a's grad mutates the shared buffer. The invariant
    // is ENFORCED: a GradAliasGuard is planted on both holders, every
    // accumulation marks it (Var::grad_beta / note_grad_mutation), and any
    // later read of the branch's aliased grad fails loudly (tape.h).
    auto guard = std::make_shared<GradAliasGuard>();
    auto attach = [&](Var& v, bool trunk) {
      if (!v.grad_init) {
        v.grad = dout;
        v.grad_init = true;
        v.alias_guard = guard;
        v.alias_trunk = trunk;
      } else {
        v.note_grad_mutation();
        ops::add_inplace(v.grad, dout, 1.0f);
      }
    };
    attach(*b, /*trunk=*/false);
    attach(*a, /*trunk=*/true);
  });
  return out;
}

// ------------------------------------------------------------------ loss ---
Tensor ce_loss(Tape& tape, VarPtr logits, const Tensor& labels) {
  Context* ctx = logits->value.ctx();
  Tensor loss(ctx, DType::F32, Shape{1});
  NZ_CHECK(!logits->grad_init, "loss applied to var with existing grad");
  logits->ensure_grad();
  logits->grad_init = true;
  ops::ce_loss(logits->value, labels, loss, &logits->grad, 1.0f);
  (void)tape;
  return loss;
}

Tensor mse_loss(Tape& tape, VarPtr y, const Tensor& target) {
  Context* ctx = y->value.ctx();
  Tensor loss(ctx, DType::F32, Shape{1});
  NZ_CHECK(!y->grad_init, "loss applied to var with existing grad");
  y->ensure_grad();
  y->grad_init = true;
  ops::mse_loss(y->value, target, loss, &y->grad, 1.0f);
  (void)tape;
  return loss;
}

}  // namespace nz
