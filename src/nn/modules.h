// NN modules: Linear (+fused bias+GELU), LayerNorm,
// MultiHeadAttention, losses. Forward builds the tape; each parameter is
// consumed exactly once per step (asserted) so its grad-ready hook can fire
// immediately after its gradient is produced (DDP overlap, C14).
#pragma once

#include <string>
#include <vector>

#include "src/autograd/tape.h"

namespace nz {

class Module {
 public:
  virtual ~Module() = default;
  virtual VarPtr forward(Tape& tape, VarPtr x) = 0;
  virtual void collect_params(std::vector<Param*>& out) = 0;
};

// y = gelu?(x @ W + b). W stored [in, out] row-major (NN forward GEMM).
class Linear : public Module {
 public:
  Linear(Context* ctx, std::string name, int64_t in, int64_t out, bool bias,
         bool gelu, uint64_t seed);
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>& out) override;
  Param w, b;
  bool has_bias, fuse_gelu;
  int64_t in_dim, out_dim;

 private:
  Context* ctx_;
};

// Post-LN LayerNorm over the last dim.
class LayerNorm : public Module {
 public:
  LayerNorm(Context* ctx, std::string name, int64_t dim);
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>& out) override;
  Param gamma, beta;
  int64_t dim;

 private:
  Context* ctx_;
};

// Multi-head self-attention (C13): fused qkv proj -> split heads ->
// batched QK^T -> scaled softmax -> batched P.V -> merge -> out proj.
// Composed from K1 batched + K4.
class MultiHeadAttention : public Module {
 public:
  MultiHeadAttention(Context* ctx, std::string name, int64_t hidden,
                     int64_t heads, int64_t seq, int64_t batch, uint64_t seed);
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>& out) override;
  Linear qkv_proj, out_proj;
  int64_t hidden, heads, dh, seq, batch;

 private:
  Context* ctx_;
};

// Standalone activation/softmax modules (the hot path
// fuses GELU into the producing GEMM — these exist for model-building
// completeness and the unfused clarity path).
class Gelu : public Module {
 public:
  explicit Gelu(Context* ctx) : ctx_(ctx) {}
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>&) override {}

 private:
  Context* ctx_;
};

class Softmax : public Module {
 public:
  Softmax(Context* ctx, float scale = 1.0f) : ctx_(ctx), scale_(scale) {}
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>&) override {}

 private:
  Context* ctx_;
  float scale_;
};

// Residual add: out = a + b (both tracked).
VarPtr residual_add(Tape& tape, VarPtr a, VarPtr b);

// Losses: return a device f32 scalar tensor; seed dy into y's grad.
Tensor mse_loss(Tape& tape, VarPtr y, const Tensor& target);
// Cross-entropy over the last dim; labels hold integral class ids (f32).
Tensor ce_loss(Tape& tape, VarPtr logits, const Tensor& labels);

}  // namespace nz
