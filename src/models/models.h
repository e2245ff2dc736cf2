// Model zoo: the two capability-contract models.
//  - MLP: L x [Linear D->D + bias + fused GELU], MSE loss (C11: L=4, D=4096).
//  - BERT-base encoder stack (C13): 12 x post-LN encoder layers,
//    H=768, FFN=3072, heads=12; MSE loss on final hidden states
//    (synthetic-data training; no embedding/vocab head by design).
#pragma once

#include <memory>

#include "src/nn/modules.h"

namespace nz {

class Model {
 public:
  virtual ~Model() = default;
  virtual VarPtr forward(Tape& tape, VarPtr x) = 0;
  virtual void collect_params(std::vector<Param*>& out) = 0;
  virtual Shape input_shape() const = 0;   // per-rank batch included
  virtual Shape output_shape() const = 0;
  std::vector<Param*> params() {
    std::vector<Param*> v;
    collect_params(v);
    return v;
  }
  int64_t num_params() {
    int64_t n = 0;
    for (auto* p : params()) n += p->numel();
    return n;
  }
  virtual int64_t samples_per_batch() const = 0;
  // Approximate training FLOPs per step (fwd+bwd GEMMs, for MFU logging).
  virtual double flops_per_step() const = 0;
};

class MlpModel : public Model {
 public:
  MlpModel(Context* ctx, int64_t layers, int64_t dim, int64_t batch, uint64_t seed);
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>& out) override;
  Shape input_shape() const override { return Shape{batch_, dim_}; }
  Shape output_shape() const override { return Shape{batch_, dim_}; }
  int64_t samples_per_batch() const override { return batch_; }
  double flops_per_step() const override {
    // 3 GEMMs/layer (fwd, wgrad, dgrad) EXCEPT layer 0, whose dgrad is
    // skipped because the input needs no grad (modules.cc Linear backward).
    return (3.0 * double(layers_.size()) - 1.0) * 2.0 * double(batch_) *
           dim_ * dim_;
  }

 private:
  int64_t dim_, batch_;
  std::vector<std::unique_ptr<Linear>> layers_;
};

struct BertConfig {
  int64_t layers = 12, hidden = 768, ffn = 3072, heads = 12;
  int64_t seq = 128, batch = 32;
};

class BertEncoderModel : public Model {
 public:
  BertEncoderModel(Context* ctx, BertConfig cfg, uint64_t seed);
  VarPtr forward(Tape& tape, VarPtr x) override;
  void collect_params(std::vector<Param*>& out) override;
  Shape input_shape() const override { return Shape{cfg_.batch * cfg_.seq, cfg_.hidden}; }
  Shape output_shape() const override { return Shape{cfg_.batch * cfg_.seq, cfg_.hidden}; }
  int64_t samples_per_batch() const override { return cfg_.batch; }
  double flops_per_step() const override;
  const BertConfig& config() const { return cfg_; }

 private:
  struct EncoderLayer {
    std::unique_ptr<MultiHeadAttention> mha;
    std::unique_ptr<LayerNorm> ln1, ln2;
    std::unique_ptr<Linear> ffn1, ffn2;
  };
  BertConfig cfg_;
  std::vector<EncoderLayer> layers_;
};

}  // namespace nz
