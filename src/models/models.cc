#include "src/models/models.h"

namespace nz {

// ------------------------------------------------------------------- MLP ---
MlpModel::MlpModel(Context* ctx, int64_t layers, int64_t dim, int64_t batch,
                   uint64_t seed)
    : dim_(dim), batch_(batch) {
  for (int64_t i = 0; i < layers; ++i)
    layers_.push_back(std::make_unique<Linear>(
        ctx, "mlp.l" + std::to_string(i), dim, dim, /*bias=*/true,
        /*gelu=*/true, seed + uint64_t(i) * 7919));
}

VarPtr MlpModel::forward(Tape& tape, VarPtr x) {
  VarPtr h = x;
  for (auto& l : layers_) h = l->forward(tape, h);
  return h;
}

void MlpModel::collect_params(std::vector<Param*>& out) {
  for (auto& l : layers_) l->collect_params(out);
}

// ------------------------------------------------------------------ BERT ---
BertEncoderModel::BertEncoderModel(Context* ctx, BertConfig cfg, uint64_t seed)
    : cfg_(cfg) {
  for (int64_t i = 0; i < cfg.layers; ++i) {
    EncoderLayer L;
    std::string nm = "bert.l" + std::to_string(i);
    uint64_t s = seed + uint64_t(i) * 104729;
    L.mha = std::make_unique<MultiHeadAttention>(ctx, nm + ".mha", cfg.hidden,
                                                 cfg.heads, cfg.seq, cfg.batch, s);
    L.ln1 = std::make_unique<LayerNorm>(ctx, nm + ".ln1", cfg.hidden);
    L.ffn1 = std::make_unique<Linear>(ctx, nm + ".ffn1", cfg.hidden, cfg.ffn,
                                      true, /*gelu=*/true, s + 3);
    L.ffn2 = std::make_unique<Linear>(ctx, nm + ".ffn2", cfg.ffn, cfg.hidden,
                                      true, /*gelu=*/false, s + 4);
    L.ln2 = std::make_unique<LayerNorm>(ctx, nm + ".ln2", cfg.hidden);
    layers_.push_back(std::move(L));
  }
}

VarPtr BertEncoderModel::forward(Tape& tape, VarPtr x) {
  // Post-LN transformer encoder (original BERT):
  //   h  = LN1(x + MHA(x));  out = LN2(h + FFN(h))
  VarPtr h = x;
  for (auto& L : layers_) {
    VarPtr attn = L.mha->forward(tape, h);
    VarPtr r1 = residual_add(tape, h, attn);
    VarPtr n1 = L.ln1->forward(tape, r1);
    VarPtr f1 = L.ffn1->forward(tape, n1);
    VarPtr f2 = L.ffn2->forward(tape, f1);
    VarPtr r2 = residual_add(tape, n1, f2);
    h = L.ln2->forward(tape, r2);
  }
  return h;
}

void BertEncoderModel::collect_params(std::vector<Param*>& out) {
  for (auto& L : layers_) {
    L.mha->collect_params(out);
    L.ln1->collect_params(out);
    L.ffn1->collect_params(out);
    L.ffn2->collect_params(out);
    L.ln2->collect_params(out);
  }
}

double BertEncoderModel::flops_per_step() const {
  // GEMM flops only, fwd+bwd ~ 3x fwd — minus layer 0's qkv-proj dgrad,
  // which Linear backward skips (the data input needs no grad).
  double T = double(cfg_.batch) * cfg_.seq;
  double qkv = 2.0 * T * cfg_.hidden * (3.0 * cfg_.hidden);
  double per_layer =
      qkv                                                // qkv proj
      + 2.0 * T * cfg_.seq * cfg_.hidden * 2.0           // QK^T and P.V
      + 2.0 * T * cfg_.hidden * cfg_.hidden              // out proj
      + 2.0 * T * cfg_.hidden * cfg_.ffn * 2.0;          // ffn pair
  return 3.0 * per_layer * double(cfg_.layers) - qkv;
}

}  // namespace nz
