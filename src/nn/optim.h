// Optimizers on flat buckets: params/grads of a model
// are flattened into ~bucket_bytes buckets (reverse registration order =
// backward production order), so (a) the Adam/SGD step is ONE kernel launch
// per bucket (C3) and (b) the DDP engine all-reduces the same flat buckets
// (C6: default 64 MB, chosen for the xGMI ring — SURVEY.md §2.4).
#pragma once

#include <functional>
#include <vector>

#include "src/autograd/tape.h"

namespace nz {

struct Bucket {
  std::vector<Param*> params;
  std::vector<int64_t> offsets;   // element offset of each param in the flats
  Tensor data_flat;               // param dtype; Param::data views into this
  Tensor grad_flat;               // param dtype; Param::grad views into this
  Tensor master, m, v, momentum;  // f32 optimizer state (m/v: adam, momentum: sgd)
  int64_t numel = 0;
  // DDP bookkeeping (driven by Ddp)
  int pending = 0;
  bool comm_inflight = false;
  Event ready_ev, comm_ev;
};

class Optimizer {
 public:
  enum class Kind { ADAM, SGD };
  Optimizer(Context* ctx, std::vector<Param*> params, Kind kind, float lr,
            int64_t bucket_bytes);
  // pre_bucket (optional): called right before each bucket's update kernel
  // is enqueued — the DDP engine uses it to make the compute stream wait
  // only THAT bucket's all-reduce, so earlier buckets' optimizer kernels
  // overlap later buckets' comm (C14).
  void step(float grad_scale, const Tensor* bc_dev = nullptr,
            const std::function<void(int)>& pre_bucket = nullptr);
  std::vector<Bucket>& buckets() { return buckets_; }
  const std::vector<Param*>& params() const { return params_; }
  int step_count() const { return step_; }
  void set_step_count(int s) { step_ = s; }

  float lr, beta1 = 0.9f, beta2 = 0.999f, eps = 1e-8f, weight_decay = 0.f;
  float sgd_momentum = 0.f;
  Kind kind;

 private:
  Context* ctx_;
  std::vector<Param*> params_;
  std::vector<Bucket> buckets_;
  int step_ = 0;
};

}  // namespace nz
