#include "src/nn/optim.h"

namespace nz {

Optimizer::Optimizer(Context* ctx, std::vector<Param*> params, Kind k, float lr_,
                     int64_t bucket_bytes)
    : lr(lr_), kind(k), ctx_(ctx), params_(std::move(params)) {
  NZ_CHECK(!params_.empty(), "optimizer with no params");
  size_t es = dtype_size(params_[0]->data.dtype());
  int64_t bucket_elems = std::max<int64_t>(1, bucket_bytes / (int64_t)es);

  // Group in reverse registration order (= order backward produces grads).
  std::vector<std::vector<Param*>> groups;
  std::vector<Param*> cur;
  int64_t cur_n = 0;
  for (auto it = params_.rbegin(); it != params_.rend(); ++it) {
    Param* p = *it;
    NZ_CHECK(p->data.dtype() == params_[0]->data.dtype(),
             "mixed param dtypes not supported");
    if (cur_n > 0 && cur_n + p->numel() > bucket_elems) {
      groups.push_back(std::move(cur));
      cur.clear();
      cur_n = 0;
    }
    cur.push_back(p);
    cur_n += p->numel();
  }
  if (!cur.empty()) groups.push_back(std::move(cur));

  DType pd = params_[0]->data.dtype();
  for (auto& g : groups) {
    Bucket b;
    b.params = g;
    for (Param* p : g) {
      b.offsets.push_back(b.numel);
      b.numel += p->numel();
    }
    b.data_flat = Tensor(ctx_, pd, Shape{b.numel});
    b.grad_flat = Tensor(ctx_, pd, Shape{b.numel});
    ops::fill(b.grad_flat, 0.f);
    // Move existing param values into the flat buffer and repoint views.
    for (size_t i = 0; i < g.size(); ++i) {
      Param* p = g[i];
      Tensor dview = Tensor::view(b.data_flat, pd, p->data.shape(), b.offsets[i]);
      ops::cast_copy(dview, p->data);
      p->data = dview;
      p->grad = Tensor::view(b.grad_flat, pd, p->data.shape(), b.offsets[i]);
      p->grad_init = false;
    }
    b.master = Tensor(ctx_, DType::F32, Shape{b.numel});
    ops::cast_copy(b.master, b.data_flat);
    if (kind == Kind::ADAM) {
      b.m = Tensor(ctx_, DType::F32, Shape{b.numel});
      b.v = Tensor(ctx_, DType::F32, Shape{b.numel});
      ops::fill(b.m, 0.f);
      ops::fill(b.v, 0.f);
    } else if (sgd_momentum != 0.f) {
      b.momentum = Tensor(ctx_, DType::F32, Shape{b.numel});
      ops::fill(b.momentum, 0.f);
    }
    buckets_.push_back(std::move(b));
  }
}

void Optimizer::step(float grad_scale, const Tensor* bc_dev,
                     const std::function<void(int)>& pre_bucket) {
  ++step_;
  for (size_t bi = 0; bi < buckets_.size(); ++bi) {
    auto& b = buckets_[bi];
    if (pre_bucket) pre_bucket((int)bi);
    if (kind == Kind::ADAM) {
      ops::adam_step(b.data_flat, b.grad_flat, b.master, b.m, b.v, lr, beta1,
                     beta2, eps, step_, grad_scale, weight_decay, bc_dev);
    } else {
      if (sgd_momentum != 0.f && !b.momentum.defined()) {
        b.momentum = Tensor(ctx_, DType::F32, Shape{b.numel});
        ops::fill(b.momentum, 0.f);
      }
      ops::sgd_step(b.data_flat, b.grad_flat, b.master,
                    b.momentum.defined() ? &b.momentum : nullptr, lr,
                    sgd_momentum, grad_scale);
    }
    for (Param* p : b.params) p->grad_init = false;
  }
}

}  // namespace nz
