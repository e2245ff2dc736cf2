#include "src/train/ddp.h"

#include "src/util/util.h"

namespace nz {

Ddp::Ddp(Context* ctx, Optimizer* opt, Comm* comm)
    : ctx_(ctx), opt_(opt), comm_(comm) {
  const char* ad = getenv("NEZHA_ALLREDUCE_DTYPE");
  allreduce_f32_ = ad && std::string(ad) == "f32";
  auto& buckets = opt_->buckets();
  if (allreduce_f32_ && buckets[0].grad_flat.dtype() == DType::BF16) {
    for (auto& b : buckets)
      f32_scratch_.emplace_back(ctx_, DType::F32, Shape{b.numel});
    NZ_LOG_INFO("DDP: all-reduce dtype f32 (upcast around the collective)");
  } else {
    allreduce_f32_ = false;
  }
  for (int bi = 0; bi < (int)buckets.size(); ++bi) {
    for (Param* p : buckets[bi].params) {
      bucket_of_[p] = bi;
      p->ready_hook = [this](Param& pp) { on_ready(pp); };
    }
  }
  NZ_LOG_INFO("DDP: %d ranks, %zu buckets", comm_->nranks(), buckets.size());
}

void Ddp::sync_params() {
  for (auto& b : opt_->buckets()) {
    comm_->broadcast(b.data_flat.data(), b.numel, b.data_flat.dtype(), 0,
                     ctx_->device() == Device::HIP ? ctx_->compute_stream() : nullptr);
    comm_->broadcast(b.master.data(), b.numel, DType::F32, 0,
                     ctx_->device() == Device::HIP ? ctx_->compute_stream() : nullptr);
  }
  if (ctx_->device() == Device::HIP) ctx_->compute_stream()->sync();
}

void Ddp::pre_step() {
  for (auto& b : opt_->buckets()) {
    b.pending = (int)b.params.size();
    b.comm_inflight = false;
  }
}

void Ddp::on_ready(Param& p) {
  auto it = bucket_of_.find(&p);
  NZ_CHECK(it != bucket_of_.end(), "ready hook for unknown param " + p.name);
  Bucket& b = opt_->buckets()[it->second];
  NZ_CHECK(b.pending > 0, "bucket over-ready (param " + p.name + " reused?)");
  if (--b.pending == 0) launch_allreduce(b, it->second);
}

void Ddp::launch_allreduce(Bucket& b, int bi) {
  ++comm_launches_;
  launch_order_.push_back(bi);
  if (ctx_->device() == Device::HIP) {
    // compute stream -> event -> comm stream all-reduce -> event
    if (allreduce_f32_) ops::cast_copy(f32_scratch_[bi], b.grad_flat);
    b.ready_ev.record(*ctx_->compute_stream());
    b.ready_ev.wait(*ctx_->comm_stream());
    if (allreduce_f32_)
      comm_->all_reduce(f32_scratch_[bi].data(), b.numel, DType::F32,
                        ReduceOp::SUM, ctx_->comm_stream());
    else
      comm_->all_reduce(b.grad_flat.data(), b.numel, b.grad_flat.dtype(),
                        ReduceOp::SUM, ctx_->comm_stream());
    b.comm_ev.record(*ctx_->comm_stream());
    b.comm_inflight = true;
  } else {
    if (allreduce_f32_) {
      ops::cast_copy(f32_scratch_[bi], b.grad_flat);
      comm_->all_reduce(f32_scratch_[bi].data(), b.numel, DType::F32,
                        ReduceOp::SUM, nullptr);
      ops::cast_copy(b.grad_flat, f32_scratch_[bi]);
    } else {
      comm_->all_reduce(b.grad_flat.data(), b.numel, b.grad_flat.dtype(),
                        ReduceOp::SUM, nullptr);
    }
  }
}

void Ddp::wait_bucket(int bi) {
  if (ctx_->device() != Device::HIP) return;
  Bucket& b = opt_->buckets()[bi];
  NZ_CHECK(b.pending == 0, "wait_bucket before bucket ready");
  if (b.comm_inflight) {
    b.comm_ev.wait(*ctx_->compute_stream());
    b.comm_inflight = false;
    // f32 mode: fold the reduced f32 values back into the bf16 bucket on
    // the compute stream (ordered after the comm event wait)
    if (allreduce_f32_) ops::cast_copy(b.grad_flat, f32_scratch_[bi]);
  }
}

void Ddp::finish_grads() {
  if (ctx_->device() != Device::HIP) return;
  for (auto& b : opt_->buckets()) {
    NZ_CHECK(b.pending == 0, "finish_grads with bucket still pending");
    if (b.comm_inflight) {
      b.comm_ev.wait(*ctx_->compute_stream());
      b.comm_inflight = false;
      if (allreduce_f32_) {
        int bi = int(&b - &opt_->buckets()[0]);
        ops::cast_copy(b.grad_flat, f32_scratch_[bi]);
      }
    }
  }
}

}  // namespace nz
