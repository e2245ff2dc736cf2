#include "src/train/ddp.h"

#include "src/util/util.h"

namespace nz {

Ddp::Ddp(Context* ctx, Optimizer* opt, Comm* comm)
    : ctx_(ctx), opt_(opt), comm_(comm) {
  auto& buckets = opt_->buckets();
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
  if (--b.pending == 0) launch_allreduce(b);
}

void Ddp::launch_allreduce(Bucket& b) {
  ++comm_launches_;
  if (ctx_->device() == Device::HIP) {
    // compute stream -> event -> comm stream all-reduce -> event
    b.ready_ev.record(*ctx_->compute_stream());
    b.ready_ev.wait(*ctx_->comm_stream());
    comm_->all_reduce(b.grad_flat.data(), b.numel, b.grad_flat.dtype(),
                      ReduceOp::SUM, ctx_->comm_stream());
    b.comm_ev.record(*ctx_->comm_stream());
    b.comm_inflight = true;
  } else {
    comm_->all_reduce(b.grad_flat.data(), b.numel, b.grad_flat.dtype(),
                      ReduceOp::SUM, nullptr);
  }
}

void Ddp::wait_bucket(int bi) {
  if (ctx_->device() != Device::HIP) return;
  Bucket& b = opt_->buckets()[bi];
  NZ_CHECK(b.pending == 0, "wait_bucket before bucket ready");
  if (b.comm_inflight) {
    b.comm_ev.wait(*ctx_->compute_stream());
    b.comm_inflight = false;
  }
}

void Ddp::finish_grads() {
  if (ctx_->device() != Device::HIP) return;
  for (auto& b : opt_->buckets()) {
    NZ_CHECK(b.pending == 0, "finish_grads with bucket still pending");
    if (b.comm_inflight) {
      b.comm_ev.wait(*ctx_->compute_stream());
      b.comm_inflight = false;
    }
  }
}

}  // namespace nz
