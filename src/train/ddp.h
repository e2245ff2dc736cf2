// DDP engine (call stack §3.3): the ONE parallelism
// strategy (C5). Gradient buckets (shared with the optimizer, C6 sizing)
// are all-reduced on the comm stream as soon as their last grad is
// produced, overlapped with the rest of backward (C14). On CPU contexts
// (FakeComm/TcpComm tests) the all-reduce runs inline and synchronously.
#pragma once

#include <unordered_map>

#include "src/comm/comm.h"
#include "src/nn/optim.h"

namespace nz {

class Ddp {
 public:
  Ddp(Context* ctx, Optimizer* opt, Comm* comm);

  // Broadcast rank-0 parameter values (flat data + f32 master) to all ranks.
  void sync_params();
  void pre_step();      // reset per-bucket pending counters
  void finish_grads();  // compute stream waits for outstanding all-reduces
  // make the compute stream wait only bucket bi's all-reduce (per-bucket
  // optimizer chaining: earlier buckets' Adam overlaps later comm)
  void wait_bucket(int bi);
  float grad_scale() const { return 1.0f / float(comm_->nranks()); }
  Comm* comm() const { return comm_; }
  int64_t comm_launches() const { return comm_launches_; }
  // bucket indices in all-reduce launch order (RCCL requires every rank to
  // issue collectives in the same order — tests assert cross-rank equality)
  const std::vector<int>& launch_order() const { return launch_order_; }

 private:
  void on_ready(Param& p);
  void launch_allreduce(Bucket& b, int bi);

  Context* ctx_;
  Optimizer* opt_;
  Comm* comm_;
  std::unordered_map<Param*, int> bucket_of_;
  int64_t comm_launches_ = 0;
  std::vector<int> launch_order_;
  // --allreduce-dtype=f32 fallback: grads are all-reduced
  // in bf16 by default (half the ring bytes on the per-link-bound xGMI
  // ring); NEZHA_ALLREDUCE_DTYPE=f32 upcasts each bucket to an f32 scratch
  // around the collective for full-precision reduction.
  bool allreduce_f32_ = false;
  std::vector<Tensor> f32_scratch_;
};

}  // namespace nz
