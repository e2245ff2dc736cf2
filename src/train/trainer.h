// Trainer (call stack §3.2): synthetic-data training
// loop. Input/target batches are device-resident and reused — zero
// host<->device transfers in steady state except optional loss readback.
#pragma once

#include <memory>

#include "src/models/models.h"
#include "src/nn/optim.h"
#include "src/train/ddp.h"
#include "src/util/util.h"

namespace nz {

// Per-rank data-shard seed stride: launchers (bench.py, cmd/train.cc) seed
// rank r with base + kDdpSeedStride*r; checkpoint restore re-derives the
// same shards from rank 0's saved base seed.
constexpr uint64_t kDdpSeedStride = 1000003ull;

class Trainer {
 public:
  Trainer(Context* ctx, std::unique_ptr<Model> model, Optimizer::Kind optim,
          float lr, int64_t bucket_bytes, uint64_t data_seed);

  // Attach DDP: installs hooks and broadcasts initial params.
  void attach_ddp(Comm* comm);

  // Regenerate the synthetic input/target batch (per-rank data shards).
  void reseed_data(uint64_t seed);

  // One training step: fwd -> loss -> bwd (-> overlapped all-reduce) -> opt.
  // Asynchronous on HIP (no sync, no download).
  void step();
  float last_loss();   // downloads the loss scalar (syncs on HIP);
                       // COLLECTIVE under DDP (mean over ranks)
  float last_loss_local();  // this rank's loss only (no collective)
  int step_count() const { return opt_->step_count(); }

  Model* model() { return model_.get(); }
  Optimizer* optimizer() { return opt_.get(); }
  Ddp* ddp() { return ddp_.get(); }
  Context* ctx() { return ctx_; }

  // Checkpoint: rank 0 writes; resume loads + broadcasts.
  // local_only: THIS rank writes its own file, no rank gating/barrier —
  // test hook for bitwise cross-rank agreement checks.
  void save_checkpoint(const std::string& path, bool local_only = false);
  void load_checkpoint(const std::string& path);

 private:
  Context* ctx_;
  std::unique_ptr<Model> model_;
  std::unique_ptr<Optimizer> opt_;
  std::unique_ptr<Ddp> ddp_;
  Tape tape_;
  Tensor input_, target_, loss_;
  uint64_t data_seed_ = 0;  // persisted as meta/data_seed; reseed on load
  bool data_stream_ = false;  // NEZHA_DATA=stream: fresh batch every step
  // per-step metrics: enabled by NEZHA_METRICS=<path>;
  // fwd/bwd/opt ms splits via hipEvents (syncs each step — observability
  // mode, off on the bench path).
  std::unique_ptr<MetricsWriter> metrics_;
  Event ev_start_, ev_fwd_, ev_bwd_, ev_opt_;
  // hipGraph capture (NEZHA_GRAPH=1, HIP + no DDP): the launch-bound step
  // is captured once and replayed; Adam bias corrections flow through a
  // 2-float device buffer refreshed before each replay.
  void step_body(const Tensor* bc_dev);
  bool graph_mode_ = false;
  bool graph_captured_ = false;
  void* graph_exec_ = nullptr;  // hipGraphExec_t (opaque here)
  Tensor bc_dev_;
  float bc_host_[2] = {1.f, 1.f};  // staging for async bc upload
};

}  // namespace nz
