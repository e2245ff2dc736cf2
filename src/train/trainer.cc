#include "src/train/trainer.h"

#include <hip/hip_runtime.h>

#include "src/ckpt/npz.h"
#include "src/util/util.h"

#include <unistd.h>

namespace nz {

Trainer::Trainer(Context* ctx, std::unique_ptr<Model> model,
                 Optimizer::Kind optim, float lr, int64_t bucket_bytes,
                 uint64_t data_seed)
    : ctx_(ctx), model_(std::move(model)) {
  opt_ = std::make_unique<Optimizer>(ctx, model_->params(), optim, lr, bucket_bytes);
  DType dt = ctx->device() == Device::HIP ? DType::BF16 : DType::F32;
  input_ = Tensor(ctx, dt, model_->input_shape());
  target_ = Tensor(ctx, dt, model_->output_shape());
  data_seed_ = data_seed;
  ops::rand_normal(input_, data_seed, 0.f, 1.f);
  ops::rand_normal(target_, data_seed + 1, 0.f, 1.f);
  if (const char* mp = getenv("NEZHA_METRICS"))
    metrics_ = std::make_unique<MetricsWriter>(mp);
  const char* dm = getenv("NEZHA_DATA");
  data_stream_ = dm && std::string(dm) == "stream";
  const char* g = getenv("NEZHA_GRAPH");
  graph_mode_ = g && g[0] == '1' && ctx->device() == Device::HIP &&
                !data_stream_;  // stream mode regenerates inputs per step
  // allocate the bias-correction buffer up front: taking a cached block
  // at capture time would starve the captured step's own allocations
  if (graph_mode_) bc_dev_ = Tensor(ctx, DType::F32, Shape{2});
}

void Trainer::reseed_data(uint64_t seed) {
  data_seed_ = seed;
  ops::rand_normal(input_, seed, 0.f, 1.f);
  ops::rand_normal(target_, seed + 1, 0.f, 1.f);
}

void Trainer::attach_ddp(Comm* comm) {
  ddp_ = std::make_unique<Ddp>(ctx_, opt_.get(), comm);
  ddp_->sync_params();
}

void Trainer::step_body(const Tensor* bc_dev) {
  if (data_stream_) {
    // streaming synthetic data: a fresh deterministic batch every step
    // (counter-based RNG keyed by data_seed + step index)
    uint64_t s = data_seed_ + 2654435761ull * (uint64_t)(opt_->step_count() + 1);
    ops::rand_normal(input_, s, 0.f, 1.f);
    ops::rand_normal(target_, s + 1, 0.f, 1.f);
  }
  if (ddp_) ddp_->pre_step();
  tape_.clear();
  const bool mx = metrics_ && ctx_->device() == Device::HIP && !graph_mode_;
  if (mx) ev_start_.record(*ctx_->compute_stream());
  VarPtr x = make_var(input_, /*needs_grad=*/false);
  VarPtr y = model_->forward(tape_, x);
  loss_ = mse_loss(tape_, y, target_);
  if (mx) ev_fwd_.record(*ctx_->compute_stream());
  tape_.backward();
  if (mx) ev_bwd_.record(*ctx_->compute_stream());
  if (ddp_ && ctx_->device() == Device::HIP) {
    // per-bucket chaining: bucket i's Adam waits only bucket i's comm
    opt_->step(ddp_->grad_scale(), bc_dev,
               [this](int bi) { ddp_->wait_bucket(bi); });
  } else {
    if (ddp_) ddp_->finish_grads();
    opt_->step(ddp_ ? ddp_->grad_scale() : 1.0f, bc_dev);
  }
  // one writer per file; never inside graph capture (last_loss_local syncs
  // + D2H-copies, which would abort hipStreamBeginCapture)
  const bool mwrite =
      metrics_ && !graph_mode_ && (!ddp_ || ddp_->comm()->rank() == 0);
  if (mx && mwrite) {
    ev_opt_.record(*ctx_->compute_stream());
    ev_opt_.sync();
    metrics_->write({{"step", (double)opt_->step_count()},
                     {"loss", last_loss_local()},
                     {"fwd_ms", ev_fwd_.elapsed_ms_since(ev_start_)},
                     {"bwd_ms", ev_bwd_.elapsed_ms_since(ev_fwd_)},
                     {"opt_ms", ev_opt_.elapsed_ms_since(ev_bwd_)},
                     {"mem_hw_gb", ctx_->arena()->high_water() / 1e9}});
  } else if (mwrite) {
    metrics_->write({{"step", (double)opt_->step_count()},
                     {"loss", last_loss_local()}});
  }
}

// NEZHA_FAULT_STEP handling + graph capture/replay wrapper.
void Trainer::step() {
  static const long fault_step = [] {
    const char* v = getenv("NEZHA_FAULT_STEP");
    return v ? atol(v) : -1L;
  }();
  if (fault_step >= 0 && opt_->step_count() + 1 == fault_step) {
    const char* fr = getenv("NEZHA_FAULT_RANK");
    int frank = fr ? atoi(fr) : 0;
    if (!ddp_ || ddp_->comm()->rank() == frank) {
      NZ_LOG_ERROR("NEZHA_FAULT_STEP: injected fault at step %ld", fault_step);
      _exit(42);
    }
  }

  // Graph capture works under DDP too when the comm backend is pure
  // stream-ordered device work (RCCL): the comm-stream collectives are
  // pulled into the capture through the ready/comm event edges, so a
  // replayed step carries the same overlapped all-reduce schedule
  // (VERDICT r01 next-step #7). Host-staged test comms are not capturable.
  if (!graph_mode_ || (ddp_ && !ddp_->comm()->capturable())) {
    step_body(nullptr);
    return;
  }

  auto upload_bc = [&](int stepno) {
    // member buffer, not stack: hipMemcpyAsync from pageable host memory
    // must outlive the call
    bc_host_[0] = 1.0f - powf(opt_->beta1, (float)stepno);
    bc_host_[1] = 1.0f - powf(opt_->beta2, (float)stepno);
    hipStream_t st = ctx_->compute_stream()->handle();
    NZ_CHECK(hipMemcpyAsync(bc_dev_.data(), bc_host_, 8,
                            hipMemcpyHostToDevice, st) == hipSuccess,
             "bc upload failed");
  };

  if (!graph_captured_) {
    if (opt_->step_count() < 2) {
      // TWO eager steps warm the arena so capture never calls hipMalloc:
      // step 1 sizes every temporary; step 2 frees step 1's loss scalar
      // (held alive across steps by loss_) back into the cache so the
      // captured step's allocation of it is a cache hit.
      step_body(nullptr);
      return;
    }
    ctx_->make_current();
    int next = opt_->step_count() + 1;
    upload_bc(next);
    ctx_->compute_stream()->sync();
    hipStream_t st = ctx_->compute_stream()->handle();
    // Capture failure must not kill the run (and under DDP must not
    // desynchronize ranks): fall back to eager. The eager step issues the
    // exact same collective sequence a replayed graph would, so a mixed
    // graph/eager world stays collectively consistent.
    int count_before = opt_->step_count();
    try {
      NZ_CHECK(hipStreamBeginCapture(st, hipStreamCaptureModeThreadLocal) ==
                   hipSuccess,
               "graph capture begin failed");
      step_body(&bc_dev_);  // increments step_count to `next`
      hipGraph_t graph = nullptr;
      NZ_CHECK(hipStreamEndCapture(st, &graph) == hipSuccess,
               "graph capture end failed");
      hipGraphExec_t exec = nullptr;
      hipError_t inst = hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0);
      (void)hipGraphDestroy(graph);
      NZ_CHECK(inst == hipSuccess, "graph instantiate failed");
      graph_exec_ = (void*)exec;
      graph_captured_ = true;
    } catch (const Error& e) {
      NZ_LOG_ERROR("hipGraph capture failed (%s) — falling back to eager",
                   e.what());
      hipGraph_t dead = nullptr;
      (void)hipStreamEndCapture(st, &dead);  // unwind a half-open capture
      if (dead) (void)hipGraphDestroy(dead);
      (void)hipGetLastError();  // clear sticky error
      graph_mode_ = false;
      opt_->set_step_count(count_before);
      // a mid-capture throw can leave per-param/bucket host state dirty
      for (auto& b : opt_->buckets())
        for (Param* p : b.params) p->grad_init = false;
      if (ddp_) ddp_->pre_step();
      step_body(nullptr);
      return;
    }
    // the captured work did NOT run; launch it for this step
    NZ_CHECK(hipGraphLaunch((hipGraphExec_t)graph_exec_, st) == hipSuccess,
             "graph launch failed");
    NZ_LOG_INFO("hipGraph captured: training step replays as one graph%s",
                ddp_ ? " (incl. bucketed RCCL all-reduce)" : "");
    return;
  }
  int next = opt_->step_count() + 1;
  upload_bc(next);
  NZ_CHECK(hipGraphLaunch((hipGraphExec_t)graph_exec_,
                          ctx_->compute_stream()->handle()) == hipSuccess,
           "graph launch failed");
  opt_->set_step_count(next);
}

float Trainer::last_loss_local() {
  NZ_CHECK(loss_.defined(), "no step taken yet");
  float v = 0.f;
  loss_.download(&v, 4);
  return v;
}

float Trainer::last_loss() {
  float v = last_loss_local();
  // under DP the reported loss is the mean over ranks (each rank trains a
  // different shard — SURVEY.md §2.4 "loss/metric reduction" call site)
  if (ddp_ && ddp_->comm() && ddp_->comm()->nranks() > 1) {
    v = ddp_->comm()->all_reduce_scalar(v, ReduceOp::SUM) /
        (float)ddp_->comm()->nranks();
  }
  return v;
}

// ------------------------------------------------------------ checkpoint ---
static NpyArray slice_to_npy(const std::vector<char>& flat_host, DType dt,
                             const Shape& shape, int64_t elem_offset) {
  NpyArray a;
  a.dtype = dt;
  a.shape = shape.dims;
  size_t es = dtype_size(dt);
  size_t off = size_t(elem_offset) * es;
  size_t n = size_t(shape.numel()) * es;
  a.data.assign(flat_host.begin() + off, flat_host.begin() + off + n);
  return a;
}

void Trainer::save_checkpoint(const std::string& path, bool local_only) {
  if (!local_only && ddp_ && ddp_->comm()->rank() != 0) {
    ddp_->comm()->barrier();
    return;
  }
  std::map<std::string, NpyArray> arrays;
  for (auto& b : opt_->buckets()) {
    std::vector<char> data_host(b.data_flat.bytes()), master_host(b.master.bytes());
    b.data_flat.download(data_host.data(), data_host.size());
    b.master.download(master_host.data(), master_host.size());
    std::vector<char> m_host, v_host;
    if (b.m.defined()) {
      m_host.resize(b.m.bytes());
      v_host.resize(b.v.bytes());
      b.m.download(m_host.data(), m_host.size());
      b.v.download(v_host.data(), v_host.size());
    }
    for (size_t i = 0; i < b.params.size(); ++i) {
      Param* p = b.params[i];
      int64_t off = b.offsets[i];
      arrays["param/" + p->name] =
          slice_to_npy(data_host, p->data.dtype(), p->data.shape(), off);
      arrays["master/" + p->name] =
          slice_to_npy(master_host, DType::F32, p->data.shape(), off);
      if (!m_host.empty()) {
        arrays["adam_m/" + p->name] =
            slice_to_npy(m_host, DType::F32, p->data.shape(), off);
        arrays["adam_v/" + p->name] =
            slice_to_npy(v_host, DType::F32, p->data.shape(), off);
      }
    }
  }
  NpyArray stepa;
  stepa.dtype = DType::F32;
  stepa.shape = {1};
  float st = (float)opt_->step_count();
  stepa.data.resize(4);
  std::memcpy(stepa.data.data(), &st, 4);
  arrays["meta/step"] = stepa;
  // RNG state (SURVEY.md §5.4 meta/rng): the generator is counter-based,
  // so the data seed IS the full state. Four 16-bit chunks, each exactly
  // representable in f32, cover all 64 seed bits.
  NpyArray rnga;
  rnga.dtype = DType::F32;
  rnga.shape = {4};
  float sd[4] = {(float)(data_seed_ & 0xffffu),
                 (float)((data_seed_ >> 16) & 0xffffu),
                 (float)((data_seed_ >> 32) & 0xffffu),
                 (float)((data_seed_ >> 48) & 0xffffu)};
  rnga.data.resize(16);
  std::memcpy(rnga.data.data(), sd, 16);
  arrays["meta/data_seed"] = rnga;
  npz_save(path, arrays);
  if (!local_only && ddp_) ddp_->comm()->barrier();
}

void Trainer::load_checkpoint(const std::string& path) {
  bool is_root = !ddp_ || ddp_->comm()->rank() == 0;
  if (is_root) {
    auto arrays = npz_load(path);
    auto fetch = [&](const std::string& key) -> NpyArray& {
      auto it = arrays.find(key);
      NZ_CHECK(it != arrays.end(), "checkpoint missing member " + key);
      return it->second;
    };
    for (auto& b : opt_->buckets()) {
      std::vector<char> data_host(b.data_flat.bytes()), master_host(b.master.bytes());
      std::vector<char> m_host(b.m.defined() ? b.m.bytes() : 0),
          v_host(b.v.defined() ? b.v.bytes() : 0);
      for (size_t i = 0; i < b.params.size(); ++i) {
        Param* p = b.params[i];
        size_t es = dtype_size(p->data.dtype());
        size_t off = size_t(b.offsets[i]) * es;
        auto& pa = fetch("param/" + p->name);
        NZ_CHECK(pa.numel() == p->numel(), "shape mismatch for " + p->name);
        if (pa.dtype == p->data.dtype()) {
          NZ_CHECK(pa.data.size() == size_t(p->numel()) * es,
                   "byte-size mismatch for " + p->name);
          std::memcpy(data_host.data() + off, pa.data.data(), pa.data.size());
        } else if (pa.dtype == DType::F32 && p->data.dtype() == DType::BF16) {
          // CPU (f32) checkpoint loaded on GPU (bf16): round-convert; exact
          // values are restored from the f32 master below anyway.
          NZ_CHECK(pa.data.size() == size_t(p->numel()) * 4,
                   "byte-size mismatch for " + p->name);
          const float* srcf = (const float*)pa.data.data();
          uint16_t* dst = (uint16_t*)(data_host.data() + off);
          for (int64_t k = 0; k < p->numel(); ++k) dst[k] = f32_to_bf16(srcf[k]);
        } else if (pa.dtype == DType::BF16 && p->data.dtype() == DType::F32) {
          // GPU (bf16) checkpoint loaded on CPU (f32): widen.
          NZ_CHECK(pa.data.size() == size_t(p->numel()) * 2,
                   "byte-size mismatch for " + p->name);
          const uint16_t* srcb = (const uint16_t*)pa.data.data();
          float* dst = (float*)(data_host.data() + off);
          for (int64_t k = 0; k < p->numel(); ++k) dst[k] = bf16_to_f32(srcb[k]);
        } else {
          NZ_CHECK(false, "unsupported dtype conversion for " + p->name);
        }
        auto& ma = fetch("master/" + p->name);
        NZ_CHECK(ma.data.size() == size_t(p->numel()) * 4,
                 "master byte-size mismatch for " + p->name);
        std::memcpy(master_host.data() + b.offsets[i] * 4, ma.data.data(),
                    ma.data.size());
        if (pa.dtype != p->data.dtype() && p->data.dtype() == DType::F32) {
          // cross-device resume onto f32 params: the f32 master is exact
          // where the bf16 param member is rounded — prefer the master
          std::memcpy(data_host.data() + off, ma.data.data(), ma.data.size());
        }
        if (!m_host.empty()) {
          std::memcpy(m_host.data() + b.offsets[i] * 4,
                      fetch("adam_m/" + p->name).data.data(), p->numel() * 4);
          std::memcpy(v_host.data() + b.offsets[i] * 4,
                      fetch("adam_v/" + p->name).data.data(), p->numel() * 4);
        }
      }
      b.data_flat.upload(data_host.data(), data_host.size());
      b.master.upload(master_host.data(), master_host.size());
      if (!m_host.empty()) {
        b.m.upload(m_host.data(), m_host.size());
        b.v.upload(v_host.data(), v_host.size());
      }
    }
    auto it = arrays.find("meta/step");
    if (it != arrays.end()) {
      float st;
      std::memcpy(&st, it->second.data.data(), 4);
      opt_->set_step_count((int)st);
    }
    auto itr = arrays.find("meta/data_seed");
    if (itr != arrays.end() && itr->second.data.size() == 16) {
      float sd[4];
      std::memcpy(sd, itr->second.data.data(), 16);
      uint64_t seed = (uint64_t)sd[0] | ((uint64_t)sd[1] << 16) |
                      ((uint64_t)sd[2] << 32) | ((uint64_t)sd[3] << 48);
      reseed_data(seed);
    } else if (itr != arrays.end() && itr->second.data.size() == 8) {
      // legacy 2x24-bit format (covers only 48 seed bits)
      float sd[2];
      std::memcpy(sd, itr->second.data.data(), 8);
      reseed_data((uint64_t)sd[0] | ((uint64_t)sd[1] << 24));
    }
  }
  if (ddp_) {
    // rank 0 loaded; broadcast params + master; broadcast step via scalar.
    ddp_->sync_params();
    for (auto& b : opt_->buckets()) {
      if (b.m.defined()) {
        Stream* s = ctx_->device() == Device::HIP ? ctx_->compute_stream() : nullptr;
        ddp_->comm()->broadcast(b.m.data(), b.numel, DType::F32, 0, s);
        ddp_->comm()->broadcast(b.v.data(), b.numel, DType::F32, 0, s);
      }
    }
    float st = ddp_->comm()->all_reduce_scalar(
        is_root ? (float)opt_->step_count() : 0.f, ReduceOp::MAX);
    opt_->set_step_count((int)st);
    // Broadcast the restored base data seed (rank 0's seed = base + 0) and
    // re-derive each rank's shard seed with the launcher's stride convention
    // (bench.py / cmd/train.cc use base + kDdpSeedStride * rank).
    uint64_t base = 0;
    for (int c = 0; c < 4; ++c) {
      float chunk = ddp_->comm()->all_reduce_scalar(
          is_root ? (float)((data_seed_ >> (16 * c)) & 0xffffu) : 0.f,
          ReduceOp::MAX);
      base |= (uint64_t)chunk << (16 * c);
    }
    if (!is_root)
      reseed_data(base + kDdpSeedStride * (uint64_t)ddp_->comm()->rank());
    if (ctx_->device() == Device::HIP) ctx_->compute_stream()->sync();
  }
}

}  // namespace nz
