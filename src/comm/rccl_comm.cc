// RcclComm: all RCCL call sites of the framework.
// Native RCCL over xGMI — includes rccl/rccl.h directly, no NCCL alias (C7).
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <unistd.h>

#include <atomic>
#include <thread>

#include "src/comm/comm.h"
#include "src/util/util.h"

namespace nz {

#define NZ_NCCL_CHECK(expr)                                                   \
  do {                                                                        \
    ncclResult_t r_ = (expr);                                                 \
    if (r_ != ncclSuccess)                                                    \
      ::nz::fail(std::string("RCCL error: ") + ncclGetErrorString(r_) +       \
                 " in " #expr);                                               \
  } while (0)

int comm_uid_bytes() { return (int)sizeof(ncclUniqueId); }

void comm_get_uid(void* out) {
  ncclUniqueId id;
  NZ_NCCL_CHECK(ncclGetUniqueId(&id));
  std::memcpy(out, &id, sizeof(id));
}

namespace {

ncclDataType_t to_nccl(DType d) {
  return d == DType::F32 ? ncclFloat32 : ncclBfloat16;
}

class RcclComm : public Comm {
 public:
  RcclComm(Context* ctx, int nranks, int rank, const void* uid)
      : ctx_(ctx), nranks_(nranks), rank_(rank) {
    ctx_->make_current();
    ncclUniqueId id;
    std::memcpy(&id, uid, sizeof(id));
    NZ_NCCL_CHECK(ncclCommInitRank(&comm_, nranks, id, rank));
    NZ_LOG_INFO("RCCL comm initialized: rank %d/%d on device %d", rank, nranks,
                ctx->device_index());
    // Failure-detection watchdog: a dead peer must not
    // hang the node — poll ncclCommGetAsyncError and abort+exit on error.
    if (nranks > 1) {
      watchdog_ = std::thread([this] {
        while (!stop_.load(std::memory_order_relaxed)) {
          ncclResult_t async = ncclSuccess;
          if (ncclCommGetAsyncError(comm_, &async) == ncclSuccess &&
              async != ncclSuccess && async != ncclInProgress) {
            NZ_LOG_ERROR("RCCL async error on rank %d: %s — aborting comm",
                         rank_, ncclGetErrorString(async));
            (void)ncclCommAbort(comm_);
            _exit(3);
          }
          std::this_thread::sleep_for(std::chrono::milliseconds(200));
        }
      });
    }
  }
  ~RcclComm() override {
    stop_.store(true);
    if (watchdog_.joinable()) watchdog_.join();
    if (comm_) (void)ncclCommDestroy(comm_);
  }
  int rank() const override { return rank_; }
  int nranks() const override { return nranks_; }
  bool capturable() const override { return true; }

  void all_reduce(void* data, int64_t count, DType dt, ReduceOp op,
                  Stream* stream) override {
    hipStream_t s = stream ? stream->handle() : ctx_->compute_stream()->handle();
    NZ_NCCL_CHECK(ncclAllReduce(data, data, (size_t)count, to_nccl(dt),
                                op == ReduceOp::SUM ? ncclSum : ncclMax, comm_, s));
  }

  void broadcast(void* data, int64_t count, DType dt, int root,
                 Stream* stream) override {
    hipStream_t s = stream ? stream->handle() : ctx_->compute_stream()->handle();
    NZ_NCCL_CHECK(ncclBroadcast(data, data, (size_t)count, to_nccl(dt), root,
                                comm_, s));
  }

  void barrier() override {
    // All-reduce of one scalar on the compute stream + host sync.
    float v = all_reduce_scalar(0.f, ReduceOp::SUM);
    (void)v;
  }

  float all_reduce_scalar(float v, ReduceOp op) override {
    if (!scratch_.defined()) scratch_ = Tensor(ctx_, DType::F32, Shape{1});
    scratch_.upload(&v, 4);
    all_reduce(scratch_.data(), 1, DType::F32, op, ctx_->compute_stream());
    float out = 0.f;
    scratch_.download(&out, 4);
    return out;
  }

 private:
  Context* ctx_;
  int nranks_, rank_;
  ncclComm_t comm_ = nullptr;
  Tensor scratch_;
  std::thread watchdog_;
  std::atomic<bool> stop_{false};
};

}  // namespace

std::unique_ptr<Comm> make_rccl_comm(Context* ctx, int nranks, int rank,
                                     const void* uid) {
  return std::make_unique<RcclComm>(ctx, nranks, rank, uid);
}

}  // namespace nz
