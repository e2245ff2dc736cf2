// FakeComm: in-process loopback Comm for N simulated
// ranks running as threads over host (CPU) buffers. The DDP engine sees
// only the Comm interface, so this exercises real 8-rank bucketing/overlap
// logic on the 8-CPU host with no cluster.
#include <hip/hip_runtime.h>

#include <condition_variable>
#include <mutex>

#include "src/comm/comm.h"

namespace nz {
namespace {

struct FakeState {
  int nranks;
  std::mutex mu;
  std::condition_variable cv;
  int arrived = 0;
  uint64_t generation = 0;
  uint64_t sig = 0;       // collective signature of this rendezvous round
  bool poisoned = false;  // a rank detected a signature mismatch
  std::vector<void*> ptrs;
  // Reusable sense barrier; `leader_fn` runs once (by the last arriver)
  // while everyone waits, then all are released.
  explicit FakeState(int n) : nranks(n), ptrs(n, nullptr) {}

  // Like RCCL, every rank must issue the SAME collective sequence. Each
  // call carries a signature (kind/op/dtype/count); the first arriver of a
  // round pins it and any mismatching later arrival fails ALL ranks loudly
  // instead of silently reducing mismatched buffers (the bug class this
  // models: DDP buckets launched in different orders on different ranks).
  template <typename Fn>
  void rendezvous(int rank, void* ptr, uint64_t s, Fn leader_fn) {
    std::unique_lock<std::mutex> lk(mu);
    NZ_CHECK(!poisoned, "FakeComm poisoned by earlier collective mismatch");
    if (arrived == 0) {
      sig = s;
    } else if (s != sig) {
      poisoned = true;
      cv.notify_all();
      NZ_CHECK(false, "FakeComm collective mismatch across ranks (different "
                      "op/dtype/count — bucket launch order diverged?)");
    }
    ptrs[rank] = ptr;
    uint64_t gen = generation;
    if (++arrived == nranks) {
      leader_fn(ptrs);
      arrived = 0;
      ++generation;
      cv.notify_all();
    } else {
      cv.wait(lk, [&] { return generation != gen || poisoned; });
      NZ_CHECK(!poisoned, "FakeComm poisoned by collective mismatch");
    }
  }
};

// signature: kind(4b) | op(4b) | dtype(4b) | count(52b)
inline uint64_t coll_sig(int kind, ReduceOp op, DType dt, int64_t count) {
  return (uint64_t(kind) << 60) | (uint64_t((int)op) << 56) |
         (uint64_t((int)dt) << 52) | (uint64_t(count) & 0xfffffffffffffull);
}

template <typename T>
void reduce_buffers(std::vector<void*>& ptrs, int64_t count, ReduceOp op,
                    bool bf16) {
  // Sum (or max) all buffers into buffer 0 in f32, then distribute.
  std::vector<float> acc(count);
  for (int r = 0; r < (int)ptrs.size(); ++r) {
    for (int64_t i = 0; i < count; ++i) {
      float v;
      if (bf16) v = bf16_to_f32(((const uint16_t*)ptrs[r])[i]);
      else v = ((const float*)ptrs[r])[i];
      if (r == 0) acc[i] = v;
      else acc[i] = op == ReduceOp::SUM ? acc[i] + v : std::max(acc[i], v);
    }
  }
  for (int r = 0; r < (int)ptrs.size(); ++r) {
    for (int64_t i = 0; i < count; ++i) {
      if (bf16) ((uint16_t*)ptrs[r])[i] = f32_to_bf16(acc[i]);
      else ((float*)ptrs[r])[i] = acc[i];
    }
  }
}

class FakeComm : public Comm {
 public:
  FakeComm(std::shared_ptr<FakeState> st, int rank, bool device)
      : st_(std::move(st)), rank_(rank), device_(device) {}
  int rank() const override { return rank_; }
  int nranks() const override { return st_->nranks; }

  void all_reduce(void* data, int64_t count, DType dt, ReduceOp op,
                  Stream* stream) override {
    if (device_ && stream) stream->sync();  // grads ready on the device
    st_->rendezvous(rank_, data, coll_sig(1, op, dt, count),
                    [&](std::vector<void*>& ptrs) {
      size_t bytes = size_t(count) * dtype_size(dt);
      if (!device_) {
        reduce_buffers<float>(ptrs, count, op, dt == DType::BF16);
        return;
      }
      // stage every rank's device buffer through the host, reduce, restore
      std::vector<std::vector<char>> host(ptrs.size());
      std::vector<void*> hptrs(ptrs.size());
      for (size_t i = 0; i < ptrs.size(); ++i) {
        host[i].resize(bytes);
        NZ_CHECK(hipMemcpy(host[i].data(), ptrs[i], bytes,
                           hipMemcpyDeviceToHost) == hipSuccess,
                 "FakeComm D2H staging copy failed");
        hptrs[i] = host[i].data();
      }
      reduce_buffers<float>(hptrs, count, op, dt == DType::BF16);
      for (size_t i = 0; i < ptrs.size(); ++i)
        NZ_CHECK(hipMemcpy(ptrs[i], host[i].data(), bytes,
                           hipMemcpyHostToDevice) == hipSuccess,
                 "FakeComm H2D staging copy failed");
    });
  }

  void broadcast(void* data, int64_t count, DType dt, int root, Stream* stream) override {
    if (device_ && stream) stream->sync();
    size_t bytes = size_t(count) * dtype_size(dt);
    bool dev = device_;
    st_->rendezvous(rank_, data, coll_sig(2, ReduceOp::SUM, dt, count + root),
                    [&, root, bytes, dev](std::vector<void*>& ptrs) {
      if (!dev) {
        for (int i = 0; i < (int)ptrs.size(); ++i)
          if (i != root) std::memcpy(ptrs[i], ptrs[root], bytes);
        return;
      }
      std::vector<char> host(bytes);
      NZ_CHECK(hipMemcpy(host.data(), ptrs[root], bytes,
                         hipMemcpyDeviceToHost) == hipSuccess,
               "FakeComm D2H broadcast copy failed");
      for (int i = 0; i < (int)ptrs.size(); ++i)
        if (i != root)
          NZ_CHECK(hipMemcpy(ptrs[i], host.data(), bytes,
                             hipMemcpyHostToDevice) == hipSuccess,
                   "FakeComm H2D broadcast copy failed");
    });
  }

  void barrier() override {
    st_->rendezvous(rank_, nullptr, coll_sig(3, ReduceOp::SUM, DType::F32, 0),
                    [](std::vector<void*>&) {});
  }

  float all_reduce_scalar(float v, ReduceOp op) override {
    // The scalar buffer is HOST memory even when this comm carries device
    // buffers — force the host reduce path (the device staging path would
    // hipMemcpy D2H from a host stack pointer and return garbage).
    float buf = v;
    st_->rendezvous(rank_, &buf, coll_sig(4, op, DType::F32, 1),
                    [&](std::vector<void*>& ptrs) {
      reduce_buffers<float>(ptrs, 1, op, /*bf16=*/false);
    });
    return buf;
  }

 private:
  std::shared_ptr<FakeState> st_;
  int rank_;
  bool device_;
};

}  // namespace

std::vector<std::unique_ptr<Comm>> make_fake_comm_group(int nranks,
                                                         bool device_buffers) {
  auto st = std::make_shared<FakeState>(nranks);
  std::vector<std::unique_ptr<Comm>> out;
  for (int r = 0; r < nranks; ++r)
    out.push_back(std::make_unique<FakeComm>(st, r, device_buffers));
  return out;
}

}  // namespace nz
