// Distributed communication backend.
// Thin Comm interface with exactly three impls:
//   RcclComm — the product path: RCCL over xGMI (native rccl/rccl.h, C7).
//   FakeComm — in-process N-rank loopback over host buffers (test tier T4:
//              exercises 8-rank DDP logic on the CPU host, no cluster).
//   TcpComm  — socket star over localhost for true multi-process CPU tests.
// All RCCL calls in the codebase live in rccl_comm.cc.
#pragma once

#include <memory>
#include <vector>

#include "src/core/tensor.h"

namespace nz {

enum class ReduceOp { SUM, MAX };

class Comm {
 public:
  virtual ~Comm() = default;
  virtual int rank() const = 0;
  virtual int nranks() const = 0;
  // In-place all-reduce of `count` elements at `data` (device ptr for RCCL,
  // host ptr for Fake/Tcp). `stream` may be null (synchronous backends).
  virtual void all_reduce(void* data, int64_t count, DType dt, ReduceOp op,
                          Stream* stream) = 0;
  virtual void broadcast(void* data, int64_t count, DType dt, int root,
                         Stream* stream) = 0;
  virtual void barrier() = 0;
  // True if this backend's collectives may be recorded into a hipGraph
  // (stream-ordered device work only — RCCL qualifies; the host-staged
  // test comms do not).
  virtual bool capturable() const { return false; }
  // Host-scalar convenience (stages through the device for RCCL).
  virtual float all_reduce_scalar(float v, ReduceOp op) = 0;
};

// RCCL (GPU). uid: ncclUniqueId bytes (see uid_bytes()).
int comm_uid_bytes();
void comm_get_uid(void* out);  // rank 0 calls, others receive the bytes
std::unique_ptr<Comm> make_rccl_comm(Context* ctx, int nranks, int rank,
                                     const void* uid);

// In-process fake comm: returns n connected instances (threads share state).
// device_buffers=true stages device pointers through the host around the
// reduce (synchronous w.r.t. the passed stream) — lets N simulated ranks
// run REAL GPU DDP on one device (test tier T4 on hardware).
std::vector<std::unique_ptr<Comm>> make_fake_comm_group(int nranks,
                                                        bool device_buffers = false);

// TCP star over localhost (CPU tensors / host scalars only).
std::unique_ptr<Comm> make_tcp_comm(int nranks, int rank, const char* host, int port);

}  // namespace nz
