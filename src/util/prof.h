// Tracing/profiling hooks (component #20).
// Op-scope timers live in the dispatch layer (src/core/ops.cc), so every
// backend op is covered without per-kernel instrumentation. Two modes:
//  - NEZHA_TRACE=<path>: hipEvent pairs per op scope on the owning stream;
//    flushed as a chrome://tracing JSON by prof_flush() (called from
//    nz_ctx_sync / Trainer teardown). Zero cost when off.
//  - NEZHA_DEBUG_SYNC=1: device-synchronize + error-check after every op
//    (turns async kernel faults into attributed ones — SURVEY.md §5.2).
#pragma once

#include <string>

#include "src/core/tensor.h"

namespace nz {
namespace prof {

bool trace_enabled();
bool debug_sync_enabled();

// RAII op scope: records start/stop events on the tensor ctx's compute
// stream when tracing; on destruction optionally debug-syncs.
class OpScope {
 public:
  OpScope(const char* name, Context* ctx);
  ~OpScope();

 private:
  const char* name_;
  Context* ctx_;
  int slot_ = -1;
};

// Write accumulated trace (chrome://tracing format) to NEZHA_TRACE path.
void flush(Context* ctx);

}  // namespace prof
}  // namespace nz
