#include "src/util/prof.h"

#include <hip/hip_runtime.h>

#include <cstdio>
#include <mutex>
#include <vector>

#include "src/util/util.h"

namespace nz {
namespace prof {

bool trace_enabled() {
  static bool on = getenv("NEZHA_TRACE") != nullptr;
  return on;
}
bool debug_sync_enabled() {
  static bool on = [] {
    const char* v = getenv("NEZHA_DEBUG_SYNC");
    return v && v[0] == '1';
  }();
  return on;
}

namespace {
struct Record {
  const char* name;
  hipEvent_t start, stop;
};
std::mutex g_mu;
std::vector<Record> g_records;  // bounded ring
constexpr size_t kMaxRecords = 1 << 16;
}  // namespace

OpScope::OpScope(const char* name, Context* ctx) : name_(name), ctx_(ctx) {
  if (!trace_enabled() || !ctx_ || ctx_->device() != Device::HIP) return;
  Record r{name, nullptr, nullptr};
  (void)hipEventCreate(&r.start);
  (void)hipEventCreate(&r.stop);
  (void)hipEventRecord(r.start, ctx_->compute_stream()->handle());
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_records.size() < kMaxRecords) {
    g_records.push_back(r);
    slot_ = (int)g_records.size() - 1;
  } else {
    (void)hipEventDestroy(r.start);
    (void)hipEventDestroy(r.stop);
  }
}

OpScope::~OpScope() {
  if (slot_ >= 0) {
    std::lock_guard<std::mutex> lk(g_mu);
    (void)hipEventRecord(g_records[slot_].stop,
                         ctx_->compute_stream()->handle());
  }
  if (debug_sync_enabled() && ctx_ && ctx_->device() == Device::HIP) {
    hipError_t e = hipDeviceSynchronize();
    if (e != hipSuccess)
      fail(std::string("NEZHA_DEBUG_SYNC: fault after op '") + name_ +
           "': " + hipGetErrorString(e));
  }
}

void flush(Context* ctx) {
  if (!trace_enabled()) return;
  const char* path = getenv("NEZHA_TRACE");
  std::lock_guard<std::mutex> lk(g_mu);
  if (g_records.empty()) return;
  if (ctx) ctx->sync();
  FILE* f = fopen(path, "w");
  if (!f) return;
  fprintf(f, "[\n");
  // timestamps relative to the first op's start event
  hipEvent_t base = g_records[0].start;
  bool first = true;
  for (auto& r : g_records) {
    float t0 = 0.f, dur = 0.f;
    if (hipEventElapsedTime(&t0, base, r.start) != hipSuccess) continue;
    if (hipEventElapsedTime(&dur, r.start, r.stop) != hipSuccess) continue;
    fprintf(f,
            "%s{\"name\":\"%s\",\"ph\":\"X\",\"ts\":%.3f,\"dur\":%.3f,"
            "\"pid\":0,\"tid\":0}",
            first ? "" : ",\n", r.name, t0 * 1e3, dur * 1e3);
    first = false;
  }
  fprintf(f, "\n]\n");
  fclose(f);
  NZ_LOG_INFO("wrote chrome trace (%zu op scopes) to %s", g_records.size(), path);
  for (auto& r : g_records) {
    (void)hipEventDestroy(r.start);
    (void)hipEventDestroy(r.stop);
  }
  g_records.clear();
}

}  // namespace prof
}  // namespace nz
