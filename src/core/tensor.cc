#include "src/core/tensor.h"

#include <hip/hip_runtime.h>

namespace nz {

#define NZ_HIP_CHECK(expr)                                                   \
  do {                                                                       \
    hipError_t nz_hip_err_ = (expr);                                                  \
    if (nz_hip_err_ != hipSuccess)                                                    \
      ::nz::fail(std::string("HIP error: ") + hipGetErrorString(nz_hip_err_) +        \
                 " in " #expr " at " + __FILE__ + ":" + std::to_string(__LINE__)); \
  } while (0)

// ------------------------------------------------------------- allocator ---
Arena::Arena(int device_index) : device_index_(device_index) {}

Arena::~Arena() {
  // Free the cached (unreferenced) blocks; live blocks are owned by
  // still-alive tensors, whose Buf frees into a dead arena otherwise —
  // contexts must outlive their tensors.
  std::lock_guard<std::mutex> lk(mu_);
  for (auto& [sz, lst] : free_lists_) {
    for (void* q : lst) {
      if (device_index_ < 0) ::free(q);
      else (void)hipFree(q);
      sizes_.erase(q);
    }
    lst.clear();
  }
  cached_ = 0;
}

void* Arena::raw_alloc(size_t bytes) {
  if (device_index_ < 0) {  // CPU
    void* p = nullptr;
    if (posix_memalign(&p, 256, bytes) != 0) return nullptr;
    return p;
  }
  NZ_HIP_CHECK(hipSetDevice(device_index_));
  void* p = nullptr;
  if (hipMalloc(&p, bytes) != hipSuccess) { (void)hipGetLastError(); return nullptr; }
  return p;
}

void* Arena::alloc(size_t bytes) {
  static const bool dbg = [] {
    const char* v = getenv("NEZHA_DEBUG_ALLOC");
    return v && v[0] == '1';
  }();
  if (bytes == 0) bytes = 16;
  bytes = (bytes + 255) & ~size_t(255);  // 256B granularity -> reuse buckets
  std::lock_guard<std::mutex> lk(mu_);
  if (dbg) {
    auto itd = free_lists_.find(bytes);
    bool hit = itd != free_lists_.end() && !itd->second.empty();
    fprintf(stderr, "[arena] alloc %zu %s\n", bytes, hit ? "hit" : "MISS");
  }
  auto it = free_lists_.find(bytes);
  if (it != free_lists_.end() && !it->second.empty()) {
    void* p = it->second.back();
    it->second.pop_back();
    cached_ -= bytes;
    in_use_ += bytes;
    if (in_use_ > high_water_) high_water_ = in_use_;
    return p;
  }
  void* p = raw_alloc(bytes);
  if (!p) {
    // Trim the cache and retry once.
    for (auto& [sz, lst] : free_lists_) {
      for (void* q : lst) {
        if (device_index_ < 0) ::free(q); else (void)hipFree(q);
        sizes_.erase(q);
      }
      cached_ -= sz * lst.size();
      lst.clear();
    }
    p = raw_alloc(bytes);
    NZ_CHECK(p != nullptr, "out of memory allocating " + std::to_string(bytes) + " bytes");
  }
  sizes_[p] = bytes;
  in_use_ += bytes;
  if (in_use_ > high_water_) high_water_ = in_use_;
  return p;
}

void Arena::free(void* p) {
  if (!p) return;
  static const bool poison = [] {  // surface use-after-free
    const char* v = getenv("NEZHA_DEBUG_POISON");
    return v && v[0] == '1';
  }();
  std::lock_guard<std::mutex> lk(mu_);
  auto it = sizes_.find(p);
  NZ_CHECK(it != sizes_.end(), "free of unknown pointer");
  size_t bytes = it->second;
  if (poison) {
    if (device_index_ < 0)
      std::memset(p, 0xCD, bytes);  // bf16 0xCDCD / f32 0xCDCDCDCD = huge negatives
    else
      (void)hipMemset(p, 0xCD, bytes);
  }
  in_use_ -= bytes;
  cached_ += bytes;
  free_lists_[bytes].push_back(p);
}

// --------------------------------------------------------------- streams ---
Stream::Stream(int device_index) : device_index_(device_index) {
  if (device_index_ >= 0) {
    NZ_HIP_CHECK(hipSetDevice(device_index_));
    NZ_HIP_CHECK(hipStreamCreateWithFlags(&s_, hipStreamNonBlocking));
  }
}
Stream::~Stream() {
  if (s_) (void)hipStreamDestroy(s_);
}
void Stream::sync() const {
  if (s_) NZ_HIP_CHECK(hipStreamSynchronize(s_));
}

Event::Event() = default;
Event::~Event() {
  if (e_) (void)hipEventDestroy(e_);
}
void Event::record(Stream& s) {
  if (!e_) NZ_HIP_CHECK(hipEventCreate(&e_));
  NZ_HIP_CHECK(hipEventRecord(e_, s.handle()));
}
void Event::wait(Stream& s) const {
  NZ_CHECK(e_, "wait on unrecorded event");
  NZ_HIP_CHECK(hipStreamWaitEvent(s.handle(), e_, 0));
}
void Event::sync() const {
  NZ_CHECK(e_, "sync on unrecorded event");
  NZ_HIP_CHECK(hipEventSynchronize(e_));
}
float Event::elapsed_ms_since(const Event& start) const {
  float ms = 0.f;
  NZ_HIP_CHECK(hipEventElapsedTime(&ms, start.e_, e_));
  return ms;
}

// --------------------------------------------------------------- context ---
int hip_device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) { (void)hipGetLastError(); return 0; }
  return n;
}

Context::Context(Device dev, int device_index) : dev_(dev), index_(device_index) {
  if (dev_ == Device::HIP) {
    NZ_CHECK(hip_device_count() > device_index, "HIP device " + std::to_string(device_index) + " not available");
    NZ_HIP_CHECK(hipSetDevice(index_));
    arena_ = std::make_unique<Arena>(index_);
    compute_ = std::make_unique<Stream>(index_);
    comm_ = std::make_unique<Stream>(index_);
  } else {
    arena_ = std::make_unique<Arena>(-1);
    compute_ = std::make_unique<Stream>(-1);
    comm_ = std::make_unique<Stream>(-1);
  }
}
Context::~Context() = default;

void Context::sync() {
  if (dev_ == Device::HIP) {
    make_current();
    NZ_HIP_CHECK(hipDeviceSynchronize());
  }
}
void Context::make_current() {
  if (dev_ == Device::HIP) NZ_HIP_CHECK(hipSetDevice(index_));
}

// ---------------------------------------------------------------- tensor ---
Tensor::Buf::~Buf() {
  if (p && ctx) ctx->arena()->free(p);
}

Tensor::Tensor(Context* ctx, DType dt, Shape shape)
    : ctx_(ctx), dtype_(dt), shape_(std::move(shape)) {
  for (int64_t d : shape_.dims)
    NZ_CHECK(d >= 0, "negative tensor dimension " + std::to_string(d));
  NZ_CHECK(shape_.numel() >= 0 && shape_.numel() < (int64_t(1) << 40),
           "tensor numel out of range");
  buf_ = std::make_shared<Buf>();
  buf_->ctx = ctx;
  buf_->p = ctx->arena()->alloc(bytes());
  raw_ptr_ = buf_->p;
  offset_bytes_ = 0;
}

Tensor Tensor::view(const Tensor& base, DType dt, Shape shape, int64_t elem_offset) {
  Tensor t;
  t.ctx_ = base.ctx_;
  t.dtype_ = dt;
  t.shape_ = std::move(shape);
  t.buf_ = base.buf_;
  t.raw_ptr_ = base.raw_ptr_;
  t.offset_bytes_ = base.offset_bytes_ + size_t(elem_offset) * dtype_size(dt);
  NZ_CHECK(t.offset_bytes_ + t.bytes() <= base.offset_bytes_ + base.bytes(),
           "view out of range");
  return t;
}

void Tensor::upload(const void* host_src, size_t n) {
  NZ_CHECK(n == bytes(), "upload size mismatch: " + std::to_string(n) + " vs " + std::to_string(bytes()));
  if (device() == Device::CPU) {
    std::memcpy(data(), host_src, n);
  } else {
    ctx_->make_current();
    // The compute stream is non-blocking w.r.t. the null stream: sync it
    // first so queued kernels (e.g. init) cannot overwrite the upload.
    ctx_->compute_stream()->sync();
    NZ_HIP_CHECK(hipMemcpy(data(), host_src, n, hipMemcpyHostToDevice));
  }
}

void Tensor::download(void* host_dst, size_t n) const {
  NZ_CHECK(n == bytes(), "download size mismatch");
  if (device() == Device::CPU) {
    std::memcpy(host_dst, data(), n);
  } else {
    ctx_->make_current();
    ctx_->compute_stream()->sync();
    NZ_HIP_CHECK(hipMemcpy(host_dst, data(), n, hipMemcpyDeviceToHost));
  }
}

}  // namespace nz
