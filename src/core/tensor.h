// Tensor engine: strided-dense tensors, caching HIP allocator, streams.
// SURVEY.md §2.1 components #1 (Tensor core), #2 (Allocator), #3 (Streams).
#pragma once

#include <memory>
#include <vector>
#include <mutex>
#include <unordered_map>

#include "src/core/common.h"

// Forward-declare HIP types so this header stays includable from pure host
// translation units; the .cc pulls in <hip/hip_runtime.h>.
struct ihipStream_t;
struct ihipEvent_t;
typedef struct ihipStream_t* hipStream_t;
typedef struct ihipEvent_t* hipEvent_t;

namespace nz {

// ----------------------------------------------------------------- shape ---
struct Shape {
  std::vector<int64_t> dims;
  Shape() = default;
  Shape(std::initializer_list<int64_t> d) : dims(d) {}
  explicit Shape(std::vector<int64_t> d) : dims(std::move(d)) {}
  int64_t numel() const {
    int64_t n = 1;
    for (auto d : dims) n *= d;
    return n;
  }
  int ndim() const { return (int)dims.size(); }
  int64_t operator[](int i) const { return dims[i]; }
  bool operator==(const Shape& o) const { return dims == o.dims; }
  std::string str() const {
    std::string s = "[";
    for (size_t i = 0; i < dims.size(); ++i)
      s += std::to_string(dims[i]) + (i + 1 < dims.size() ? "," : "");
    return s + "]";
  }
};

// ------------------------------------------------------------- allocator ---
// Caching device allocator: hipMalloc is slow and synchronizing, so freed
// blocks are kept in per-size free lists and reused. Sized against 288 GB
// HBM3E (C6): we never proactively trim unless allocation fails.
class Arena {
 public:
  explicit Arena(int device_index);
  ~Arena();
  void* alloc(size_t bytes);
  void free(void* p);
  size_t high_water() const { return high_water_; }
  size_t in_use() const { return in_use_; }
  int device_index() const { return device_index_; }

 private:
  int device_index_;
  std::mutex mu_;
  std::unordered_map<void*, size_t> sizes_;                 // live + cached
  std::unordered_map<size_t, std::vector<void*>> free_lists_;
  size_t in_use_ = 0, cached_ = 0, high_water_ = 0;
  void* raw_alloc(size_t bytes);
};

// --------------------------------------------------------------- streams ---
// One compute stream + one comm stream per device context (C14 overlap).
class Stream {
 public:
  explicit Stream(int device_index);
  ~Stream();
  hipStream_t handle() const { return s_; }
  void sync() const;

 private:
  int device_index_;
  hipStream_t s_ = nullptr;
};

class Event {
 public:
  Event();
  ~Event();
  Event(const Event&) = delete;             // owns a hipEvent_t
  Event& operator=(const Event&) = delete;
  Event(Event&& o) noexcept : e_(o.e_) { o.e_ = nullptr; }
  Event& operator=(Event&& o) noexcept {
    if (this != &o) {
      this->~Event();
      e_ = o.e_;
      o.e_ = nullptr;
    }
    return *this;
  }
  void record(Stream& s);
  void wait(Stream& s) const;  // makes s wait on this event (device-side)
  void sync() const;
  float elapsed_ms_since(const Event& start) const;

 private:
  hipEvent_t e_ = nullptr;
};

// --------------------------------------------------------------- context ---
// A Context owns the device, its arena and its streams. CPU contexts have
// no HIP state.
class Context {
 public:
  Context(Device dev, int device_index);
  ~Context();
  Device device() const { return dev_; }
  int device_index() const { return index_; }
  Arena* arena() { return arena_.get(); }
  Stream* compute_stream() { return compute_.get(); }
  Stream* comm_stream() { return comm_.get(); }
  void sync();
  void make_current();  // hipSetDevice for HIP contexts

 private:
  Device dev_;
  int index_;
  std::unique_ptr<Arena> arena_;
  std::unique_ptr<Stream> compute_, comm_;
};

int hip_device_count();  // 0 when no GPU / no HIP runtime

// ---------------------------------------------------------------- tensor ---
// Dense row-major contiguous tensor owning (or viewing) a device buffer.
class Tensor {
 public:
  Tensor() = default;  // null tensor
  Tensor(Context* ctx, DType dt, Shape shape);

  static Tensor view(const Tensor& base, DType dt, Shape shape, int64_t elem_offset);

  bool defined() const { return buf_ != nullptr; }
  DType dtype() const { return dtype_; }
  Device device() const { return ctx_ ? ctx_->device() : Device::CPU; }
  Context* ctx() const { return ctx_; }
  const Shape& shape() const { return shape_; }
  int64_t numel() const { return shape_.numel(); }
  size_t bytes() const { return size_t(numel()) * dtype_size(dtype_); }
  void* data() const { return (char*)raw_ptr_ + offset_bytes_; }

  // host <-> device
  void upload(const void* host_src, size_t bytes);
  void download(void* host_dst, size_t bytes) const;

  // 2D helpers (row-major)
  int64_t rows() const { NZ_CHECK(shape_.ndim() >= 1, "rows on 0-d"); return numel() / shape_.dims.back(); }
  int64_t cols() const { NZ_CHECK(shape_.ndim() >= 1, "cols on 0-d"); return shape_.dims.back(); }

 private:
  struct Buf {  // shared ownership of the allocation
    Context* ctx = nullptr;
    void* p = nullptr;
    ~Buf();
  };
  Context* ctx_ = nullptr;
  DType dtype_ = DType::F32;
  Shape shape_;
  std::shared_ptr<Buf> buf_;
  void* raw_ptr_ = nullptr;
  size_t offset_bytes_ = 0;
};

}  // namespace nz
