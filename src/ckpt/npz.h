// Checkpointing: genuine .npz = a zip archive of
// numpy-format-v1.0 .npy members, stored (uncompressed), zlib crc32.
// Interop contract: numpy's np.load must read our files and vice versa
// (bf16 tensors are stored with descr '<u2' — raw bf16 bit patterns).
#pragma once

#include <map>
#include <string>
#include <vector>

#include "src/core/common.h"

namespace nz {

struct NpyArray {
  DType dtype;                  // F32 <-> '<f4', BF16 <-> '<u2'
  std::vector<int64_t> shape;
  std::vector<char> data;       // raw C-order payload
  int64_t numel() const {
    int64_t n = 1;
    for (auto d : shape) n *= d;
    return n;
  }
};

// Write members in map order (byte-stable for a given input).
void npz_save(const std::string& path, const std::map<std::string, NpyArray>& arrays);
std::map<std::string, NpyArray> npz_load(const std::string& path);

}  // namespace nz
