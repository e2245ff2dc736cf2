// T1 unit tests (C++): tensor/shape/view/arena/npz/flags.
#include <cassert>
#include <cstdio>

#include "src/ckpt/npz.h"
#include "src/core/ops.h"
#include "src/util/util.h"

using namespace nz;

static void test_shape_tensor() {
  Context ctx(Device::CPU, 0);
  Tensor t(&ctx, DType::F32, Shape{4, 8});
  assert(t.numel() == 32);
  assert(t.rows() == 4 && t.cols() == 8);
  ops::fill(t, 2.0f);
  float host[32];
  t.download(host, sizeof(host));
  for (float v : host) assert(v == 2.0f);

  Tensor v2 = Tensor::view(t, DType::F32, Shape{8}, 8);
  ops::fill(v2, 7.0f);
  t.download(host, sizeof(host));
  for (int i = 0; i < 8; ++i) assert(host[8 + i] == 7.0f);
  for (int i = 16; i < 32; ++i) assert(host[i] == 2.0f);
  printf("shape/tensor/view OK\n");
}

static void test_arena_reuse() {
  Context ctx(Device::CPU, 0);
  void* p1 = ctx.arena()->alloc(1 << 20);
  ctx.arena()->free(p1);
  void* p2 = ctx.arena()->alloc(1 << 20);
  assert(p1 == p2);  // cached block reused
  ctx.arena()->free(p2);
  assert(ctx.arena()->high_water() >= (1 << 20));
  printf("arena reuse OK\n");
}

static void test_npz_roundtrip() {
  std::map<std::string, NpyArray> in;
  NpyArray a;
  a.dtype = DType::F32;
  a.shape = {2, 3};
  a.data.resize(24);
  float vals[6] = {1, 2, 3, 4, 5, 6};
  memcpy(a.data.data(), vals, 24);
  in["x/y"] = a;
  NpyArray b;
  b.dtype = DType::BF16;
  b.shape = {4};
  b.data.resize(8);
  uint16_t bv[4] = {f32_to_bf16(1.5f), f32_to_bf16(-2.f), 0, 0x3f80};
  memcpy(b.data.data(), bv, 8);
  in["b"] = b;
  npz_save("/tmp/nz_test.npz", in);
  auto out = npz_load("/tmp/nz_test.npz");
  assert(out.size() == 2);
  assert(out["x/y"].shape == std::vector<int64_t>({2, 3}));
  assert(memcmp(out["x/y"].data.data(), vals, 24) == 0);
  assert(out["b"].dtype == DType::BF16);
  assert(memcmp(out["b"].data.data(), bv, 8) == 0);
  printf("npz roundtrip OK\n");
}

static void test_flags() {
  Flags f;
  const char* argv[] = {"--model=mlp", "--gpus", "4", "--verbose"};
  f.parse(4, argv);
  assert(f.get_str("model", "") == "mlp");
  assert(f.get_int("gpus", 0) == 4);
  assert(f.get_bool("verbose", false));
  assert(f.get_int("missing", 42) == 42);
  printf("flags OK\n");
}

int main() {
  test_shape_tensor();
  test_arena_reuse();
  test_npz_roundtrip();
  test_flags();
  printf("test_core: ALL OK\n");
  return 0;
}
