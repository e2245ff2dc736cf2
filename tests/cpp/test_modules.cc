// C++ unit test for the standalone module classes (Gelu, Softmax) and the
// tape-level ce_loss — the unfused clarity path.
#include <cassert>
#include <cmath>
#include <cstdio>
#include <vector>

#include "src/core/act.h"
#include "src/nn/modules.h"

using namespace nz;

static std::vector<float> host(const Tensor& t) {
  std::vector<float> v(t.numel());
  t.download(v.data(), v.size() * 4);
  return v;
}

int main() {
  Context ctx(Device::CPU, 0);
  Tape tape;

  // Gelu module forward/backward vs the scalar reference
  {
    Tensor x(&ctx, DType::F32, Shape{4, 8});
    ops::rand_normal(x, 3, 0.f, 1.f);
    auto xv = make_var(x);
    Gelu g(&ctx);
    auto y = g.forward(tape, xv);
    auto xs = host(x), ys = host(y->value);
    for (size_t i = 0; i < xs.size(); ++i)
      assert(std::abs(ys[i] - gelu_fwd_scalar(xs[i])) < 1e-5f);
    y->ensure_grad();
    ops::fill(y->grad, 1.0f);
    y->grad_init = true;
    tape.backward();
    auto gx = host(xv->grad);
    for (size_t i = 0; i < xs.size(); ++i)
      assert(std::abs(gx[i] - gelu_bwd_scalar(xs[i])) < 1e-5f);
    printf("Gelu module OK\n");
  }

  // Softmax module rows sum to 1; backward of sum(y) is ~0
  {
    Tensor x(&ctx, DType::F32, Shape{5, 16});
    ops::rand_normal(x, 7, 0.f, 2.f);
    auto xv = make_var(x);
    Softmax sm(&ctx, 0.5f);
    auto y = sm.forward(tape, xv);
    auto ys = host(y->value);
    for (int r = 0; r < 5; ++r) {
      float s = 0;
      for (int c = 0; c < 16; ++c) s += ys[r * 16 + c];
      assert(std::abs(s - 1.0f) < 1e-5f);
    }
    y->ensure_grad();
    ops::fill(y->grad, 1.0f);
    y->grad_init = true;
    tape.backward();
    auto gx = host(xv->grad);
    for (float v : gx) assert(std::abs(v) < 1e-5f);  // d sum(softmax)/dx = 0
    printf("Softmax module OK\n");
  }

  // ce_loss: uniform logits -> loss = log(C); grad rows sum to 0
  {
    Tensor logits(&ctx, DType::F32, Shape{6, 10});
    ops::fill(logits, 0.3f);
    Tensor labels(&ctx, DType::F32, Shape{6});
    ops::fill(labels, 4.0f);
    auto lv = make_var(logits);
    Tensor loss = ce_loss(tape, lv, labels);
    float l;
    loss.download(&l, 4);
    assert(std::abs(l - std::log(10.0f)) < 1e-5f);
    auto g = host(lv->grad);
    for (int r = 0; r < 6; ++r) {
      float s = 0;
      for (int c = 0; c < 10; ++c) s += g[r * 10 + c];
      assert(std::abs(s) < 1e-6f);
    }
    printf("ce_loss OK\n");
  }

  printf("test_modules: ALL OK\n");
  return 0;
}
