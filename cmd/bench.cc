// nezha-bench: kernel microbenchmarks — per-kernel TFLOP/s
// / GB/s on the contract shapes, for the M2 tuning loop and rocprof runs.
#include <cstdio>
#include <memory>
#include <string>

#include "src/core/ops.h"
#include "src/util/util.h"

using namespace nz;

static double time_op(Context& ctx, int iters, const std::function<void()>& fn) {
  fn();  // warm
  fn();
  ctx.sync();
  double t0 = now_sec();
  for (int i = 0; i < iters; ++i) fn();
  ctx.sync();
  return (now_sec() - t0) / iters;
}

static void bench_gemm(Context& ctx, int64_t M, int64_t N, int64_t K,
                       bool tA, bool tB, const char* tag, int iters) {
  Tensor A(&ctx, DType::BF16, tA ? Shape{K, M} : Shape{M, K});
  Tensor B(&ctx, DType::BF16, tB ? Shape{N, K} : Shape{K, N});
  Tensor C(&ctx, DType::BF16, Shape{M, N});
  ops::rand_normal(A, 1, 0.f, 1.f);  // random data (guide §5.4 rule 25)
  ops::rand_normal(B, 2, 0.f, 1.f);
  double sec = time_op(ctx, iters, [&] { ops::gemm(A, B, C, tA, tB, 1.f, 0.f); });
  double tf = 2.0 * M * N * K / sec / 1e12;
  printf("gemm %-22s M=%-6lld N=%-6lld K=%-6lld  %8.3f ms  %7.1f TF/s\n", tag,
         (long long)M, (long long)N, (long long)K, sec * 1e3, tf);
}

static void bench_mem(Context& ctx, const char* name, int64_t n, double bytes,
                      const std::function<void()>& fn, int iters) {
  double sec = time_op(ctx, iters, fn);
  printf("%-28s n=%-10lld %8.3f ms  %7.2f TB/s\n", name, (long long)n,
         sec * 1e3, bytes / sec / 1e12);
}

int main(int argc, const char** argv) {
  Flags flags;
  flags.parse(argc - 1, argv + 1);
  if (flags.get_bool("help", false)) {
    puts("nezha-bench — kernel microbenchmarks on the contract shapes\n"
         "  --op=all|gemm|mem|norm   which suite (default all)\n"
         "  --iters=N                timed iterations (20)\n"
         "  --batch=N                GEMM M dimension (8192)");
    return 0;
  }
  int iters = (int)flags.get_int("iters", 20);
  std::string which = flags.get_str("op", "all");
  NZ_CHECK(hip_device_count() > 0, "nezha-bench requires a GPU");
  Context ctx(Device::HIP, 0);

  int64_t B = flags.get_int("batch", 8192);
  if (which == "shape") {
    // one arbitrary GEMM shape: --m= --n= --k= [--ta=1] [--tb=1]
    bench_gemm(ctx, flags.get_int("m", 4096), flags.get_int("n", 4096),
               flags.get_int("k", 4096), flags.get_bool("ta", false),
               flags.get_bool("tb", false), "custom", iters);
    return 0;
  }
  if (which == "bert32k") {
    // the per-layer BERT-base linear GEMMs at the contract batch
    // (B=256, S=128 -> M=32768) — the shapes that dominate the step.
    bench_gemm(ctx, 32768, 2304, 768, false, false, "qkv-fwd (NN)", iters);
    bench_gemm(ctx, 32768, 768, 768, false, false, "out-fwd (NN)", iters);
    bench_gemm(ctx, 32768, 3072, 768, false, false, "ffn1-fwd (NN)", iters);
    bench_gemm(ctx, 32768, 768, 3072, false, false, "ffn2-fwd (NN)", iters);
    bench_gemm(ctx, 32768, 768, 2304, false, true, "qkv-dgrad (NT)", iters);
    bench_gemm(ctx, 32768, 768, 768, false, true, "out-dgrad (NT)", iters);
    bench_gemm(ctx, 32768, 3072, 768, false, true, "ffn1-dgrad (NT)", iters);
    bench_gemm(ctx, 32768, 768, 3072, false, true, "ffn2-dgrad (NT)", iters);
    bench_gemm(ctx, 768, 2304, 32768, true, false, "qkv-wgrad (TN)", iters);
    bench_gemm(ctx, 768, 3072, 32768, true, false, "ffn1-wgrad (TN)", iters);
    bench_gemm(ctx, 3072, 768, 32768, true, false, "ffn2-wgrad (TN)", iters);
    return 0;
  }
  if (which == "all" || which == "gemm") {
    // the MLP contract shapes (C11): fwd, dgrad, wgrad at batch 8192
    bench_gemm(ctx, B, 4096, 4096, false, false, "mlp-fwd (NN)", iters);
    bench_gemm(ctx, B, 4096, 4096, false, true, "mlp-dgrad (NT)", iters);
    bench_gemm(ctx, 4096, 4096, B, true, false, "mlp-wgrad (TN)", iters);
    // BERT shapes
    bench_gemm(ctx, 4096, 2304, 768, false, false, "bert-qkv (NN)", iters);
    bench_gemm(ctx, 4096, 3072, 768, false, false, "bert-ffn1 (NN)", iters);
    bench_gemm(ctx, 4096, 768, 3072, false, false, "bert-ffn2 (NN)", iters);
    // square reference points
    bench_gemm(ctx, 4096, 4096, 4096, false, false, "square-4k (NN)", iters);
    bench_gemm(ctx, 8192, 8192, 8192, false, false, "square-8k (NN)", 5);
  }
  if (which == "all" || which == "attn") {
    // fused attention pair at the contract shape (B=256, H=12, S=128, dh=64)
    int64_t Bb = 256, H = 12, S = 128, D = 64, HID = H * D;
    Tensor qkv(&ctx, DType::BF16, Shape{Bb * S, 3 * HID});
    Tensor probs(&ctx, DType::BF16, Shape{Bb * H * S, S});
    Tensor out(&ctx, DType::BF16, Shape{Bb * S, HID});
    Tensor dm(&ctx, DType::BF16, Shape{Bb * S, HID});
    Tensor dqkv(&ctx, DType::BF16, Shape{Bb * S, 3 * HID});
    ops::rand_normal(qkv, 11, 0.f, 0.5f);
    ops::rand_normal(dm, 12, 0.f, 0.5f);
    float sc = 0.125f;
    double fw = time_op(ctx, iters, [&] {
      ops::attn_fwd_packed(qkv, probs, out, Bb, S, H, D, sc);
    });
    double bw = time_op(ctx, iters, [&] {
      ops::attn_bwd_packed(qkv, probs, dm, dqkv, Bb, S, H, D, sc);
    });
    double fwf = 2.0 * Bb * H * (2.0 * S * S * D);            // QK^T + PV
    double bwf = 2.0 * Bb * H * (4.0 * S * S * D);            // dP,dV,dQ,dK
    printf("attn fwd  (fused, packed)  B=256 H=12 S=128 dh=64 %8.3f ms  %7.1f TF/s\n",
           fw * 1e3, fwf / fw / 1e12);
    printf("attn bwd  (fused, packed)  B=256 H=12 S=128 dh=64 %8.3f ms  %7.1f TF/s\n",
           bw * 1e3, bwf / bw / 1e12);
  }
  if (which == "all" || which == "mem") {
    int64_t n = 256 << 20;  // 256M bf16 elements = 512 MB
    Tensor x(&ctx, DType::BF16, Shape{n});
    Tensor y(&ctx, DType::BF16, Shape{n});
    ops::rand_normal(x, 3, 0.f, 1.f);
    ops::rand_normal(y, 4, 0.f, 1.f);
    bench_mem(ctx, "add_inplace bf16", n, 3.0 * n * 2,
              [&] { ops::add_inplace(x, y, 1.0f); }, iters);
    Tensor z(&ctx, DType::BF16, Shape{1 << 22, 64});
    Tensor zt(&ctx, DType::BF16, Shape{64, 1 << 22});
    bench_mem(ctx, "transpose2d bf16", z.numel(), 2.0 * z.numel() * 2,
              [&] { ops::transpose2d(z, zt); }, iters);
    Tensor z2(&ctx, DType::BF16, Shape{65536, 3072});
    Tensor z2t(&ctx, DType::BF16, Shape{3072, 65536});
    bench_mem(ctx, "transpose2d bf16 (wgrad shape)", z2.numel(),
              2.0 * z2.numel() * 2, [&] { ops::transpose2d(z2, z2t); }, iters);
    // adam on a 64 MB bucket
    int64_t bn = 32 << 20;
    Tensor p(&ctx, DType::BF16, Shape{bn}), g(&ctx, DType::BF16, Shape{bn});
    Tensor ma(&ctx, DType::F32, Shape{bn}), m(&ctx, DType::F32, Shape{bn}),
        v(&ctx, DType::F32, Shape{bn});
    ops::rand_normal(p, 5, 0.f, 1.f);
    ops::rand_normal(g, 6, 0.f, 1.f);
    ops::fill(ma, 0.f); ops::fill(m, 0.f); ops::fill(v, 0.f);
    bench_mem(ctx, "adam_step (64MB bucket)", bn, (2. + 4 * 3) * bn + (2. + 4 * 3) * bn,
              [&] { ops::adam_step(p, g, ma, m, v, 1e-4f, .9f, .999f, 1e-8f, 1, 1.f, 0.f); },
              iters);
  }
  if (which == "all" || which == "norm") {
    Tensor x(&ctx, DType::BF16, Shape{8192, 4096});
    Tensor y(&ctx, DType::BF16, Shape{8192, 4096});
    Tensor gma(&ctx, DType::BF16, Shape{4096}), be(&ctx, DType::BF16, Shape{4096});
    Tensor mean(&ctx, DType::F32, Shape{8192}), rstd(&ctx, DType::F32, Shape{8192});
    ops::rand_normal(x, 7, 0.f, 1.f);
    ops::fill(gma, 1.f); ops::fill(be, 0.f);
    bench_mem(ctx, "layernorm_fwd 8192x4096", x.numel(), 2.0 * x.numel() * 2,
              [&] { ops::layernorm_fwd(x, gma, be, y, mean, rstd, 1e-5f); }, iters);
    bench_mem(ctx, "softmax_fwd 8192x4096", x.numel(), 2.0 * x.numel() * 2,
              [&] { ops::softmax_fwd(x, y, 1.0f); }, iters);
  }
  return 0;
}
