// T4: DDP logic against FakeComm — 8 simulated ranks as threads on CPU
// tensors. Validates bucketing, grad-ready hook ordering,
// param sync, and gradient AVERAGING semantics:
//   with SGD, params_ddp(1 step) == mean_r(params_single_r(1 step))
// because p' = p - lr * mean_r(g_r) and each single run gives p - lr*g_r.
#include <cassert>
#include <cmath>
#include <cstdio>
#include <thread>
#include <vector>

#include "src/comm/comm.h"
#include "src/models/models.h"
#include "src/train/trainer.h"

using namespace nz;

static constexpr int NRANKS = 8;
static constexpr int LAYERS = 2, DIM = 32, BATCH = 8;
static constexpr float LR = 0.01f;
static constexpr int64_t BUCKET = 4096;  // bytes -> several buckets

static std::unique_ptr<Trainer> make_trainer(Context* ctx) {
  auto model = std::make_unique<MlpModel>(ctx, LAYERS, DIM, BATCH, /*seed=*/42);
  return std::make_unique<Trainer>(ctx, std::move(model), Optimizer::Kind::SGD,
                                   LR, BUCKET, /*data_seed=*/0);
}

static std::vector<float> dump_params(Trainer& tr) {
  std::vector<float> out;
  for (auto& b : tr.optimizer()->buckets()) {
    std::vector<float> host(b.numel);
    b.data_flat.download(host.data(), host.size() * 4);
    out.insert(out.end(), host.begin(), host.end());
  }
  return out;
}

int main() {
  // --- single-rank references: one SGD step on rank-specific data ---------
  std::vector<std::vector<float>> single(NRANKS);
  for (int r = 0; r < NRANKS; ++r) {
    Context ctx(Device::CPU, 0);
    auto tr = make_trainer(&ctx);
    tr->reseed_data(777 + r);
    tr->step();
    single[r] = dump_params(*tr);
  }
  size_t n = single[0].size();
  std::vector<double> mean(n, 0.0);
  for (int r = 0; r < NRANKS; ++r)
    for (size_t i = 0; i < n; ++i) mean[i] += single[r][i] / NRANKS;

  // --- DDP run: 8 fake ranks in threads ------------------------------------
  auto comms = make_fake_comm_group(NRANKS);
  std::vector<std::vector<float>> ddp(NRANKS);
  std::vector<int64_t> launches(NRANKS, 0);
  std::vector<std::vector<int>> order(NRANKS);
  std::vector<std::thread> threads;
  for (int r = 0; r < NRANKS; ++r) {
    threads.emplace_back([&, r]() {
      Context ctx(Device::CPU, 0);
      auto tr = make_trainer(&ctx);
      tr->reseed_data(777 + r);
      tr->attach_ddp(comms[r].get());
      tr->step();
      ddp[r] = dump_params(*tr);
      launches[r] = tr->ddp()->comm_launches();
      order[r] = tr->ddp()->launch_order();
    });
  }
  for (auto& t : threads) t.join();

  // every rank issued its bucket all-reduces in the SAME order (the RCCL
  // ordering contract; a divergence here is silent corruption on hardware)
  for (int r = 1; r < NRANKS; ++r) assert(order[r] == order[0]);

  // all ranks identical
  for (int r = 1; r < NRANKS; ++r)
    for (size_t i = 0; i < n; ++i)
      assert(ddp[r][i] == ddp[0][i]);

  // equals the average-gradient step
  double max_err = 0;
  for (size_t i = 0; i < n; ++i)
    max_err = std::max(max_err, std::abs(ddp[0][i] - mean[i]));
  printf("ddp-vs-mean max err = %.3e (n=%zu)\n", max_err, n);
  assert(max_err < 1e-6);

  // multiple buckets actually exercised, one all-reduce per bucket
  {
    Context ctx(Device::CPU, 0);
    auto tr = make_trainer(&ctx);
    size_t nbuckets = tr->optimizer()->buckets().size();
    printf("buckets = %zu, comm launches = %lld\n", nbuckets,
           (long long)launches[0]);
    assert(nbuckets >= 2);
    assert(launches[0] == (int64_t)nbuckets);
  }

  // --- adversarial: mismatched collective sequences must fail loudly ------
  // Two ranks all-reduce different-size "buckets" concurrently (the bug
  // class: DDP buckets launched in diverging order across ranks). FakeComm
  // pins the first arriver's collective signature and poisons the group on
  // mismatch — BOTH ranks must see an Error, never a silent wrong reduce.
  {
    auto comms2 = make_fake_comm_group(2);
    std::vector<int> caught(2, 0);
    std::vector<std::thread> ths;
    for (int r = 0; r < 2; ++r) {
      ths.emplace_back([&, r]() {
        std::vector<float> buf(r == 0 ? 16 : 8, 1.0f);
        try {
          comms2[r]->all_reduce(buf.data(), (int64_t)buf.size(), DType::F32,
                                ReduceOp::SUM, nullptr);
        } catch (const Error&) {
          caught[r] = 1;
        }
      });
    }
    for (auto& t : ths) t.join();
    assert(caught[0] == 1 && caught[1] == 1);
    printf("adversarial collective mismatch detected on both ranks OK\n");
  }

  printf("test_ddp_fake: ALL OK\n");
  return 0;
}
