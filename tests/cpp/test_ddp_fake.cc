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
    });
  }
  for (auto& t : threads) t.join();

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

  printf("test_ddp_fake: ALL OK\n");
  return 0;
}
