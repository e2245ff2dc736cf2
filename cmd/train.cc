// nezha-train: single-binary training CLI with multi-process DP launch
// (call stack §3.1). With --gpus=N it forks N rank
// processes, passes the ncclUniqueId over an inherited pipe, sets one HIP
// device per rank, and trains with bucketed RCCL all-reduce overlap.
#include <sys/wait.h>
#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <memory>

#include "src/comm/comm.h"
#include "src/models/models.h"
#include "src/train/trainer.h"
#include "src/util/util.h"

using namespace nz;

static int run_rank(const Flags& flags, int rank, int nranks, int uid_fd) {
  log_set_rank(rank);
  std::string device = flags.get_str("device", hip_device_count() > 0 ? "hip" : "cpu");
  std::unique_ptr<Context> ctx;
  if (device == "hip") {
    ctx = std::make_unique<Context>(Device::HIP, rank);
  } else {
    ctx = std::make_unique<Context>(Device::CPU, 0);
  }

  std::string model = flags.get_str("model", "mlp");
  uint64_t seed = (uint64_t)flags.get_int("seed", 1234);
  int64_t bucket = flags.get_int("bucket-mb", 64) << 20;
  float lr = (float)flags.get_float("lr", 1e-4);
  std::string optim = flags.get_str("optim", "adam");
  auto kind = optim == "sgd" ? Optimizer::Kind::SGD : Optimizer::Kind::ADAM;

  std::unique_ptr<Model> m;
  if (model == "mlp") {
    m = std::make_unique<MlpModel>(ctx.get(), flags.get_int("layers", 4),
                                   flags.get_int("dim", 4096),
                                   flags.get_int("batch", 8192), seed);
  } else if (model == "bert") {
    BertConfig cfg;
    cfg.layers = flags.get_int("layers", 12);
    cfg.hidden = flags.get_int("dim", 768);
    cfg.ffn = flags.get_int("ffn", 3072);
    cfg.heads = flags.get_int("heads", 12);
    cfg.seq = flags.get_int("seq", 128);
    cfg.batch = flags.get_int("batch", 256);
    m = std::make_unique<BertEncoderModel>(ctx.get(), cfg, seed);
  } else {
    NZ_LOG_ERROR("unknown model %s", model.c_str());
    return 1;
  }
  int64_t samples = m->samples_per_batch();
  double flops = m->flops_per_step();
  Trainer trainer(ctx.get(), std::move(m), kind, lr, bucket, seed + 12345);
  trainer.optimizer()->weight_decay = (float)flags.get_float("wd", 0.0);
  trainer.optimizer()->sgd_momentum = (float)flags.get_float("momentum", 0.0);

  std::unique_ptr<Comm> comm;
  if (nranks > 1) {
    std::vector<char> uid(comm_uid_bytes());
    // every rank (incl. 0, which generated it) reads the uid from the pipe
    size_t got = 0;
    while (got < uid.size()) {
      ssize_t k = read(uid_fd, uid.data() + got, uid.size() - got);
      NZ_CHECK(k > 0, "uid pipe read failed");
      got += (size_t)k;
    }
    comm = make_rccl_comm(ctx.get(), nranks, rank, uid.data());
    trainer.attach_ddp(comm.get());
    trainer.reseed_data(seed + 12345 + 1000003ull * rank);
  }

  int warmup = (int)flags.get_int("warmup", 10);
  int steps = (int)flags.get_int("steps", 50);
  std::string ckpt_out = flags.get_str("save", "");
  std::string ckpt_in = flags.get_str("load", "");
  if (!ckpt_in.empty()) trainer.load_checkpoint(ckpt_in);

  MetricsWriter metrics(rank == 0 ? flags.get_str("metrics", "") : "");
  if (rank == 0) flags.warn_unused();  // typo'd flags must not die silently
  for (int i = 0; i < warmup; ++i) trainer.step();
  ctx->sync();
  if (comm) comm->barrier();
  double t0 = now_sec();
  for (int i = 0; i < steps; ++i) trainer.step();
  ctx->sync();
  double elapsed = now_sec() - t0;
  if (comm) elapsed = comm->all_reduce_scalar((float)elapsed, ReduceOp::MAX);
  float loss = trainer.last_loss();
  double sps = double(samples) * nranks * steps / elapsed;
  if (rank == 0) {
    printf("{\"model\":\"%s\",\"n_gpus\":%d,\"steps\":%d,\"samples_per_sec\":%.1f,"
           "\"ms_per_step\":%.3f,\"loss\":%.6f,\"tflops_per_gpu\":%.1f}\n",
           model.c_str(), nranks, steps, sps, elapsed / steps * 1e3, loss,
           flops * steps / elapsed / 1e12);
    metrics.write({{"samples_per_sec", sps},
                   {"ms_per_step", elapsed / steps * 1e3},
                   {"loss", loss}},
                  {{"model", model}, {"config", flags.dump_json()}});
  }
  if (!ckpt_out.empty()) trainer.save_checkpoint(ckpt_out);
  return 0;
}

static const char* kHelp =
    "nezha-train — synthetic-data training CLI (MI355X-native engine)\n"
    "  --model=mlp|bert      model family (default mlp)\n"
    "  --device=hip|cpu      backend (default: hip when a GPU is present)\n"
    "  --gpus=N              fork N DP ranks over RCCL/xGMI (default 1)\n"
    "  --layers --dim        model size (mlp: 4x4096; bert: 12x768)\n"
    "  --ffn --heads --seq   bert geometry (3072 / 12 / 128)\n"
    "  --batch=N             per-rank batch (mlp 8192, bert 256)\n"
    "  --optim=adam|sgd --lr=F --wd=F --momentum=F --bucket-mb=N (64)\n"
    "  --steps=N --warmup=N  timed window (50 / 10)\n"
    "  --save=path --load=path  .npz checkpoint\n"
    "  --metrics=path        per-run JSONL summary\n"
    "  --seed=N              init seed (1234)\n"
    "  env: NEZHA_* (see docs/api.md)\n";

int main(int argc, const char** argv) {
  Flags flags;
  flags.parse(argc - 1, argv + 1);
  if (flags.get_bool("help", false)) {
    fputs(kHelp, stdout);
    return 0;
  }
  int nranks = (int)flags.get_int("gpus", 1);
  if (nranks <= 1) return run_rank(flags, 0, 1, -1);

  // Launcher: fork N ranks, share the ncclUniqueId via
  // pipes. The PARENT must never touch HIP/RCCL (forking after runtime
  // init is undefined), so rank 0 generates the uid in its own process
  // and sends it up; the parent only relays bytes.
  int uid_bytes = comm_uid_bytes();  // sizeof only, no runtime init
  int up[2];
  NZ_CHECK(pipe(up) == 0, "pipe failed");
  std::vector<pid_t> pids;
  std::vector<int> down_fds;
  for (int r = 0; r < nranks; ++r) {
    int fds[2];
    NZ_CHECK(pipe(fds) == 0, "pipe failed");
    pid_t pid = fork();
    if (pid == 0) {
      close(fds[1]);
      close(up[0]);
      if (r == 0) {
        std::vector<char> uid(uid_bytes);
        comm_get_uid(uid.data());  // rank 0's own process inits RCCL/HIP
        NZ_CHECK(write(up[1], uid.data(), uid.size()) == (ssize_t)uid.size(),
                 "uid up-pipe write failed");
      }
      close(up[1]);
      int rc = run_rank(flags, r, nranks, fds[0]);
      _exit(rc);
    }
    close(fds[0]);
    down_fds.push_back(fds[1]);
    pids.push_back(pid);
  }
  close(up[1]);
  std::vector<char> uid(uid_bytes);
  {
    size_t got = 0;
    while (got < uid.size()) {
      ssize_t k = read(up[0], uid.data() + got, uid.size() - got);
      NZ_CHECK(k > 0, "uid up-pipe read failed");
      got += (size_t)k;
    }
  }
  close(up[0]);
  for (int fd : down_fds)
    NZ_CHECK(write(fd, uid.data(), uid.size()) == (ssize_t)uid.size(),
             "uid down-pipe write failed");
  int fail = 0;
  for (size_t i = 0; i < pids.size(); ++i) {
    int status = 0;
    waitpid(pids[i], &status, 0);
    if (!WIFEXITED(status) || WEXITSTATUS(status) != 0) fail = 1;
    close(down_fds[i]);
  }
  return fail;
}
