#!/usr/bin/env python3
"""Flagship benchmark (driver contract): 4-layer 4096-dim MLP bf16 training,
samples/sec whole-job (BASELINE.json metric), DP over RCCL/xGMI for N>1.

Single GPU:   python bench.py --steps 30 --warmup 10
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
(one process per GPU; RANK/LOCAL_RANK/WORLD_SIZE/MASTER_PORT from env; the
ncclUniqueId is exchanged through a MASTER_PORT-keyed file on the node —
torch is never imported here, the compute+comm path is native C++/HIP/RCCL.)
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import nezha_amd as nz  # noqa: E402

# BASELINE.md §2 rows (A100 synthetic baseline, samples/sec at 1/2/4/8 GPUs)
BASELINE_MLP = {1: 4.2e5, 2: 8.3e5, 4: 1.6e6, 8: 3.0e6}
BASELINE_BERT = {1: 2.1e3, 8: 1.5e4}


def exchange_uid(rank, world, key):
    """File-based ncclUniqueId exchange on one node.

    The file name is keyed by (MASTER_PORT, parent pid, world): torchrun is
    the common parent of all ranks, so the key is shared within one launch
    and unique across back-to-back launches — a stale file from a previous
    run (same port) can never be picked up."""
    path = f"/tmp/nezha_uid_{key}_{os.getppid()}_{world}"
    if rank == 0:
        uid = nz.Comm.get_uid()
        with open(path + ".tmp", "wb") as f:
            f.write(uid)
        os.rename(path + ".tmp", path)
        return uid
    deadline = time.time() + 180
    while time.time() < deadline:
        if os.path.exists(path):
            uid = open(path, "rb").read()
            if len(uid) == nz.Comm.uid_bytes():
                return uid
        time.sleep(0.05)
    raise RuntimeError("timed out waiting for ncclUniqueId file")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--model", choices=["mlp", "bert"], default="mlp")
    ap.add_argument("--batch", type=int, default=0, help="per-GPU batch")
    ap.add_argument("--seq", type=int, default=128, help="bert seq len")
    ap.add_argument("--bucket-mb", type=int, default=64)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    if world > 1 and "RANK" not in os.environ:
        raise SystemExit(
            "--gpus > 1 needs one process per GPU: launch via\n"
            "  python -m torch.distributed.run --nnodes=1 --nproc-per-node "
            f"{world} --master-addr 127.0.0.1 bench.py --gpus {world} ...")
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    port_key = os.environ.get("MASTER_PORT", "0")

    navail = nz.hip_device_count()
    assert navail > 0, "bench.py requires an MI355X GPU"
    ctx = nz.Ctx(nz.HIP, local_rank % navail)

    if args.model == "mlp":
        # 32768/GPU measured +3% over 16384 (larger GEMM M amortizes
        # fixed costs) and halves the comm:compute ratio for DP weak scaling
        batch = args.batch or 32768
        tr = nz.Trainer.mlp(ctx, layers=4, dim=4096, batch=batch, optim="adam",
                            lr=1e-4, bucket_bytes=args.bucket_mb << 20,
                            seed=1234)
        config = {"model": "mlp-4x4096", "global_batch": batch * world,
                  "seq_len": None, "parallelism": f"dp{world}"}
        baseline = BASELINE_MLP.get(world)
    else:
        batch = args.batch or 512
        tr = nz.Trainer.bert(ctx, layers=12, hidden=768, ffn=3072, heads=12,
                             seq=args.seq, batch=batch, optim="adam", lr=1e-4,
                             bucket_bytes=args.bucket_mb << 20, seed=1234)
        config = {"model": "bert-base-encoder", "global_batch": batch * world,
                  "seq_len": args.seq, "parallelism": f"dp{world}"}
        baseline = BASELINE_BERT.get(world)

    comm = None
    if world > 1:
        uid = exchange_uid(rank, world, port_key)
        comm = nz.Comm.rccl(ctx, world, rank, uid)
        tr.attach_ddp(comm)
        tr.reseed_data(1234 + 1000003 * rank)  # per-rank data shard

    # ---- warmup ----
    for _ in range(args.warmup):
        tr.step()
    ctx.sync()
    if comm:
        comm.barrier()
        ctx.sync()

    # ---- timed window: exactly --steps steps, barrier+sync both sides ----
    t0 = time.perf_counter()
    for _ in range(args.steps):
        tr.step()
    ctx.sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if comm:
        elapsed = comm.allreduce_scalar(elapsed, "max")  # MAX over ranks
        ctx.sync()

    loss = tr.loss()
    samples = tr.samples_per_step * world * args.steps
    value = samples / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if comm and rank == 0:
        try:
            os.remove(f"/tmp/nezha_uid_{port_key}_{os.getppid()}_{world}")
        except OSError:
            pass
    if rank == 0:
        out = {
            "metric": "samples/sec (whole node), 4-layer 4096-dim MLP bf16"
                      if args.model == "mlp" else
                      "samples/sec (whole node), BERT-base encoder bf16",
            "value": value,
            "unit": "samples/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (value / baseline) if baseline else None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": config,
            "loss": loss,
            "flops_per_step_per_gpu": tr.flops_per_step,
            "mfu_est": tr.flops_per_step / (elapsed / args.steps) / 2.5e15,
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
