#!/usr/bin/env python3
"""Flagship benchmark (driver contract): 4-layer 4096-dim MLP bf16 training,
samples/sec whole-job (BASELINE.json metric), DP over RCCL/xGMI for N>1.

Single GPU:   python bench.py --steps 30 --warmup 10
Multi GPU:    python bench.py --gpus N --steps K --warmup W
              (SELF-LAUNCHING: forks one rank process per GPU, exchanges the
              ncclUniqueId over a key-named file — no torchrun required)
        or:   python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
(one process per GPU; RANK/LOCAL_RANK/WORLD_SIZE read from the env when a
launcher sets them; torch is never imported here — the compute+comm path is
native C++/HIP/RCCL.)
"""
import argparse
import json
import os
import subprocess
import sys
import time
import uuid
import zlib

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import nezha_amd as nz  # noqa: E402

# BASELINE.md §2 rows (A100 synthetic baseline, samples/sec at 1/2/4/8 GPUs)
BASELINE_MLP = {1: 4.2e5, 2: 8.3e5, 4: 1.6e6, 8: 3.0e6}
BASELINE_BERT = {1: 2.1e3, 8: 1.5e4}

SEED = 1234
RANK_SEED_STRIDE = 1000003  # must match nz::kDdpSeedStride (trainer.h)


def self_launch(args):
    """Fork one rank process per GPU and wait.

    The driver may invoke `bench.py --gpus N` directly; torchrun must not be
    required on the box. Children re-exec this script with RANK/LOCAL_RANK/
    WORLD_SIZE set and share a fresh NEZHA_LAUNCH_KEY so the uid-file
    rendezvous can never pick up a stale file from a crashed earlier run.
    """
    env_base = dict(os.environ)
    env_base["WORLD_SIZE"] = str(args.gpus)
    env_base["MASTER_ADDR"] = "127.0.0.1"
    env_base.setdefault("MASTER_PORT", "29500")
    env_base["NEZHA_LAUNCH_KEY"] = uuid.uuid4().hex[:12]
    procs = []
    for r in range(args.gpus):
        env = dict(env_base, RANK=str(r), LOCAL_RANK=str(r))
        procs.append(subprocess.Popen([sys.executable, os.path.abspath(__file__)]
                                      + sys.argv[1:], env=env))
    rc = 0
    deadline = time.time() + 3600
    try:
        for p in procs:
            rc = max(rc, p.wait(timeout=max(1, deadline - time.time())))
    except subprocess.TimeoutExpired:
        sys.stderr.write("[bench] rank timeout — killing remaining ranks\n")
        rc = 124
    finally:
        for p in procs:
            if p.poll() is None:
                p.kill()
    sys.exit(rc)


def launch_key(world):
    """Rendezvous key shared by all ranks of ONE launch, unique across
    launches: the self-launch parent passes a fresh uuid; under torchrun the
    (MASTER_PORT, common parent pid) pair plays that role."""
    k = os.environ.get("NEZHA_LAUNCH_KEY")
    if k:
        return f"{k}_{world}"
    return f"{os.environ.get('MASTER_PORT', '0')}_{os.getppid()}_{world}"


def exchange_uid(rank, world, key):
    """File-based ncclUniqueId exchange on one node."""
    path = f"/tmp/nezha_uid_{key}"
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
    raise RuntimeError("timed out waiting for ncclUniqueId file " + path)


def make_comm(args, ctx, rank, world, key):
    if args.device == "cpu":
        # CPU rehearsal path for the launch machinery (tests): TcpComm star
        # over localhost; the GPU path below is RCCL over xGMI.
        port = 29710 + (zlib.crc32(key.encode()) % 1000)  # same on all ranks
        return nz.Comm.tcp(world, rank, "127.0.0.1", port)
    uid = exchange_uid(rank, world, key)
    return nz.Comm.rccl(ctx, world, rank, uid)


def preflight(comm, rank, world):
    """First-contact check BEFORE the timed window: a world-size all-reduce
    of the rank ids. A wrong sum means ranks joined different communicators
    (stale uid / split rendezvous) — fail loudly now, not 0.9-scaling-lately.
    """
    s = comm.allreduce_scalar(float(rank), "sum")
    want = world * (world - 1) / 2.0
    if abs(s - want) > 1e-3:
        sys.stderr.write(f"[bench rank {rank}] PREFLIGHT FAILED: "
                         f"sum(ranks)={s} want {want}\n")
        sys.exit(13)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--model", choices=["mlp", "bert"], default="mlp")
    ap.add_argument("--batch", type=int, default=0, help="per-GPU batch")
    ap.add_argument("--seq", type=int, default=128, help="bert seq len")
    ap.add_argument("--bucket-mb", type=int, default=64)
    ap.add_argument("--device", choices=["hip", "cpu"], default="hip",
                    help="cpu = launch-machinery rehearsal only (f32, tiny)")
    args = ap.parse_args()

    if args.gpus > 1 and "RANK" not in os.environ:
        self_launch(args)  # does not return

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    key = launch_key(world)

    if args.device == "hip":
        navail = nz.hip_device_count()
        assert navail > 0, "bench.py requires an MI355X GPU (or --device cpu)"
        if world > navail and not os.environ.get("NEZHA_OVERSUBSCRIBE"):
            sys.stderr.write(f"[bench rank {rank}] {world} ranks but only "
                             f"{navail} GPUs visible — RCCL cannot share a "
                             "device across ranks\n")
            sys.exit(12)
        ctx = nz.Ctx(nz.HIP, local_rank % navail)
    else:
        ctx = nz.Ctx(nz.CPU, 0)

    if args.model == "mlp":
        # 32768/GPU measured +3% over 16384 (larger GEMM M amortizes
        # fixed costs) and halves the comm:compute ratio for DP weak scaling
        batch = args.batch or (32768 if args.device == "hip" else 64)
        dim = 4096 if args.device == "hip" else 128
        tr = nz.Trainer.mlp(ctx, layers=4, dim=dim, batch=batch, optim="adam",
                            lr=1e-4, bucket_bytes=args.bucket_mb << 20,
                            seed=SEED)
        config = {"model": f"mlp-4x{dim}", "global_batch": batch * world,
                  "seq_len": None, "parallelism": f"dp{world}"}
        baseline = BASELINE_MLP.get(world) if args.device == "hip" else None
    else:
        batch = args.batch or (512 if args.device == "hip" else 4)
        tr = nz.Trainer.bert(ctx, layers=12, hidden=768, ffn=3072, heads=12,
                             seq=args.seq, batch=batch, optim="adam", lr=1e-4,
                             bucket_bytes=args.bucket_mb << 20, seed=SEED)
        config = {"model": "bert-base-encoder", "global_batch": batch * world,
                  "seq_len": args.seq, "parallelism": f"dp{world}"}
        # the BASELINE.md BERT row is quoted at S=128 only
        baseline = (BASELINE_BERT.get(world)
                    if args.device == "hip" and args.seq == 128 else None)

    comm = None
    if world > 1:
        comm = make_comm(args, ctx, rank, world, key)
        preflight(comm, rank, world)
        tr.attach_ddp(comm)
        tr.reseed_data(SEED + RANK_SEED_STRIDE * rank)  # per-rank data shard

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
    local_elapsed = t1 - t0
    elapsed = local_elapsed
    if comm:
        # per-rank diagnostics BEFORE the collective: a <0.9 scaling verdict
        # must be attributable to a rank (slow GPU / comm wait / stragglers)
        sys.stderr.write(f"[bench rank {rank}] local {local_elapsed*1e3/args.steps:.3f} "
                         f"ms/step over {args.steps} steps\n")
        elapsed = comm.allreduce_scalar(local_elapsed, "max")  # MAX over ranks
        ctx.sync()

    loss = tr.loss()
    samples = tr.samples_per_step * world * args.steps
    value = samples / elapsed
    ms_per_step = elapsed / args.steps * 1e3

    if comm and rank == 0 and args.device == "hip":
        try:
            os.remove(f"/tmp/nezha_uid_{key}")
        except OSError:
            pass
    if rank == 0:
        peak = 2.5e15 if args.device == "hip" else None
        out = {
            "metric": ("samples/sec (whole node), 4-layer 4096-dim MLP bf16"
                       if args.model == "mlp" and args.device == "hip" else
                       f"samples/sec (whole node), {config['model']}"
                       if args.model == "mlp" else
                       "samples/sec (whole node), BERT-base encoder bf16"),
            "value": value,
            "unit": "samples/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (value / baseline) if baseline else None,
            "dtype": "bf16" if args.device == "hip" else "f32",
            "data": "synthetic",
            "config": config,
            "loss": loss,
            "flops_per_step_per_gpu": tr.flops_per_step,
        }
        if peak:
            out["mfu_est"] = tr.flops_per_step / (elapsed / args.steps) / peak
        print(json.dumps(out))
    if rank == 0 and os.environ.get("NEZHA_TRACE"):
        ctx.prof_flush()  # write the chrome://tracing JSON (op scopes)


if __name__ == "__main__":
    main()
