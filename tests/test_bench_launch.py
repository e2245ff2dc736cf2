"""bench.py launch-machinery tests.

The driver may invoke `python bench.py --gpus N` directly — bench.py must
self-launch its DP ranks (no torchrun). These tests exercise that machinery
end-to-end on CPU (TcpComm rehearsal path): fork, rendezvous, preflight,
timed window, MAX-over-ranks aggregation, single JSON line from rank 0.
"""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def run_bench(args, timeout=180):
    r = subprocess.run([sys.executable, BENCH] + args, cwd=REPO,
                       capture_output=True, timeout=timeout)
    return r


def parse_json_line(out):
    lines = [ln for ln in out.decode().splitlines() if ln.strip().startswith("{")]
    assert len(lines) == 1, f"expected exactly one JSON line, got: {out!r}"
    return json.loads(lines[0])


def test_bench_single_rank_cpu():
    r = run_bench(["--device", "cpu", "--steps", "2", "--warmup", "1"])
    assert r.returncode == 0, r.stderr.decode()
    j = parse_json_line(r.stdout)
    assert j["n_gpus"] == 1
    assert j["steps"] == 2
    assert j["data"] == "synthetic"
    assert j["config"]["parallelism"] == "dp1"
    assert j["value"] > 0


def test_bench_self_launch_two_ranks_cpu():
    """--gpus 2 with no RANK in env: bench.py forks its own rank processes."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run([sys.executable, BENCH, "--gpus", "2", "--device",
                        "cpu", "--steps", "2", "--warmup", "1"], cwd=REPO,
                       env=env, capture_output=True, timeout=180)
    assert r.returncode == 0, r.stderr.decode() + r.stdout.decode()
    j = parse_json_line(r.stdout)
    assert j["n_gpus"] == 2
    assert j["config"]["parallelism"] == "dp2"
    assert j["config"]["global_batch"] == 2 * (j["config"]["global_batch"] // 2)
    # per-rank diagnostics on stderr (attribution for scaling verdicts)
    err = r.stderr.decode()
    assert "[bench rank 0]" in err and "[bench rank 1]" in err


def test_bench_env_rank_path_cpu():
    """Launcher-set RANK env (the torchrun contract): each rank process runs
    one shard; rank 0 prints the JSON line, rank 1 prints nothing."""
    import threading

    outs = {}

    def run(rank):
        env = dict(os.environ, RANK=str(rank), LOCAL_RANK=str(rank),
                   WORLD_SIZE="2", MASTER_ADDR="127.0.0.1",
                   MASTER_PORT="29877", NEZHA_LAUNCH_KEY="envtest42")
        outs[rank] = subprocess.run(
            [sys.executable, BENCH, "--gpus", "2", "--device", "cpu",
             "--steps", "2", "--warmup", "1"], cwd=REPO, env=env,
            capture_output=True, timeout=180)

    ts = [threading.Thread(target=run, args=(r,)) for r in range(2)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert outs[0].returncode == 0, outs[0].stderr.decode()
    assert outs[1].returncode == 0, outs[1].stderr.decode()
    j = parse_json_line(outs[0].stdout)
    assert j["n_gpus"] == 2
    assert outs[1].stdout.strip() == b""
