"""The driver contract: bench.py emits one valid JSON line, and the
torchrun multi-rank path (the exact launch the driver uses for SCALE_rNN,
gloo here instead of RCCL) works end-to-end."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")
TINY = ["--steps", "2", "--warmup", "1", "--rollout-length", "8",
        "--batch-size", "8", "--envs-per-actor", "4", "--num-actors", "2",
        "--device", "cpu", "--dtype", "fp32"]


def _parse_last_json(stdout: str) -> dict:
    for line in reversed(stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{stdout}")


def test_bench_single_rank_json_contract():
    r = subprocess.run([sys.executable, BENCH] + TINY, capture_output=True,
                       text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    j = _parse_last_json(r.stdout)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in j, key
    assert j["value"] > 0
    assert j["scaling"] == "weak"
    assert j["config"]["parallelism"] == "dp1"


@pytest.mark.timeout(900)
def test_bench_torchrun_two_ranks_cpu():
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29617", BENCH] + TINY
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=800,
                       cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-3000:])
    j = _parse_last_json(r.stdout)
    assert j["config"]["parallelism"] == "dp2"
    # whole-job value: 2 ranks × 2 steps × 8×8 env steps / elapsed
    assert j["value"] > 0


@pytest.mark.timeout(1200)
def test_bench_torchrun_eight_ranks_cpu_dryrun():
    """8-rank process topology dry-run on CPU (tiny shapes): proves the
    torchrun launch, per-rank actor farms, gloo all-reduce and the timed
    loop all come up at the driver's SCALE rank count."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "8", "--master-addr", "127.0.0.1",
           "--master-port", "29629", BENCH, "--steps", "2", "--warmup", "1",
           "--rollout-length", "8", "--batch-size", "8", "--envs-per-actor",
           "4", "--num-actors", "2", "--device", "cpu", "--dtype", "fp32"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=1100,
                       cwd=REPO, env=env)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-3000:])
    j = _parse_last_json(r.stdout)
    assert j["config"]["parallelism"] == "dp8"
    assert j["n_gpus"] == 0  # CPU dry-run
    assert j["value"] > 0
