"""Driver-contract protection for bench.py: the JSON line parses, carries
the required fields, and the data-parallel (torchrun, gloo world=2) path
works -- the driver runs N=1,2,4,8 on a real node, so the multi-process
rendezvous + barrier/all-reduce path must be correct by construction."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ("metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config")


def _env():
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    env["CMLS_BENCH_REQ"] = "16"
    env["CMLS_BENCH_DISTINCT"] = "4"
    return env


@pytest.mark.timeout(600)
def test_bench_single_process_json_contract():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=580, env=_env(), cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = r.stdout.strip().splitlines()[-1]
    out = json.loads(line)
    for k in REQUIRED:
        assert k in out, k
    assert out["n_gpus"] == 1 and out["steps"] == 2
    assert out["scaling"] == "weak" and out["higher_is_better"] is True
    assert out["config"]["requests_per_step"] == 16
    assert out["config"]["outputs_consumed"] > 0


@pytest.mark.timeout(900)
def test_bench_dp2_gloo():
    """The exact launch shape the driver uses (torch.distributed.run,
    --master-addr 127.0.0.1), world=2 on CPU/gloo."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29561", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=880, env=_env(), cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.strip().splitlines()
            if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    # whole-job aggregate: world * steps * req_per_step
    assert out["value"] > 0


@pytest.mark.timeout(900)
def test_bench_dp4_gloo():
    """World=4 on CPU/gloo: half the driver's 8-GPU ladder; catches any
    rendezvous/barrier/all-reduce issue that only appears beyond 2 ranks
    (each rank builds its own serving replica, so this is the heaviest
    CPU test -- tiny step counts keep it bounded)."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29562", os.path.join(REPO, "bench.py"),
         "--gpus", "4", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=880, env=_env(), cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.strip().splitlines()
            if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 4
    assert out["config"]["parallelism"] == "dp4"
    assert out["value"] > 0
