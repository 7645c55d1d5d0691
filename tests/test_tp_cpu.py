"""Tensor-parallel path over gloo, world_size=2, CPU (no GPU needed)."""

import os
import subprocess
import sys

import pytest


@pytest.mark.timeout(300)
def test_tp2_math_and_engine_protocol():
    helper = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "helpers", "tp_check.py")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", helper],
        capture_output=True, text=True, timeout=280, env=env,
    )
    assert out.returncode == 0, out.stdout + "\n" + out.stderr
    assert "TP-MATH-OK" in out.stdout
    assert "TP-ENGINE-OK" in out.stdout
