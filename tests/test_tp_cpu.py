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
    assert "TP-MICROBATCH-OK" in out.stdout
    assert "TP-PREFIXCACHE-OK" in out.stdout


@pytest.mark.timeout(300)
def test_tp4_math_and_engine_protocol():
    """Same checks at world=4 (widened llama-tiny: TP degree must divide
    head counts): catches world>2-only bugs -- shard arithmetic, plan-codec
    world independence, sampling vocab offsets -- before the first real
    multi-GPU window."""
    helper = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "helpers", "tp_check.py")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29532", helper],
        capture_output=True, text=True, timeout=280, env=env,
    )
    assert out.returncode == 0, out.stdout + "\n" + out.stderr
    assert "TP-MATH-OK" in out.stdout
    assert "TP-SAMPLE-OK" in out.stdout
    assert "TP-ENGINE-OK" in out.stdout
    assert "TP-MICROBATCH-OK" in out.stdout
    assert "TP-PREFIXCACHE-OK" in out.stdout


@pytest.mark.timeout(300)
def test_tp_serve_launcher_smoke(tmp_path):
    """The tp_serve LAUNCHER end to end under torchrun (gloo, world=2):
    store + endpoint prepared, rank 0 serves one chat completion through
    the processor, worker rank follows the plan broadcast."""
    import json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="tps", force_create=True)
    card = tmp_path / "card.json"
    card.write_text(json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu"}))
    rec = store.register_model(name="t", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tpllm",
                                    model_id=rec.model_id))
    proc.serialize()

    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = store_root
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(
        os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29533", "-m",
         "clearml_serving_amd.parallel.tp_serve", "--smoke"],
        capture_output=True, text=True, timeout=280, env=env,
    )
    assert out.returncode == 0, out.stdout + "\n" + out.stderr
    assert "TP-SERVE-OK" in out.stdout
