"""The examples are executable fixtures (the reference's substitute for
tests): run the sklearn example exactly as its readme does --
train script, CLI session + endpoint, HTTP serve -- in an isolated store."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(180)
def test_sklearn_example_readme_flow(tmp_path, monkeypatch):
    pytest.importorskip("sklearn")
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    # 1. train + register (writes sklearn-model.pkl into cwd)
    run_py([os.path.join(ROOT, "examples/sklearn/train_model.py")],
           cwd=str(tmp_path))
    # 2. CLI: create session + add endpoint (readme commands)
    run_py(["-m", "clearml_serving_amd", "create", "--name",
            "serving example"], cwd=ROOT)
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "sklearn", "--endpoint", "test_model_sklearn", "--preprocess",
            "examples/sklearn/preprocess.py", "--name",
            "train sklearn model", "--project", "serving examples"],
           cwd=ROOT)
    # 3. HTTP serve (readme curl)
    monkeypatch.setenv("CLEARML_SERVING_AMD_STORE", str(tmp_path / "store"))
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/test_model_sklearn",
                        json={"x0": 1, "x1": 2})
        assert r.status_code == 200, r.text
        assert r.json() == {"y": [1]}


@pytest.mark.timeout(120)
def test_custom_example_readme_flow(tmp_path):
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=90)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    run_py(["-m", "clearml_serving_amd", "create", "--name", "custom ex"])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "custom", "--endpoint", "test_model_custom", "--preprocess",
            "examples/custom/preprocess.py"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/test_model_custom", json={"x0": 1, "x1": 2})
        assert r.status_code == 200, r.text
        assert r.json() == {"y": 3.0}


@pytest.mark.timeout(180)
def test_pipeline_example_readme_flow(tmp_path):
    """Ensemble example: custom_async endpoint fanning out to the sklearn
    endpoint via the injected in-process send_request hook."""
    pytest.importorskip("sklearn")
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    run_py([os.path.join(ROOT, "examples/sklearn/train_model.py")],
           cwd=str(tmp_path))
    run_py(["-m", "clearml_serving_amd", "create", "--name", "pipe ex"])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "sklearn", "--endpoint", "test_model_sklearn", "--preprocess",
            "examples/sklearn/preprocess.py", "--name",
            "train sklearn model", "--project", "serving examples"])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "custom_async", "--endpoint", "ensemble", "--preprocess",
            "examples/pipeline/async_preprocess.py"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/ensemble", json={"x0": 1, "x1": 2})
        assert r.status_code == 200, r.text
        assert r.json() == {"y": [1, 1]}


@pytest.mark.timeout(240)
def test_canary_autoupdate_example_readme_flow(tmp_path):
    """Canary + auto-update example: registering a SECOND model version
    materializes /2 on sync and the prefix canary routes to both."""
    pytest.importorskip("sklearn")
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    run_py([os.path.join(ROOT, "examples/sklearn/train_model.py")],
           cwd=str(tmp_path))
    run_py(["-m", "clearml_serving_amd", "create", "--name", "canary ex"])
    run_py(["-m", "clearml_serving_amd", "model", "auto-update",
            "--engine", "sklearn", "--endpoint", "test_model_sklearn_auto",
            "--preprocess", "examples/sklearn/preprocess.py", "--name",
            "train sklearn model", "--project", "serving examples",
            "--max-versions", "2"])
    run_py(["-m", "clearml_serving_amd", "model", "canary", "--endpoint",
            "test_model_sklearn_canary", "--weights", "0.9", "0.1",
            "--input-endpoint-prefix", "test_model_sklearn_auto"])
    # second registration = new model version
    run_py([os.path.join(ROOT, "examples/sklearn/train_model.py")],
           cwd=str(tmp_path))

    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        # force one sync (the daemon would do this within a poll interval)
        proc = app.state.processor
        proc.deserialize()
        proc._update_monitored_models()
        proc._update_canary_lookup()
        eps = proc.get_synced_endpoints()
        assert "test_model_sklearn_auto/1" in eps
        assert "test_model_sklearn_auto/2" in eps
        for _ in range(10):
            r = client.post("/serve/test_model_sklearn_canary",
                            json={"x0": 1, "x1": 2})
            assert r.status_code == 200, r.text
            assert r.json() == {"y": [1]}


@pytest.mark.timeout(240)
def test_llm_example_readme_flow(tmp_path):
    """LLM example flow with a scaled-down model card: upload the card via
    the CLI, add the llm endpoint, chat through the OpenAI route."""
    import json as _json

    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_refs = {}

    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    card = tmp_path / "llama_card.json"
    card.write_text(_json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu"}))
    run_py(["-m", "clearml_serving_amd", "create", "--name", "llm ex"])
    run_py(["-m", "clearml_serving_amd", "model", "upload", "--name",
            "llama card", "--project", "examples", "--path", str(card)])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine", "llm",
            "--endpoint", "test_llm", "--name", "llama card", "--project",
            "examples"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    try:
        with TestClient(app) as client:
            r = client.post("/serve/openai/v1/chat/completions", json={
                "model": "test_llm", "max_tokens": 4, "temperature": 0.0,
                "ignore_eos": True,
                "messages": [{"role": "user", "content": "hi"}]})
            assert r.status_code == 200, r.text
            out = r.json()
            assert out["choices"][0]["finish_reason"] == "length"
            assert out["usage"]["completion_tokens"] == 4
    finally:
        LlmPreprocessRequest._engine_singleton = None
        LlmPreprocessRequest._engines = {}
        LlmPreprocessRequest._engine_refs = {}


@pytest.mark.timeout(240)
def test_bert_example_readme_flow(tmp_path):
    """BERT example flow (the reference's huggingface/Triton recipe) on
    CPU fp32: card upload, hip-engine endpoint with I/O spec + aux batch
    config via CLI, pre-tokenized payload through HTTP."""
    import json as _json

    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=150)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    card = tmp_path / "bert_card.json"
    card.write_text(_json.dumps({"arch": "bert-base", "num_labels": 2,
                                 "dtype": "float32", "vocab_size": 30522}))
    run_py(["-m", "clearml_serving_amd", "create", "--name", "bert ex"])
    run_py(["-m", "clearml_serving_amd", "model", "upload", "--name",
            "bert base card", "--project", "examples", "--path", str(card)])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine", "hip",
            "--endpoint", "transformer_model", "--name", "bert base card",
            "--project", "examples", "--preprocess",
            "examples/bert/preprocess.py",
            "--input-size", "[8]", "[8]", "--input-type", "int64", "int32",
            "--input-name", "input_ids", "attention_mask",
            "--aux-config", "max_batch_size=8", "max_queue_delay_us=1000",
            "use_graphs=false"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    payload = _json.load(open(os.path.join(ROOT,
                                           "examples/bert/example_payload.json")))
    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/transformer_model", json=payload)
        assert r.status_code == 200, r.text
        out = r.json()
        assert out["label"] in (0, 1) and len(out["logits"]) == 2


@pytest.mark.timeout(240)
def test_llm_speculative_example_flow(tmp_path):
    """The readme's speculative-decoding recipe end to end via CLI + HTTP:
    a card with {'speculative': {...}} serves greedy chat identically to a
    plain card, and the engine reports speculation activity."""
    import json as _json

    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_refs = {}

    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    base = {"arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
            "block_size": 16, "max_model_len": 128, "device": "cpu"}
    plain_card = tmp_path / "plain.json"
    plain_card.write_text(_json.dumps(base))
    spec_card = tmp_path / "spec.json"
    spec_card.write_text(_json.dumps({
        **base, "speculative": {"method": "ngram", "num_spec_tokens": 4,
                                "ngram": 2}}))
    run_py(["-m", "clearml_serving_amd", "create", "--name", "spec ex"])
    for name, card in (("plain card", plain_card), ("spec card", spec_card)):
        run_py(["-m", "clearml_serving_amd", "model", "upload", "--name",
                name, "--project", "examples", "--path", str(card)])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine", "llm",
            "--endpoint", "plain_llm", "--name", "plain card", "--project",
            "examples"])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine", "vllm",
            "--endpoint", "spec_llm", "--name", "spec card", "--project",
            "examples"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    try:
        with TestClient(app) as client:
            body = {"max_tokens": 24, "temperature": 0.0, "ignore_eos": True,
                    "messages": [{"role": "user", "content": "abcabcabc"}]}
            outs = {}
            for model in ("plain_llm", "spec_llm"):
                r = client.post("/serve/openai/v1/chat/completions",
                                json={**body, "model": model})
                assert r.status_code == 200, r.text
                outs[model] = r.json()["choices"][0]["message"]["content"]
            # exactness: speculation must not change greedy output
            assert outs["plain_llm"] == outs["spec_llm"]
            # speculation fired on the spec endpoint
            specs = [e for e in LlmPreprocessRequest._engines.values()
                     if e.cfg.speculative]
            assert specs and specs[0].stats["spec_proposed"] > 0
    finally:
        LlmPreprocessRequest._engine_singleton = None
        LlmPreprocessRequest._engines = {}
        LlmPreprocessRequest._engine_refs = {}


@pytest.mark.timeout(240)
def test_ensemble_example_readme_flow(tmp_path):
    """The ensemble readme flow verbatim: train VotingRegressor, upload,
    add sklearn endpoint with the example preprocess, query over HTTP."""
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    pkl = tmp_path / "ensemble-vr.pkl"
    run_py(["examples/ensemble/train_ensemble.py", "--out", str(pkl)])
    run_py(["-m", "clearml_serving_amd", "create", "--name", "ens ex"])
    run_py(["-m", "clearml_serving_amd", "model", "upload", "--name",
            "train model ensemble", "--project", "serving examples",
            "--path", str(pkl)])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "sklearn", "--endpoint", "test_model_ensemble",
            "--preprocess", os.path.join(ROOT, "examples/ensemble/preprocess.py"),
            "--name", "train model ensemble", "--project",
            "serving examples"])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/test_model_ensemble",
                        json={"x0": -5.0, "x1": -2.0})
        assert r.status_code == 200, r.text
        assert "y" in r.json()
        assert isinstance(r.json()["y"], (int, float, list))


@pytest.mark.timeout(240)
def test_preprocess_template_is_a_valid_preprocess(tmp_path):
    """The shipped template is directly servable: register it verbatim on
    a custom endpoint whose load()/process() are inherited defaults."""
    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = ROOT

    def run_py(args, cwd=ROOT):
        out = subprocess.run([sys.executable] + args, cwd=cwd, env=env,
                             capture_output=True, text=True, timeout=120)
        assert out.returncode == 0, out.stdout + "\n" + out.stderr
        return out.stdout

    run_py(["-m", "clearml_serving_amd", "create", "--name", "tpl ex"])
    run_py(["-m", "clearml_serving_amd", "model", "add", "--engine",
            "custom", "--endpoint", "tpl", "--preprocess",
            os.path.join(ROOT,
                         "examples/preprocess_template/"
                         "preprocess_template.py")])
    from fastapi.testclient import TestClient

    from clearml_serving_amd.serving.app import create_app

    app = create_app(store_root=str(tmp_path / "store"),
                     poll_frequency_sec=3600)
    with TestClient(app) as client:
        r = client.post("/serve/tpl", json={"anything": 1})
        # template's process() returns None -> served as null, 200
        assert r.status_code == 200, r.text
        assert r.json() is None
