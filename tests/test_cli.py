import subprocess
import sys

import numpy as np
import pytest


@pytest.fixture()
def cli_env(tmp_path, monkeypatch):
    import os

    env = dict(os.environ)
    env["CLEARML_SERVING_AMD_STORE"] = str(tmp_path / "store")
    env["PYTHONPATH"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    return env


def run_cli(env, *args):
    out = subprocess.run(
        [sys.executable, "-m", "clearml_serving_amd.cli"] + list(args),
        capture_output=True, text=True, env=env, timeout=120,
    )
    return out


def test_create_list(cli_env):
    out = run_cli(cli_env, "create", "--name", "svc1", "--project", "proj")
    assert out.returncode == 0, out.stderr
    assert "New Serving Service created" in out.stdout
    out = run_cli(cli_env, "list")
    assert "svc1" in out.stdout


def test_model_upload_add_and_ls(cli_env, tmp_path):
    import joblib
    from sklearn.linear_model import LinearRegression

    model = LinearRegression().fit(np.array([[0.0], [1.0]]), np.array([0.0, 2.0]))
    p = tmp_path / "m.pkl"
    joblib.dump(model, str(p))

    assert run_cli(cli_env, "create", "--name", "svc").returncode == 0
    out = run_cli(cli_env, "model", "upload", "--name", "lin", "--project",
                  "p", "--path", str(p), "--publish")
    assert out.returncode == 0, out.stderr
    model_id = out.stdout.strip().split("id=")[-1]

    out = run_cli(cli_env, "model", "add", "--engine", "sklearn",
                  "--endpoint", "lin_ep", "--version", "1",
                  "--model-id", model_id)
    assert out.returncode == 0, out.stderr
    assert "lin_ep/1" in out.stdout

    out = run_cli(cli_env, "model", "list")
    assert "lin_ep/1" in out.stdout

    out = run_cli(cli_env, "model", "remove", "--endpoint", "lin_ep/1")
    assert "removed" in out.stdout


def test_model_add_by_name_query(cli_env, tmp_path):
    import joblib
    from sklearn.linear_model import LinearRegression

    model = LinearRegression().fit(np.array([[0.0], [1.0]]), np.array([0.0, 2.0]))
    p = tmp_path / "m.pkl"
    joblib.dump(model, str(p))
    run_cli(cli_env, "create", "--name", "svc")
    run_cli(cli_env, "model", "upload", "--name", "qmod", "--project", "p",
            "--path", str(p))
    out = run_cli(cli_env, "model", "add", "--engine", "sklearn",
                  "--endpoint", "q_ep", "--name", "^qmod$", "--project", "p")
    assert out.returncode == 0, out.stderr


def test_auto_update_and_canary(cli_env, tmp_path):
    run_cli(cli_env, "create", "--name", "svc")
    out = run_cli(cli_env, "model", "auto-update", "--engine", "sklearn",
                  "--endpoint", "auto_ep", "--name", "^automod$",
                  "--max-versions", "2")
    assert out.returncode == 0, out.stderr
    out = run_cli(cli_env, "model", "canary", "--endpoint", "auto_ep_canary",
                  "--weights", "0.9", "0.1",
                  "--input-endpoint-prefix", "auto_ep")
    assert out.returncode == 0, out.stderr
    out = run_cli(cli_env, "model", "list")
    assert "auto_ep" in out.stdout and "auto_ep_canary" in out.stdout


def test_metrics_add_list_remove(cli_env):
    run_cli(cli_env, "create", "--name", "svc")
    out = run_cli(cli_env, "metrics", "add", "--endpoint", "m/*",
                  "--log-freq", "0.5",
                  "--variable-scalar", "x1=0,0.5,1", "x2=0.0/1.0/4",
                  "--variable-enum", "detect=cat,dog",
                  "--variable-value", "latency")
    assert out.returncode == 0, out.stderr
    out = run_cli(cli_env, "metrics", "list")
    assert "x1" in out.stdout and "detect" in out.stdout
    data = out.stdout[out.stdout.index("{"):]
    import json

    cfg = json.loads(data)
    assert cfg["m/*"]["metrics"]["x2"]["buckets"] == [0.0, 0.25, 0.5, 0.75, 1.0]
    out = run_cli(cli_env, "metrics", "remove", "--endpoint", "m/*",
                  "--variable", "x1")
    assert out.returncode == 0
    out = run_cli(cli_env, "metrics", "list")
    assert '"x1"' not in out.stdout


def test_config_cmd(cli_env):
    run_cli(cli_env, "create", "--name", "svc")
    out = run_cli(cli_env, "config", "--base-serving-url",
                  "http://127.0.0.1:8080/serve", "--metric-log-freq", "0.1")
    assert out.returncode == 0, out.stderr
