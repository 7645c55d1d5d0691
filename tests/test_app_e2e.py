import gzip
import json

import numpy as np
import pytest
from fastapi.testclient import TestClient

from clearml_serving_amd.schemas import ModelEndpoint
from clearml_serving_amd.serving.app import create_app


@pytest.fixture()
def client(processor, store, tmp_path):
    import joblib
    from sklearn.linear_model import LinearRegression

    X = np.array([[0.0, 0.0], [1.0, 0.0], [0.0, 1.0]])
    y = np.array([0.0, 2.0, 3.0])
    model = LinearRegression().fit(X, y)
    p = tmp_path / "model.pkl"
    joblib.dump(model, str(p))
    rec = store.register_model(name="lin2", project="p", path=str(p))

    code = tmp_path / "preprocess.py"
    code.write_text(
        "import numpy as np\n"
        "class Preprocess(object):\n"
        "    def preprocess(self, body, state, collect_custom_statistics_fn=None):\n"
        "        return [[body['x0'], body['x1']]]\n"
        "    def postprocess(self, data, state, collect_custom_statistics_fn=None):\n"
        "        return {'y': float(np.asarray(data).ravel()[0])}\n"
    )
    processor.add_endpoint(
        ModelEndpoint(engine_type="sklearn", serving_url="test_model_sklearn",
                      model_id=rec.model_id),
        preprocess_code=str(code),
    )
    processor.serialize()
    app = create_app(processor=processor, poll_frequency_sec=3600)
    with TestClient(app) as c:
        yield c


def test_serve_roundtrip(client):
    r = client.post("/serve/test_model_sklearn", json={"x0": 1.0, "x1": 0.0})
    assert r.status_code == 200, r.text
    assert abs(r.json()["y"] - 2.0) < 1e-6


def test_serve_gzip_body(client):
    body = gzip.compress(json.dumps({"x0": 0.0, "x1": 1.0}).encode())
    r = client.post(
        "/serve/test_model_sklearn", content=body,
        headers={"Content-Encoding": "gzip", "Content-Type": "application/json"},
    )
    assert r.status_code == 200, r.text
    assert abs(r.json()["y"] - 3.0) < 1e-6


def test_serve_unknown_endpoint_404(client):
    r = client.post("/serve/ghost_model", json={})
    assert r.status_code == 404


def test_health(client):
    r = client.get("/health")
    assert r.status_code == 200
    assert "test_model_sklearn" in r.json()["endpoints"]


def test_stats_flow_to_prometheus(client, processor):
    from clearml_serving_amd.statistics.collector import StatsRegistry
    from prometheus_client import CollectorRegistry

    reg = CollectorRegistry()
    stats_registry = StatsRegistry(processor=processor, registry=reg)
    processor.set_stats_sink(stats_registry.report_batch)
    processor._metric_log_freq = 1.0

    for _ in range(3):
        client.post("/serve/test_model_sklearn", json={"x0": 1.0, "x1": 0.0})
    # drain the queue synchronously (the background thread also does this)
    batch = []
    item = processor._stats_queue.get(timeout=0)
    while item:
        batch.append(item)
        item = processor._stats_queue.get(timeout=0)
    stats_registry.report_batch(batch)
    # reserved metrics present under reference naming: {url}:{var} -> _
    val = reg.get_sample_value("test_model_sklearn:_count_total")
    assert val == 3.0
    lat_count = reg.get_sample_value(
        "test_model_sklearn:_latency_count")
    assert lat_count == 3.0


def test_status_endpoint(client):
    client.post("/serve/test_model_sklearn", json={"x0": 1.0, "x1": 0.0})
    r = client.get("/status")
    assert r.status_code == 200
    body = r.json()
    assert "test_model_sklearn" in body["endpoints"]
    assert body["revision"] is not None


def test_engine_exception_maps_to_422(processor, store, tmp_path):
    code = tmp_path / "bad.py"
    code.write_text(
        "class Preprocess(object):\n"
        "    def process(self, data, state, collect_custom_statistics_fn=None):\n"
        "        raise ValueError('bad input shape')\n")
    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.app import create_app

    processor.add_endpoint(
        ModelEndpoint(engine_type="custom", serving_url="bad_ep"),
        preprocess_code=str(code))
    processor.serialize()  # app startup re-deserializes from the store
    app = create_app(processor=processor, poll_frequency_sec=3600)
    with TestClient(app) as c:
        r = c.post("/serve/bad_ep", json={})
        assert r.status_code == 422
        assert "bad input shape" in r.json()["detail"]


def test_engine_crash_maps_to_500(processor, store, tmp_path):
    code = tmp_path / "crash.py"
    code.write_text(
        "class Preprocess(object):\n"
        "    def process(self, data, state, collect_custom_statistics_fn=None):\n"
        "        raise RuntimeError('boom')\n")
    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.app import create_app

    processor.add_endpoint(
        ModelEndpoint(engine_type="custom", serving_url="crash_ep"),
        preprocess_code=str(code))
    processor.serialize()
    app = create_app(processor=processor, poll_frequency_sec=3600)
    with TestClient(app) as c:
        r = c.post("/serve/crash_ep", json={})
        assert r.status_code == 500


def test_endpoint_telemetry_counters(client):
    """Reference endpoint telemetry parity (model_request_processor.py:
    162-187): per-endpoint request/response counters, surfaced on
    GET /status."""
    for _ in range(3):
        r = client.post("/serve/test_model_sklearn",
                        json={"x0": 1.0, "x1": 0.0})
        assert r.status_code == 200
    r = client.post("/serve/nope_model", json=[[1.0]])
    assert r.status_code == 404
    tel = client.get("/status").json()["endpoint_telemetry"]
    assert tel["test_model_sklearn"]["requests"] == 3
    assert tel["test_model_sklearn"]["responses"] == 3
    assert tel["test_model_sklearn"]["in_flight"] == 0
    # failed request: counted in, never counted out (reference increments
    # on_response only after success)
    assert tel["nope_model"]["requests"] == 1
    assert tel["nope_model"]["responses"] == 0
