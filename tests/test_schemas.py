import pytest

from clearml_serving_amd.schemas import (
    CanaryEP,
    EndpointMetricLogging,
    MetricType,
    ModelEndpoint,
    ModelMonitoring,
)


def test_endpoint_roundtrip():
    ep = ModelEndpoint(
        engine_type="sklearn", serving_url="test_model", version="1",
        input_size=[1, 28, 28], input_type="float32", input_name="x",
        output_size=[10], output_type="float32", output_name="y",
    )
    # flat input_size becomes nested (single-input shorthand)
    assert ep.input_size == [[1, 28, 28]]
    assert ep.input_type == ["float32"]
    assert ep.input_name == ["x"]
    d = ep.as_dict()
    ep2 = ModelEndpoint.from_dict(d)
    assert ep2.as_dict() == d


def test_endpoint_multi_input():
    ep = ModelEndpoint(
        engine_type="custom", serving_url="bert",
        input_size=[[-1, -1], [-1, -1], [-1, -1]],
        input_type=["int32", "int32", "int32"],
    )
    assert ep.input_size == [[-1, -1], [-1, -1], [-1, -1]]
    assert len(ep.input_type) == 3


def test_bad_engine_rejected():
    with pytest.raises(TypeError):
        ModelEndpoint(engine_type="not_an_engine", serving_url="x")


def test_bad_dtype_rejected():
    with pytest.raises(TypeError):
        ModelEndpoint(
            engine_type="sklearn", serving_url="x", input_type="floatzz"
        )


def test_monitoring_defaults():
    m = ModelMonitoring(base_serving_url="m", engine_type="sklearn")
    assert m.monitor_tags == []
    assert m.max_versions is None


def test_metric_logging_nested_dict():
    m = EndpointMetricLogging(
        endpoint="model/1",
        metrics={"x1": {"type": "scalar", "buckets": [0, 1, 2]},
                 "y": {"type": "enum", "buckets": ["cat", "dog"]}},
    )
    assert isinstance(m.metrics["x1"], MetricType)
    d = m.as_dict()
    m2 = EndpointMetricLogging.from_dict(d)
    assert m2.metrics["y"].buckets == ["cat", "dog"]


def test_metric_bad_type():
    with pytest.raises(TypeError):
        MetricType(type="bogus")


def test_canary():
    c = CanaryEP(endpoint="m", weights=[0.1, 0.9], load_endpoints=["m/1", "m/2"])
    assert c.load_endpoint_prefix is None


def test_all_schemas_roundtrip_through_dict():
    """Every config dataclass must survive as_dict -> ctor unchanged (this
    is exactly what serialize/deserialize does through the store)."""
    ep = ModelEndpoint(engine_type="llm", serving_url="m", version="2",
                       model_id="abc", preprocess_artifact="py_code_m",
                       input_size=[3, 224, 224], input_type="float32",
                       input_name="x", output_size=[10],
                       output_type="float32", output_name="y",
                       auxiliary_cfg={"gpu": 1, "overrides": {"layers": 2}})
    assert ModelEndpoint(**ep.as_dict()).as_dict() == ep.as_dict()

    mon = ModelMonitoring(engine_type="hip", base_serving_url="auto",
                          monitor_name="^m", monitor_project="p",
                          monitor_tags=["t"], only_published=True,
                          max_versions=3, input_size=[2], input_type="float32")
    assert ModelMonitoring(**mon.as_dict()).as_dict() == mon.as_dict()

    can = CanaryEP(endpoint="c", weights=[0.7, 0.3],
                   load_endpoints=["m/1", "m/2"])
    assert CanaryEP(**can.as_dict()).as_dict() == can.as_dict()

    met = EndpointMetricLogging(endpoint="m/1", log_frequency=0.5, metrics={
        "x0": MetricType(type="scalar", buckets=[0.0, 1.0, 2.0]),
        "lang": MetricType(type="enum", buckets=["en", "de"]),
        "conf": MetricType(type="value")})
    rt = EndpointMetricLogging(**met.as_dict())
    assert rt.as_dict() == met.as_dict()
    assert rt.metrics["x0"].buckets == [0.0, 1.0, 2.0]
