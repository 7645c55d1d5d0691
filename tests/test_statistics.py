"""Statistics registry: metric types, naming compatibility, prefix configs."""

from prometheus_client import CollectorRegistry

from clearml_serving_amd.schemas import EndpointMetricLogging, MetricType
from clearml_serving_amd.statistics.collector import StatsRegistry, _prom_name


def test_prom_name_folding():
    # reference naming: {url}:{var} with non-alnum -> "_"
    assert _prom_name("model/1", "_latency") == "model_1:_latency"
    assert _prom_name("a-b", "x.y") == "a_b:x_y"


class FakeProcessor:
    def __init__(self, cfg):
        self._cfg = cfg

    def list_endpoint_logging(self):
        return self._cfg


def test_all_metric_types():
    cfg = {
        "ep/1": EndpointMetricLogging(endpoint="ep/1", metrics={
            "x1": MetricType(type="scalar", buckets=[0, 1, 2]),
            "detect": MetricType(type="enum", buckets=["cat", "dog"]),
            "gauge_v": MetricType(type="value"),
            "cnt": MetricType(type="counter"),
        }),
    }
    reg = CollectorRegistry()
    sr = StatsRegistry(processor=FakeProcessor(cfg), registry=reg)
    sr.report_batch([
        {"_url": "ep/1", "_latency": 0.02, "_count": 2, "x1": 1.5,
         "detect": "cat", "gauge_v": 42.0, "cnt": 3},
        {"_url": "ep/1", "_latency": 0.2, "_count": 2, "detect": "dog"},
    ])
    assert reg.get_sample_value("ep_1:_count_total") == 4.0
    assert reg.get_sample_value("ep_1:_latency_count") == 2.0
    # scalar histogram with custom buckets
    assert reg.get_sample_value("ep_1:x1_bucket", {"le": "2.0"}) == 1.0
    # enum counters per value
    assert reg.get_sample_value("ep_1:detect_total", {"value": "cat"}) == 1.0
    assert reg.get_sample_value("ep_1:detect_total", {"value": "dog"}) == 1.0
    assert reg.get_sample_value("ep_1:gauge_v") == 42.0
    assert reg.get_sample_value("ep_1:cnt_total") == 3.0


def test_prefix_metric_config_applies():
    cfg = {"models/*": EndpointMetricLogging(endpoint="models/*", metrics={
        "score": MetricType(type="scalar", buckets=[0, 0.5, 1.0])})}
    reg = CollectorRegistry()
    sr = StatsRegistry(processor=FakeProcessor(cfg), registry=reg)
    sr.report_batch([{"_url": "models/alpha/1", "score": 0.7}])
    assert reg.get_sample_value(
        "models_alpha_1:score_bucket", {"le": "1.0"}) == 1.0


def test_unconfigured_endpoint_gets_reserved_metrics():
    reg = CollectorRegistry()
    sr = StatsRegistry(processor=FakeProcessor({}), registry=reg)
    sr.report_batch([{"_url": "ghost", "_latency": 0.01, "_count": 1}])
    assert reg.get_sample_value("ghost:_count_total") == 1.0
    assert reg.get_sample_value("ghost:_latency_count") == 1.0


def test_batcher_stages_collector():
    from clearml_serving_amd.statistics.collector import BatcherStagesCollector

    class FakeBatcher:
        stats = {"batches": 4, "requests": 10, "occupancy_sum": 3.0,
                 "stage_ms_sum": 8.0, "gpu_wait_ms_sum": 12.0,
                 "queue_wait_ms_sum": 20.0}

    class FakeEngine:
        _batcher = FakeBatcher()

    class FakeProc:
        _engine_processor_lookup = {"ep/1": FakeEngine()}

    fams = {f.name: f for f in BatcherStagesCollector(FakeProc()).collect()}
    assert fams["serving_batches"].samples[0].value == 4
    assert abs(fams["serving_batch_occupancy"].samples[0].value - 0.75) < 1e-9
    assert fams["serving_queue_wait_ms"].samples[0].value == 20.0
    assert fams["serving_gpu_wait_ms"].samples[0].value == 12.0


def test_kafka_forwarder_with_fake_kafka(monkeypatch):
    """Kafka topology parity exercised with a stub kafka module: payloads
    are JSON batches on the reference topic; oversized batches split
    recursively in half (reference: model_request_processor.py:1097-1102)."""
    import json
    import sys
    import types

    sent = []

    class _Future:
        def get(self, timeout=None):
            return None

    class FakeProducer:
        max_bytes = 10_000

        def __init__(self, **kwargs):
            self.kwargs = kwargs

        def send(self, topic, payload):
            if len(payload) > self.max_bytes:
                raise fake_errors.MessageSizeTooLargeError()
            sent.append((topic, payload))
            return _Future()

    fake_kafka = types.ModuleType("kafka")
    fake_kafka.KafkaProducer = FakeProducer
    fake_errors = types.ModuleType("kafka.errors")

    class MessageSizeTooLargeError(Exception):
        pass

    fake_errors.MessageSizeTooLargeError = MessageSizeTooLargeError
    fake_kafka.errors = fake_errors
    monkeypatch.setitem(sys.modules, "kafka", fake_kafka)
    monkeypatch.setitem(sys.modules, "kafka.errors", fake_errors)

    from clearml_serving_amd.statistics.kafka_forwarder import (
        KafkaStatsForwarder,
    )

    fwd = KafkaStatsForwarder("kafka:9092")
    small = [{"_url": "ep", "_latency": 0.01, "_count": 1}] * 3
    fwd(small)
    assert len(sent) == 1
    topic, payload = sent[0]
    assert topic == "clearml_inference_stats"
    assert json.loads(payload) == small

    # oversized batch splits recursively until each chunk fits
    sent.clear()
    big = [{"_url": "ep", "pad": "x" * 200, "i": i} for i in range(100)]
    fwd(big)
    assert len(sent) >= 2
    recovered = []
    for _, p in sent:
        assert len(p) <= FakeProducer.max_bytes
        recovered.extend(json.loads(p))
    assert recovered == big


def test_standalone_service_consume_loop(monkeypatch, processor):
    """statistics/__main__.py consume(): Kafka batches land as Prometheus
    metrics through the same registry the in-process sink uses."""
    import json
    import sys
    import types

    msgs = [types.SimpleNamespace(value=json.dumps(
        [{"_url": "svc_ep", "_latency": 0.02, "_count": 2}]).encode())]

    fake_kafka = types.ModuleType("kafka")
    fake_kafka.KafkaConsumer = lambda *a, **k: iter(msgs)
    monkeypatch.setitem(sys.modules, "kafka", fake_kafka)

    from prometheus_client import CollectorRegistry

    from clearml_serving_amd.statistics.__main__ import consume
    from clearml_serving_amd.statistics.collector import StatsRegistry

    reg = CollectorRegistry()
    sr = StatsRegistry(processor=processor, registry=reg)
    n = consume(sr, "kafka:9092", max_batches=1)
    assert n == 1
    assert reg.get_sample_value("svc_ep:_count_total") == 2.0
