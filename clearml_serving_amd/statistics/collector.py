"""Per-request statistics -> Prometheus, metric-name compatible with the
reference's Grafana dashboards.

The reference ships per-request stat dicts over Kafka to a statistics
container that converts them to Prometheus metrics (reference:
clearml_serving/statistics/metrics.py:219-344). On a single MI355X node the
idiomatic equivalent is an in-process registry: same metric names
``{endpoint}:{variable}`` with non-alphanumerics folded to ``_``
(metrics.py:322-324), same reserved metrics ``_latency`` (12-bucket histogram
5 ms..5 s, metrics.py:189-192) and ``_count`` counter, same type mapping
scalar->histogram, enum->per-value counter, value->gauge, counter->counter.
A Kafka forwarder can be layered for multi-node topologies (kafka-python is
optional; absent in this image).

GPU-level gauges (HBM used/total, GPU utilization) are exported alongside --
the reference scrapes these from Triton's own exporter (triton_helper.py:45-89);
here they come straight from torch.cuda / amdsmi on the serving process.
"""

import os
import re
import threading
import time
from typing import Dict, List, Optional

from prometheus_client import (
    REGISTRY,
    Counter,
    Gauge,
    Histogram,
    start_http_server,
)

# reference reserved-latency buckets (statistics/metrics.py:190)
LATENCY_BUCKETS = (
    0.005, 0.01, 0.025, 0.05, 0.075, 0.1, 0.25, 0.5, 0.75, 1.0, 2.5, 5.0,
)
RESERVED = ("_latency", "_count", "_url")


def _prom_name(endpoint: str, variable: str) -> str:
    # "{url}:{var}" -- the reference's regex [^a-zA-Z0-9_:] KEEPS colons
    # (metrics.py:322-324), so reference-authored Grafana dashboards match
    return re.sub(r"[^a-zA-Z0-9_:]", "_", "{}:{}".format(endpoint, variable))


class StatsRegistry:
    """Dynamic Prometheus metric creation per endpoint/variable."""

    def __init__(self, processor=None, registry=None):
        self._observe_warned = set()
        self._processor = processor
        self._registry = registry or REGISTRY
        self._metrics: Dict[str, object] = {}
        self._types: Dict[str, dict] = {}
        self._lock = threading.Lock()

    def _metric_conf(self, endpoint: str, variable: str) -> dict:
        if variable == "_latency":
            return {"type": "scalar", "buckets": list(LATENCY_BUCKETS)}
        if variable == "_count":
            return {"type": "counter"}
        if self._processor is not None:
            logging_cfg = self._processor.list_endpoint_logging()
            cfg = logging_cfg.get(endpoint)
            if cfg is None:
                for name, c in logging_cfg.items():
                    if name.endswith("*") and endpoint.startswith(
                            name[:-1].rstrip("/")):
                        cfg = c
                        break
            if cfg is not None and variable in cfg.metrics:
                m = cfg.metrics[variable]
                return {"type": m.type, "buckets": m.buckets}
        return {"type": "value"}

    def _get_metric(self, endpoint: str, variable: str):
        key = _prom_name(endpoint, variable)
        with self._lock:
            if key in self._metrics:
                return self._metrics[key], self._types[key]
            conf = self._metric_conf(endpoint, variable)
            t = conf["type"]
            try:
                if t == "scalar":
                    m = Histogram(
                        key, "scalar metric {}".format(variable),
                        buckets=conf.get("buckets") or LATENCY_BUCKETS,
                        registry=self._registry,
                    )
                elif t == "enum":
                    m = Counter(
                        key, "enum metric {}".format(variable),
                        labelnames=("value",), registry=self._registry,
                    )
                elif t == "counter":
                    m = Counter(
                        key, "counter metric {}".format(variable),
                        registry=self._registry,
                    )
                else:
                    m = Gauge(
                        key, "value metric {}".format(variable),
                        registry=self._registry,
                    )
            except ValueError:
                # already registered in this process (e.g. two processors)
                m = None
            self._metrics[key] = m
            self._types[key] = conf
            return m, conf

    def report_batch(self, batch: List[dict]) -> None:
        for stats in batch:
            endpoint = stats.get("_url")
            if not endpoint:
                continue
            for variable, value in stats.items():
                if variable == "_url":
                    continue
                metric, conf = self._get_metric(endpoint, variable)
                if metric is None:
                    continue
                try:
                    self._observe(metric, conf["type"], value)
                except Exception as ex:
                    # stats must never break serving, but a silently
                    # mis-typed metric is an advisor trap: warn once per key
                    key = (endpoint, variable)
                    if key not in self._observe_warned:
                        self._observe_warned.add(key)
                        print("[stats] dropping metric {}:{} ({}: {})".format(
                            endpoint, variable, type(ex).__name__, ex))

    @staticmethod
    def _observe(metric, mtype: str, value) -> None:
        values = value if isinstance(value, (list, tuple)) else [value]
        for v in values:
            if mtype == "scalar":
                metric.observe(float(v))
            elif mtype == "enum":
                metric.labels(value=str(v)).inc()
            elif mtype == "counter":
                metric.inc(float(v))
            else:
                metric.set(float(v))


class GpuStatsExporter:
    """HBM + utilization gauges from the serving process itself.

    The reference scrapes these from Triton's own Prometheus exporter
    (triton_helper.py:45-89 -- nv_gpu_utilization / nv_gpu_memory_*);
    here HBM comes from torch.cuda and GPU/memory-controller activity
    from amdsmi (amdsmi_get_gpu_activity), when available."""

    def __init__(self, registry=None, interval_sec: float = 10.0):
        self._registry = registry or REGISTRY
        self._interval = interval_sec
        self._stop = False
        try:
            self._hbm_used = Gauge(
                "gpu_hbm_used_bytes", "HBM3E bytes in use", ("gpu",),
                registry=self._registry)
            self._hbm_total = Gauge(
                "gpu_hbm_total_bytes", "HBM3E bytes total", ("gpu",),
                registry=self._registry)
            self._gfx_util = Gauge(
                "gpu_utilization_percent", "GFX engine activity", ("gpu",),
                registry=self._registry)
            self._umc_util = Gauge(
                "gpu_memory_activity_percent",
                "memory-controller (UMC) activity", ("gpu",),
                registry=self._registry)
        except ValueError:
            self._hbm_used = self._hbm_total = None
            self._gfx_util = self._umc_util = None
        self._thread: Optional[threading.Thread] = None
        self._amdsmi = None
        self._amdsmi_devs = []

    def start(self):
        import torch

        if not torch.cuda.is_available() or self._hbm_used is None:
            return
        try:
            import amdsmi

            amdsmi.amdsmi_init()
            self._amdsmi = amdsmi
            self._amdsmi_devs = amdsmi.amdsmi_get_processor_handles()
        except Exception:
            self._amdsmi = None  # HBM-only export
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def _loop(self):
        import torch

        while not self._stop:
            try:
                for i in range(torch.cuda.device_count()):
                    free, total = torch.cuda.mem_get_info(i)
                    self._hbm_used.labels(gpu=str(i)).set(total - free)
                    self._hbm_total.labels(gpu=str(i)).set(total)
            except Exception:
                pass
            if self._amdsmi is not None:
                try:
                    for i, dev in enumerate(self._amdsmi_devs):
                        act = self._amdsmi.amdsmi_get_gpu_activity(dev)
                        gfx = act.get("gfx_activity")
                        umc = act.get("umc_activity")
                        if gfx is not None:
                            self._gfx_util.labels(gpu=str(i)).set(gfx)
                        if umc is not None:
                            self._umc_util.labels(gpu=str(i)).set(umc)
                except Exception:
                    pass
            time.sleep(self._interval)

    def stop(self):
        self._stop = True
        if self._amdsmi is not None:
            try:
                self._amdsmi.amdsmi_shut_down()
            except Exception:
                pass
            self._amdsmi = None


class BatcherStagesCollector:
    """Per-stage serving telemetry: queue-wait, batch
    staging and GPU time per endpoint, exported as _sum/_count pairs
    (rate(sum)/rate(count) gives the live mean in Prometheus)."""

    def __init__(self, processor):
        self._processor = processor

    def collect(self):
        from prometheus_client.core import (
            CounterMetricFamily,
            GaugeMetricFamily,
        )

        occ = GaugeMetricFamily(
            "serving_batch_occupancy", "mean batch fill of its bucket",
            labels=["endpoint"])
        batches = CounterMetricFamily(
            "serving_batches_total", "executed batches", labels=["endpoint"])
        fams = {
            s: CounterMetricFamily(
                "serving_{}_ms_total".format(s),
                "cumulative {} milliseconds".format(s), labels=["endpoint"])
            for s in ("queue_wait", "stage", "gpu_wait")
        }
        for url, engine in list(
                self._processor._engine_processor_lookup.items()):
            b = getattr(engine, "_batcher", None)
            if b is None or not b.stats["batches"]:
                continue
            s = b.stats
            label = _prom_name(url, "")[:-1] or url
            batches.add_metric([label], s["batches"])
            occ.add_metric([label], s["occupancy_sum"] / s["batches"])
            fams["queue_wait"].add_metric([label], s["queue_wait_ms_sum"])
            fams["stage"].add_metric([label], s["stage_ms_sum"])
            fams["gpu_wait"].add_metric([label], s["gpu_wait_ms_sum"])
        yield batches
        yield occ
        for f in fams.values():
            yield f


_http_started = False


def install_stats_sink(processor, port: Optional[int] = None) -> StatsRegistry:
    """Wire the processor's stats queue to Prometheus and expose /metrics.

    Port 9999 matches the reference statistics container's scrape port
    (statistics/main.py:35)."""
    global _http_started
    registry = StatsRegistry(processor=processor)
    processor.set_stats_sink(registry.report_batch)
    try:
        REGISTRY.register(BatcherStagesCollector(processor))
    except Exception:
        pass  # already registered in this process
    port = port if port is not None else int(
        os.environ.get("CLEARML_SERVING_STATS_PORT", 9999))
    if port > 0 and not _http_started:
        try:
            start_http_server(port)
            _http_started = True
        except OSError:
            pass  # busy port: another worker already exports
        try:
            exporter = GpuStatsExporter()
            exporter.start()
        except Exception:
            pass
    return registry
