"""Orchestration core: endpoint registry, canary routing, auto-update,
zero-downtime config reload, request dispatch, stats emission.

This is the MI355X build's equivalent of the reference's
``ModelRequestProcessor`` (reference: clearml_serving/serving/
model_request_processor.py:104-1569) with the ClearML Task replaced by the
local ``ServingStore``. Semantics preserved:

- endpoint key = "{serving_url}/{version}", version optional
  (model_request_processor.py:1455-1457)
- canary draw via numpy.random.choice before endpoint lookup (:279-281, 306-313)
- canary prefix mode selects the newest N endpoints by zero-padded version
  descending, first weight -> newest (:795-808)
- auto-update assigns monotonically increasing integer versions to newly seen
  model IDs and keeps only max_versions newest (:874-923)
- config hot-reload: set update flag, wait for the in-flight request counter to
  drain, swap tables, release; stalled requests spin+retry (:258-270, 700-720)
- lock-free hot path: request counting via paired itertools.count (atomic under
  the GIL), stats queue via deque + throttled Event (:58-101)
"""

import asyncio
import itertools
import logging
import os
import random
import socket
import threading
import time
from collections import deque
from typing import Any, Dict, List, Optional, Union

import numpy as np

from ..schemas import (
    CanaryEP,
    EndpointMetricLogging,
    ModelEndpoint,
    ModelMonitoring,
)
from ..store import ServingStore
from .preprocess import BasePreprocessRequest

logger = logging.getLogger("clearml_serving_amd.processor")


class FastWriteCounter:
    """Lock-free inc/dec counter: two itertools.count objects, each atomic
    under the GIL (same idiom as model_request_processor.py:58-70)."""

    def __init__(self):
        self._inc = itertools.count()
        self._dec = itertools.count()

    def inc(self):
        next(self._inc)

    def dec(self):
        next(self._dec)

    def value(self) -> int:
        # reading advances both counters equally, so the difference is exact
        return next(self._inc) - next(self._dec)


class FastSimpleQueue:
    """deque + throttled Event: producers notify at most every ``_notify_every``
    seconds, the consumer polls with a timeout (model_request_processor.py:73-101)."""

    _notify_every = 10.0

    def __init__(self):
        self._q = deque()
        self._event = threading.Event()
        self._last_notify = time.time()

    def put(self, obj) -> None:
        self._q.append(obj)
        if time.time() - self._last_notify > self._notify_every:
            self._last_notify = time.time()
            self._event.set()

    def get(self, timeout: Optional[float] = None):
        while True:
            try:
                return self._q.popleft()
            except IndexError:
                if not self._event.wait(timeout=timeout):
                    return None
                self._event.clear()


class EndpointNotFoundError(ValueError):
    pass


class ServingInitError(ValueError):
    pass


class ModelRequestProcessor:
    _config_key_endpoints = "endpoints"
    _config_key_monitoring = "model_monitoring"
    _config_key_canary = "canary"
    _config_key_metric_logging = "metric_logging"
    _config_key_monitoring_eps = "model_monitoring_eps"

    def __init__(
        self,
        task_id: Optional[str] = None,
        store: Optional[ServingStore] = None,
        update_lock_guard: Optional[threading.Lock] = None,
        name: Optional[str] = None,
        project: Optional[str] = None,
        tags: Optional[List[str]] = None,
        force_create: bool = False,
    ):
        self._store = store or ServingStore()
        if force_create or (task_id is None and name is not None):
            self._session_id = self._store.create_session(
                name=name or "Serving-Service", project=project or "DevOps",
                tags=tags,
            )
            from .. import __version__

            self._store.set_params(self._session_id,
                                   {"serving_version": __version__})
        else:
            self._session_id = self._store.resolve_session(task_id)
            # version guard (reference: __main__.py:24-40): warn when the
            # session was created by a different package version
            from .. import __version__

            created_by = self._store.get_params(self._session_id).get(
                "serving_version")
            if created_by and created_by.split(".")[0] !=                     __version__.split(".")[0]:
                print("Warning: serving session created by version {} but "
                      "this package is {}".format(created_by, __version__))
        # give engine instances access to store + session (preprocess.py)
        self._store._session_id = self._session_id

        self._endpoints: Dict[str, ModelEndpoint] = {}
        self._model_monitoring: Dict[str, ModelMonitoring] = {}
        self._model_monitoring_versions: Dict[str, Dict[str, int]] = {}
        self._model_monitoring_endpoints: Dict[str, ModelEndpoint] = {}
        self._canary_endpoints: Dict[str, CanaryEP] = {}
        self._canary_route: Dict[str, dict] = {}
        self._metric_logging: Dict[str, EndpointMetricLogging] = {}
        self._metric_cfg_cache: Dict[str, Any] = {}
        self._engine_processor_lookup: Dict[str, BasePreprocessRequest] = {}

        self._last_revision: Optional[int] = None
        self._update_lock_flag = False
        self._update_lock_guard = update_lock_guard or threading.Lock()
        self._request_processing_state = FastWriteCounter()
        self._stats_queue = FastSimpleQueue()
        self._stats_sink = None  # callable(list_of_dicts) -> None
        self._serving_base_url: Optional[str] = None
        self._metric_log_freq: float = float(
            os.environ.get("CLEARML_DEFAULT_METRIC_LOG_FREQ", 1.0)
        )
        self._sync_daemon_thread: Optional[threading.Thread] = None
        self._stats_send_thread: Optional[threading.Thread] = None
        self._stop = False
        self._instance_messages: List[str] = []
        # per-endpoint request/response counters (reference endpoint
        # telemetry, model_request_processor.py:162-187 -- there it feeds
        # the ClearML router UI; here GET /status + Prometheus read it)
        self._endpoint_telemetry: Dict[str, dict] = {}
        self._enable_endpoint_telemetry = os.environ.get(
            "CLEARML_ENABLE_ENDPOINT_TELEMETRY", "1") != "0"

    # ------------------------------------------------------------------ #
    # identity / config
    # ------------------------------------------------------------------ #
    @property
    def instance_id(self) -> str:
        return "{}:{}".format(socket.gethostname(), os.getpid())

    def get_id(self) -> str:
        return self._session_id

    @property
    def store(self) -> ServingStore:
        return self._store

    def get_endpoints(self) -> Dict[str, ModelEndpoint]:
        return dict(self._endpoints)

    def get_synced_endpoints(self) -> Dict[str, ModelEndpoint]:
        """All materialized endpoints: static + monitored versions
        (reference: get_synced_endpoints, used by the engine manager)."""
        out = dict(self._endpoints)
        out.update(self._model_monitoring_endpoints)
        return out

    def get_canary_endpoints(self) -> Dict[str, CanaryEP]:
        return dict(self._canary_endpoints)

    def get_model_monitoring(self) -> Dict[str, ModelMonitoring]:
        return dict(self._model_monitoring)

    def get_metric_logging(self) -> Dict[str, EndpointMetricLogging]:
        return dict(self._metric_logging)

    def set_metric_log_freq(self, value: float) -> None:
        self._metric_log_freq = float(value)
        self._store.set_params(self._session_id, {"metric_logging_freq": value})

    def configure(
        self,
        external_serving_base_url: Optional[str] = None,
        external_kafka_service_server: Optional[str] = None,
        default_metric_log_freq: Optional[float] = None,
    ) -> None:
        params = {}
        if external_serving_base_url is not None:
            params["serving_base_url"] = external_serving_base_url
        if external_kafka_service_server is not None:
            params["kafka_service_server"] = external_kafka_service_server
        if default_metric_log_freq is not None:
            params["metric_logging_freq"] = default_metric_log_freq
            self._metric_log_freq = float(default_metric_log_freq)
        if params:
            self._store.set_params(self._session_id, params)

    # ------------------------------------------------------------------ #
    # mutations (CLI surface)
    # ------------------------------------------------------------------ #
    def add_endpoint(
        self,
        endpoint: Union[ModelEndpoint, dict],
        preprocess_code: Optional[str] = None,
        model_name: Optional[str] = None,
        model_project: Optional[str] = None,
        model_tags: Optional[List[str]] = None,
        model_published: Optional[bool] = None,
    ) -> str:
        if not isinstance(endpoint, ModelEndpoint):
            endpoint = ModelEndpoint(**endpoint)

        url = self._normalize_endpoint_url(endpoint.serving_url, endpoint.version)
        if url in self._endpoints:
            print("Warning: Model endpoint '{}' overwritten".format(url))

        if not endpoint.model_id and (model_name or model_project or model_tags):
            models = self._store.query_models(
                project=model_project, name=model_name, tags=model_tags,
                only_published=bool(model_published), max_results=1,
            )
            if not models:
                raise ValueError(
                    "Could not find any model to serve (name={} project={} "
                    "tags={} published={})".format(
                        model_name, model_project, model_tags, model_published
                    )
                )
            endpoint.model_id = models[0].model_id
        if endpoint.model_id and not self._store.get_model(endpoint.model_id):
            raise ValueError("model id '{}' not found".format(endpoint.model_id))

        if preprocess_code:
            endpoint.preprocess_artifact = self._upload_preprocess(
                url, preprocess_code
            )
        self._endpoints[url] = endpoint
        return url

    def remove_endpoint(self, endpoint_url: str) -> bool:
        endpoint_url = self._normalize_endpoint_url(endpoint_url)
        for table in (self._endpoints, self._model_monitoring,
                      self._canary_endpoints):
            if endpoint_url in table:
                table.pop(endpoint_url, None)
                return True
        return False

    def add_model_monitoring(
        self,
        monitoring: Union[ModelMonitoring, dict],
        preprocess_code: Optional[str] = None,
    ) -> str:
        if not isinstance(monitoring, ModelMonitoring):
            monitoring = ModelMonitoring(**monitoring)
        url = self._normalize_endpoint_url(monitoring.base_serving_url)
        if url in self._model_monitoring:
            print("Warning: Model monitoring '{}' overwritten".format(url))
        if preprocess_code:
            monitoring.preprocess_artifact = self._upload_preprocess(
                url, preprocess_code
            )
        monitoring.base_serving_url = url
        self._model_monitoring[url] = monitoring
        return url

    def remove_model_monitoring(self, model_base_url: str) -> bool:
        url = self._normalize_endpoint_url(model_base_url)
        if url not in self._model_monitoring:
            return False
        self._model_monitoring.pop(url, None)
        return True

    def add_canary_endpoint(self, canary: Union[CanaryEP, dict]) -> str:
        if not isinstance(canary, CanaryEP):
            canary = CanaryEP(**canary)
        url = self._normalize_endpoint_url(canary.endpoint)
        if url in self._canary_endpoints:
            print("Warning: Canary endpoint '{}' overwritten".format(url))
        if canary.load_endpoints and canary.load_endpoint_prefix:
            raise ValueError(
                "Could not add canary endpoint with both fixed endpoints and "
                "endpoint prefix"
            )
        if canary.load_endpoints and len(canary.load_endpoints) != len(canary.weights):
            raise ValueError("Canary endpoints and weights must match in length")
        canary.endpoint = url
        self._canary_endpoints[url] = canary
        return url

    def remove_canary_endpoint(self, endpoint_url: str) -> bool:
        return self._canary_endpoints.pop(
            self._normalize_endpoint_url(endpoint_url), None
        ) is not None

    def add_metric_logging(self, metric: Union[EndpointMetricLogging, dict]) -> bool:
        if not isinstance(metric, EndpointMetricLogging):
            metric = EndpointMetricLogging(**metric)
        name = str(metric.endpoint).strip("/")
        metric.endpoint = name
        if "*" not in name and name not in self.get_synced_endpoints() \
                and name not in self._model_monitoring:
            raise ValueError("Endpoint '{}' not found".format(name))
        if name in self._metric_logging:
            print("Warning: Metric logging '{}' overwritten".format(name))
        self._metric_logging[name] = metric
        return True

    def remove_metric_logging(
        self, endpoint: str, variable_names: Optional[List[str]] = None
    ) -> bool:
        name = str(endpoint).strip("/")
        if name not in self._metric_logging:
            return False
        if not variable_names:
            self._metric_logging.pop(name, None)
        else:
            entry = self._metric_logging[name]
            for v in variable_names:
                entry.metrics.pop(v, None)
        return True

    def list_endpoint_logging(self) -> Dict[str, EndpointMetricLogging]:
        return dict(self._metric_logging)

    # reference API parity (model_request_processor.py:592-599): same
    # content as list_endpoint_logging against the local store
    def list_metric_logging(self) -> Dict[str, EndpointMetricLogging]:
        return dict(self._metric_logging)

    def get_configuration(self) -> Dict[str, Any]:
        """Session parameters (reference :371-372: the General/* section)."""
        return self._store.get_params(self._session_id)

    def get_version(self) -> str:
        """Serving package version recorded on the session at create time
        (reference :762-770)."""
        return self._store.get_params(self._session_id).get(
            "serving_version") or "1.0.0"

    def reload(self) -> None:
        """Force a full state reload from the store (reference :734-739),
        bypassing the revision no-op check."""
        self._last_revision = None
        self.deserialize(skip_sync=False)

    @classmethod
    def list_control_plane_tasks(
        cls, name: Optional[str] = None, project: Optional[str] = None,
        tags: Optional[List[str]] = None,
    ) -> List[dict]:
        """List serving sessions ("control plane tasks"), optionally
        filtered (reference :1372-1395)."""
        out = []
        for s in ServingStore().list_sessions():
            if name and name not in s["name"]:
                continue
            if project and project != s["project"]:
                continue
            if tags and not set(tags).issubset(set(s["tags"])):
                continue
            out.append(s)
        return out

    # public telemetry hooks (reference :165-187); process_request calls
    # the counters directly, these exist for API compatibility
    def on_request_endpoint_telemetry(self, base_url=None,
                                      version=None) -> None:
        t = self._telemetry_counters(
            self._normalize_endpoint_url(base_url or "", version))
        if t:
            t["requests"].inc()

    def on_response_endpoint_telemetry(self, base_url=None,
                                       version=None) -> None:
        t = self._telemetry_counters(
            self._normalize_endpoint_url(base_url or "", version))
        if t:
            t["responses"].inc()

    def _upload_preprocess(self, url: str, preprocess_code: str) -> str:
        if not os.path.exists(preprocess_code):
            raise ValueError(
                "Preprocess code '{}' not found".format(preprocess_code)
            )
        artifact_name = "py_code_{}".format(url.replace("/", "_"))
        self._store.upload_artifact(self._session_id, artifact_name, preprocess_code)
        return artifact_name

    # ------------------------------------------------------------------ #
    # persistence (serialize/deserialize + hot reload)
    # ------------------------------------------------------------------ #
    def serialize(self) -> None:
        s, sid = self._store, self._session_id
        s.set_config_object(
            sid, self._config_key_endpoints,
            {k: v.as_dict(remove_null_entries=True)
             for k, v in self._endpoints.items()},
        )
        s.set_config_object(
            sid, self._config_key_monitoring,
            {k: v.as_dict(remove_null_entries=True)
             for k, v in self._model_monitoring.items()},
        )
        s.set_config_object(
            sid, self._config_key_canary,
            {k: v.as_dict(remove_null_entries=True)
             for k, v in self._canary_endpoints.items()},
        )
        s.set_config_object(
            sid, self._config_key_metric_logging,
            {k: v.as_dict(remove_null_entries=True)
             for k, v in self._metric_logging.items()},
        )

    def _serialize_monitoring_eps(self) -> None:
        self._store.set_config_object(
            self._session_id, self._config_key_monitoring_eps,
            {
                "endpoints": {
                    k: v.as_dict(remove_null_entries=True)
                    for k, v in self._model_monitoring_endpoints.items()
                },
                "versions": self._model_monitoring_versions,
            },
        )

    def deserialize(
        self, skip_sync: bool = False, update_current_task: bool = True,
    ) -> bool:
        """Reload state from the store. Returns True when anything changed.

        When change is detected while serving, runs the reference's
        zero-downtime stall-swap protocol (:700-720)."""
        revision = self._store.revision(self._session_id)
        if self._last_revision is not None and revision == self._last_revision:
            return False

        s = self._store
        endpoints = {
            k: ModelEndpoint.from_dict(v)
            for k, v in (s.get_config_object(
                self._session_id, self._config_key_endpoints, {}) or {}).items()
        }
        monitoring = {
            k: ModelMonitoring.from_dict(v)
            for k, v in (s.get_config_object(
                self._session_id, self._config_key_monitoring, {}) or {}).items()
        }
        canary = {
            k: CanaryEP.from_dict(v)
            for k, v in (s.get_config_object(
                self._session_id, self._config_key_canary, {}) or {}).items()
        }
        metric_logging = {
            k: EndpointMetricLogging.from_dict(v)
            for k, v in (s.get_config_object(
                self._session_id, self._config_key_metric_logging, {}) or {}).items()
        }
        mon_eps_blob = s.get_config_object(
            self._session_id, self._config_key_monitoring_eps, {}) or {}
        mon_eps = {
            k: ModelEndpoint.from_dict(v)
            for k, v in (mon_eps_blob.get("endpoints") or {}).items()
        }
        mon_versions = mon_eps_blob.get("versions") or {}
        params = s.get_params(self._session_id)

        if skip_sync:
            self._endpoints = endpoints
            self._model_monitoring = monitoring
            self._canary_endpoints = canary
            self._metric_logging = metric_logging
            self._model_monitoring_endpoints = mon_eps
            self._model_monitoring_versions = mon_versions
            self._apply_params(params)
            self._last_revision = revision
            return True

        # --- stall-swap: block new requests, drain in-flight, swap -------- #
        with self._update_lock_guard:
            self._update_lock_flag = True
            t0 = time.time()
            while self._request_processing_state.value() > 0:
                if time.time() - t0 > 60.0:
                    break
                time.sleep(0.05)
            self._endpoints = endpoints
            self._model_monitoring = monitoring
            self._canary_endpoints = canary
            self._metric_logging = metric_logging
            self._model_monitoring_endpoints = mon_eps
            self._model_monitoring_versions = mon_versions
            self._apply_params(params)
            self._update_canary_lookup()
            # flush engine instances whose endpoint config OR preprocess
            # artifact content changed (the reference hashes artifacts into
            # its config hash, model_request_processor.py:636-654)
            synced = self.get_synced_endpoints()
            for url in list(self._engine_processor_lookup.keys()):
                cached = self._engine_processor_lookup[url]
                ep = synced.get(url)
                stale = ep is None or \
                    ep.as_dict() != cached.model_endpoint.as_dict()
                if not stale and ep.preprocess_artifact:
                    art = self._store.get_artifact(self._session_id,
                                                   ep.preprocess_artifact)
                    stale = art is not None and \
                        art["sha256"] != cached._artifact_sha
                if stale:
                    engine = self._engine_processor_lookup.pop(url, None)
                    batcher = getattr(engine, "_batcher", None)
                    if batcher is not None:
                        batcher.shutdown()  # free graphs/streams/HBM
                    shutdown = getattr(engine, "shutdown", None)
                    if callable(shutdown):
                        shutdown()  # e.g. LLM engine refcount/teardown
            self._metric_cfg_cache.clear()
            self._last_revision = revision
            self._update_lock_flag = False
        import gc

        gc.collect()
        return True

    def _apply_params(self, params: Dict[str, Any]) -> None:
        if "serving_base_url" in params:
            self._serving_base_url = params["serving_base_url"]
        if "metric_logging_freq" in params and params["metric_logging_freq"]:
            self._metric_log_freq = float(params["metric_logging_freq"])

    # ------------------------------------------------------------------ #
    # canary routing
    # ------------------------------------------------------------------ #
    def _update_canary_lookup(self) -> None:
        canary_route: Dict[str, dict] = {}
        synced = self.get_synced_endpoints()
        for url, canary in self._canary_endpoints.items():
            if canary.load_endpoints:
                total = sum(canary.weights) or 1.0
                canary_route[url] = {
                    "endpoints": list(canary.load_endpoints),
                    "weights": [w / total for w in canary.weights],
                }
            elif canary.load_endpoint_prefix:
                prefix = canary.load_endpoint_prefix.rstrip("/")
                matching = [
                    ep for ep in synced
                    if ep == prefix or ep.startswith(prefix + "/")
                ]
                # newest first: sort by zero-padded version descending
                # (reference: model_request_processor.py:795-808)
                def _version_key(ep: str) -> str:
                    ver = ep[len(prefix):].strip("/")
                    return ver.zfill(32)

                matching = sorted(matching, key=_version_key, reverse=True)
                selected = matching[: len(canary.weights)]
                weights = canary.weights[: len(selected)]
                total = sum(weights) or 1.0
                canary_route[url] = {
                    "endpoints": selected,
                    "weights": [w / total for w in weights],
                }
        self._canary_route = canary_route

    def _process_canary(self, base_url: str) -> Optional[dict]:
        route = self._canary_route.get(base_url)
        if not route or not route["endpoints"]:
            return None
        ep = np.random.choice(route["endpoints"], p=route["weights"])
        return {"endpoint": str(ep)}

    # ------------------------------------------------------------------ #
    # auto-update (monitored models)
    # ------------------------------------------------------------------ #
    def _update_monitored_models(self) -> bool:
        """Query the model registry for every monitoring spec; assign integer
        versions to newly seen model IDs; keep max_versions newest; materialize
        endpoints under {base_url}/{version} (reference :836-923)."""
        dirty = False
        new_eps: Dict[str, ModelEndpoint] = {}
        for base_url, mon in self._model_monitoring.items():
            versions = self._model_monitoring_versions.get(base_url, {})
            models = self._store.query_models(
                project=mon.monitor_project or None,
                name=mon.monitor_name or None,
                tags=mon.monitor_tags or None,
                only_published=mon.only_published,
                max_results=max(mon.max_versions or 1, 1) * 4,
            )
            # oldest-first so versions increase monotonically with recency
            for rec in reversed(models):
                if rec.model_id not in versions:
                    next_ver = max(versions.values(), default=0) + 1
                    versions[rec.model_id] = next_ver
                    dirty = True
            # keep only max_versions newest versions
            max_v = max(mon.max_versions or 1, 1)
            keep = dict(
                sorted(versions.items(), key=lambda kv: kv[1], reverse=True)[:max_v]
            )
            if keep != self._model_monitoring_versions.get(base_url):
                dirty = True
            self._model_monitoring_versions[base_url] = keep

            for model_id, ver in keep.items():
                url = "{}/{}".format(base_url, ver)
                new_eps[url] = ModelEndpoint(
                    engine_type=mon.engine_type,
                    serving_url=base_url,
                    model_id=model_id,
                    version=str(ver),
                    preprocess_artifact=mon.preprocess_artifact,
                    input_size=mon.input_size, input_type=mon.input_type,
                    input_name=mon.input_name, output_size=mon.output_size,
                    output_type=mon.output_type, output_name=mon.output_name,
                    auxiliary_cfg=mon.auxiliary_cfg,
                )
        # drop version maps for removed monitors
        for base_url in list(self._model_monitoring_versions.keys()):
            if base_url not in self._model_monitoring:
                self._model_monitoring_versions.pop(base_url)
                dirty = True
        if new_eps != self._model_monitoring_endpoints:
            dirty = True
        self._model_monitoring_endpoints = new_eps
        if dirty:
            self._update_canary_lookup()
            self._serialize_monitoring_eps()
            # our own serialize bumped the revision; absorb it
            self._last_revision = self._store.revision(self._session_id)
        return dirty

    # ------------------------------------------------------------------ #
    # request dispatch (hot path)
    # ------------------------------------------------------------------ #
    def _telemetry_counters(self, url: str) -> Optional[dict]:
        if not self._enable_endpoint_telemetry:
            return None
        t = self._endpoint_telemetry.get(url)
        if t is None:
            t = self._endpoint_telemetry.setdefault(
                url, {"requests": FastWriteCounter(),
                      "responses": FastWriteCounter()})
        return t

    def endpoint_telemetry_snapshot(self) -> Dict[str, dict]:
        """{url: {requests, responses, in_flight}} -- reading advances the
        lock-free counters equally, so snapshots stay exact."""
        out = {}
        for url, t in list(self._endpoint_telemetry.items()):
            req = t["requests"].value()
            resp = t["responses"].value()
            out[url] = {"requests": req, "responses": resp,
                        "in_flight": max(req - resp, 0)}
        return out

    async def process_request(
        self, base_url: str, version: str, request_body: Any,
        serve_type: str = "process",
    ) -> Any:
        self._request_processing_state.inc()
        # config-update stall: spin until the swap completes (:258-270)
        while self._update_lock_flag:
            self._request_processing_state.dec()
            await asyncio.sleep(0.5 + random.random())
            self._request_processing_state.inc()

        telemetry = self._telemetry_counters(
            self._normalize_endpoint_url(base_url, version))
        if telemetry:
            telemetry["requests"].inc()
        try:
            result = await self._process_request_inner(
                base_url=base_url, version=version, request_body=request_body,
                serve_type=serve_type,
            )
            if telemetry:
                telemetry["responses"].inc()
            return result
        finally:
            self._request_processing_state.dec()

    async def _process_request_inner(
        self, base_url: str, version: str, request_body: Any, serve_type: str,
    ) -> Any:
        url = self._normalize_endpoint_url(base_url, version)

        # canary resolution happens before endpoint lookup (:279-281)
        canary = self._process_canary(base_url=url)
        if canary:
            url = canary["endpoint"]

        ep = self._endpoints.get(url) or self._model_monitoring_endpoints.get(url)
        if not ep:
            raise EndpointNotFoundError("Model inference endpoint '{}' not found".format(url))

        processor = self._engine_processor_lookup.get(url)
        if not processor:
            processor_cls = BasePreprocessRequest.get_engine_cls(ep.engine_type)
            processor = processor_cls(model_endpoint=ep, task=self._store)
            # inject pipeline fan-out hook (send_request) bound to this
            # processor so ensembles dispatch in-process, no HTTP hop
            # (reference goes through HTTP, preprocess_service.py:255-264;
            # in-process dispatch is the single-node-native equivalent).
            self._bind_send_request(processor)
            self._engine_processor_lookup[url] = processor

        return await self._do_process_request(
            processor, url=url, body=request_body, serve_type=serve_type
        )

    def _bind_send_request(self, processor: BasePreprocessRequest) -> None:
        if processor._preprocess is None:
            return

        # the loop serving requests: sync send_request (user threads, e.g.
        # the reference's ThreadPoolExecutor pipeline,
        # examples/pipeline/preprocess.py:18-32) schedules onto it
        try:
            serving_loop = asyncio.get_running_loop()
        except RuntimeError:
            serving_loop = None

        if processor.is_process_async:
            async def _send(endpoint, version=None, data=None):
                try:
                    return await self.process_request(
                        base_url=endpoint, version=version, request_body=data
                    )
                except Exception:
                    # reference parity: pipeline fan-out failures return None
                    # (the reference's HTTP hop does the same,
                    # preprocess_service.py:255-264) -- but log them loudly
                    logger.exception(
                        "pipeline send_request to '%s' failed", endpoint)
                    return None
        else:
            def _send(endpoint, version=None, data=None):
                coro = self.process_request(
                    base_url=endpoint, version=version, request_body=data)
                try:
                    asyncio.get_running_loop()
                except RuntimeError:
                    pass  # off-loop (worker thread): the supported sync path
                else:
                    # called inline ON the event loop: blocking here would
                    # deadlock (the loop can't progress the inner request).
                    # Fail loudly instead of silently returning None.
                    coro.close()
                    raise RuntimeError(
                        "sync send_request called on the serving event loop; "
                        "call it from a worker thread (ThreadPoolExecutor) or "
                        "use engine_type=custom_async with "
                        "'await self.send_request(...)'")
                try:
                    if serving_loop is not None and not serving_loop.is_closed():
                        fut = asyncio.run_coroutine_threadsafe(
                            coro, serving_loop)
                        return fut.result()
                    # no serving loop captured (offline/unit-test use): run
                    # the coroutine on a private loop in this thread
                    loop = asyncio.new_event_loop()
                    try:
                        return loop.run_until_complete(coro)
                    finally:
                        loop.close()
                except Exception:
                    logger.exception(
                        "pipeline send_request to '%s' failed", endpoint)
                    return None

        processor._preprocess.send_request = _send

    def _resolve_metric_cfg(self, url: str):
        # per-request prefix scans are hot-path cost: resolve once per url
        # (cache invalidated on config reload)
        try:
            return self._metric_cfg_cache[url]
        except KeyError:
            cfg = self._metric_logging.get(url) or self._prefix_metric(url)
            self._metric_cfg_cache[url] = cfg
            return cfg

    async def _do_process_request(
        self, processor: BasePreprocessRequest, url: str, body: Any,
        serve_type: str,
    ) -> Any:
        state: Dict[str, Any] = {}

        # metric sampling decision (:1316-1323)
        freq = self._metric_log_freq
        metric_cfg = self._resolve_metric_cfg(url)
        if metric_cfg and metric_cfg.log_frequency is not None:
            freq = metric_cfg.log_frequency
        collect = bool(freq) and (freq >= 1.0 or random.random() <= freq)
        tic = time.time() if collect else 0.0
        stats: Dict[str, Any] = {}
        collect_fn = stats.update if collect else None

        if processor.is_preprocess_async:
            preprocessed = await processor.preprocess(body, state, collect_fn)
        else:
            preprocessed = processor.preprocess(body, state, collect_fn)

        if serve_type == "process":
            if processor.is_process_async:
                processed = await processor.process(preprocessed, state, collect_fn)
            elif processor.run_process_off_loop:
                # blocking user code (custom engine / threaded pipelines):
                # keep the event loop live so sync send_request can schedule
                # nested requests onto it
                processed = await asyncio.to_thread(
                    processor.process, preprocessed, state, collect_fn)
            else:
                processed = processor.process(preprocessed, state, collect_fn)
        else:
            # OpenAI-style route: method name from serve_type
            # ("v1/chat/completions" -> processor.v1_chat_completions,
            # reference :1331)
            method = getattr(processor, serve_type.replace("/", "_"), None)
            if method is None:
                raise EndpointNotFoundError(
                    "engine '{}' does not implement serve type '{}'".format(
                        processor.__class__.__name__, serve_type
                    )
                )
            processed = await method(preprocessed, state, collect_fn)

        if processor.is_postprocess_async:
            result = await processor.postprocess(processed, state, collect_fn)
        else:
            result = processor.postprocess(processed, state, collect_fn)

        if collect:
            stats["_latency"] = round(time.time() - tic, 4)
            stats["_count"] = int(1.0 / freq)
            stats["_url"] = url
            # auto-log configured request/response variables (:1341-1360)
            if metric_cfg:
                names = set(metric_cfg.metrics.keys())
                if isinstance(body, dict):
                    stats.update({k: v for k, v in body.items() if k in names})
                if isinstance(result, dict):
                    stats.update({k: v for k, v in result.items() if k in names})
            self._stats_queue.put(stats)
        return result

    def _prefix_metric(self, url: str) -> Optional[EndpointMetricLogging]:
        for name, cfg in self._metric_logging.items():
            if name.endswith("*") and url.startswith(name[:-1].rstrip("/")):
                return cfg
        return None

    # ------------------------------------------------------------------ #
    # service lifecycle
    # ------------------------------------------------------------------ #
    def launch(self, poll_frequency_sec: float = 300.0) -> None:
        """Bootstrap: full deserialize + monitored-model sync + background
        sync daemon and stats-send loop (reference launch :951-982)."""
        self.deserialize(skip_sync=True)
        self._update_monitored_models()
        self._update_canary_lookup()
        self._stop = False
        self._sync_daemon_thread = threading.Thread(
            target=self._sync_daemon, args=(poll_frequency_sec,), daemon=True
        )
        self._stats_send_thread = threading.Thread(
            target=self._stats_send_loop, daemon=True
        )
        self._sync_daemon_thread.start()
        self._stats_send_thread.start()

    def stop(self) -> None:
        self._stop = True

    def _sync_daemon(self, poll_frequency_sec: float) -> None:
        while not self._stop:
            try:
                # keep-alive: this serving instance announces itself (the
                # reference pings the control Task, :999-1007); no revision
                # bump, so other instances don't see it as a config change
                self._store.ping_instance(
                    self._session_id, self.instance_id,
                    {"pid": os.getpid(), "host": socket.gethostname()})
                changed = self.deserialize(skip_sync=False)
                changed |= self._update_monitored_models()
                if changed:
                    self._report_text("serving config synced (revision {})".format(
                        self._last_revision))
            except Exception as ex:
                self._report_text("sync daemon error: {}".format(ex))
            # fine-grained sleep so stop() is responsive
            t0 = time.time()
            while not self._stop and time.time() - t0 < poll_frequency_sec:
                time.sleep(0.5)

    def _stats_send_loop(self) -> None:
        while not self._stop:
            batch = []
            obj = self._stats_queue.get(timeout=10.0)
            while obj is not None:
                batch.append(obj)
                obj = self._stats_queue.get(timeout=0.0) if self._stats_queue._q \
                    else None
            if batch and self._stats_sink:
                try:
                    self._stats_sink(batch)
                except Exception as ex:
                    self._report_text("stats sink error: {}".format(ex))

    def set_stats_sink(self, sink) -> None:
        """sink: callable(list_of_stat_dicts). The statistics module installs
        an in-process Prometheus sink; a Kafka forwarder can be set instead
        for reference-topology parity."""
        self._stats_sink = sink

    def _report_text(self, msg: str) -> None:
        self._instance_messages.append(msg)
        print("[clearml-serving-amd] {}".format(msg))

    # ------------------------------------------------------------------ #
    @classmethod
    def _normalize_endpoint_url(cls, endpoint: str, version: Optional[str] = None) -> str:
        # reference: model_request_processor.py:1455-1457
        return "{}/{}".format(endpoint.rstrip("/"), version or "").rstrip("/")
