"""Typed endpoint/monitoring/canary/metric structs for the serving session.

Semantics mirror the reference structs (reference: clearml_serving/serving/
endpoints.py:44-124) -- same field names, same list/nested-list conversions,
same dtype validation against numpy -- implemented as plain dataclasses so the
control-plane store can round-trip them as JSON.
"""

import dataclasses
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import numpy as np


def _validate_engine(value: str) -> None:
    # deferred import: the engine registry self-populates on package import
    from .serving.preprocess import BasePreprocessRequest

    if not BasePreprocessRequest.validate_engine_type(value):
        # GPU engines register from the engines package; import it lazily so
        # plain schema use does not pull in torch
        try:
            from . import engines  # noqa: F401
        except Exception:
            pass
    if not BasePreprocessRequest.validate_engine_type(value):
        raise TypeError("{} not supported engine type".format(value))


def _validate_matrix_type(value) -> None:
    # mirrors endpoints.py:11-18 -- every entry must be a valid numpy dtype name
    if value is None:
        return
    values = value if isinstance(value, (tuple, list)) else [value]
    for v in values:
        if v:
            np.dtype(v)  # raises TypeError on unknown names


def _to_list(value) -> Optional[list]:
    if value is None:
        return None
    return list(value) if isinstance(value, (tuple, list)) else [value]


def _to_nested_list(value) -> Optional[list]:
    # a flat [d0, d1, ...] becomes [[d0, d1, ...]] (single-input shorthand,
    # endpoints.py:27-33)
    if value is None:
        return None
    if isinstance(value, (tuple, list)) and all(
        not isinstance(i, (tuple, list)) for i in value
    ):
        return [list(value)]
    value = value if isinstance(value, (tuple, list)) else [value]
    return [list(v) if isinstance(v, (tuple, list)) else v for v in value]


class BaseStruct:
    def as_dict(self, remove_null_entries: bool = False) -> Dict[str, Any]:
        d = dataclasses.asdict(self)
        if remove_null_entries:
            d = {k: v for k, v in d.items() if v is not None}
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "BaseStruct":
        names = {f.name for f in dataclasses.fields(cls)}
        return cls(**{k: v for k, v in d.items() if k in names})


@dataclass
class ModelEndpoint(BaseStruct):
    """One serving endpoint: engine + model + optional I/O spec.

    Field-for-field the reference's ModelEndpoint (endpoints.py:64-78).
    """

    engine_type: str
    serving_url: str
    model_id: Optional[str] = None
    version: str = ""
    preprocess_artifact: Optional[str] = None
    input_size: Optional[list] = None
    input_type: Optional[list] = None
    input_name: Optional[list] = None
    output_size: Optional[list] = None
    output_type: Optional[list] = None
    output_name: Optional[list] = None
    auxiliary_cfg: Optional[Any] = None

    def __post_init__(self):
        _validate_engine(self.engine_type)
        self.input_size = _to_nested_list(self.input_size)
        self.output_size = _to_nested_list(self.output_size)
        self.input_type = _to_list(self.input_type)
        self.output_type = _to_list(self.output_type)
        _validate_matrix_type(self.input_type)
        _validate_matrix_type(self.output_type)
        self.input_name = _to_list(self.input_name)
        self.output_name = _to_list(self.output_name)


@dataclass
class ModelMonitoring(BaseStruct):
    """Auto-update spec: watch the model registry, materialize versioned
    endpoints under base_serving_url/{version} (endpoints.py:44-61)."""

    base_serving_url: str
    engine_type: str
    monitor_project: Optional[str] = None
    monitor_name: Optional[str] = None
    monitor_tags: List[str] = field(default_factory=list)
    only_published: bool = False
    max_versions: Optional[int] = None
    input_size: Optional[list] = None
    input_type: Optional[list] = None
    input_name: Optional[list] = None
    output_size: Optional[list] = None
    output_type: Optional[list] = None
    output_name: Optional[list] = None
    preprocess_artifact: Optional[str] = None
    auxiliary_cfg: Optional[Any] = None

    def __post_init__(self):
        _validate_engine(self.engine_type)
        self.input_size = _to_nested_list(self.input_size)
        self.output_size = _to_nested_list(self.output_size)
        self.input_type = _to_list(self.input_type)
        self.output_type = _to_list(self.output_type)
        _validate_matrix_type(self.input_type)
        _validate_matrix_type(self.output_type)
        self.input_name = _to_list(self.input_name)
        self.output_name = _to_list(self.output_name)


@dataclass
class CanaryEP(BaseStruct):
    """Canary/A-B routing entry (endpoints.py:81-88): either a fixed endpoint
    list + weights, or a prefix resolved to the newest N versions at sync."""

    endpoint: str
    weights: List[float]
    load_endpoints: List[str] = field(default_factory=list)
    load_endpoint_prefix: Optional[str] = None


@dataclass
class MetricType(BaseStruct):
    type: str
    buckets: Optional[list] = None

    _VALID = ("scalar", "enum", "value", "counter")

    def __post_init__(self):
        if self.type not in self._VALID:
            raise TypeError(
                "metric type '{}' not in {}".format(self.type, self._VALID)
            )


@dataclass
class EndpointMetricLogging(BaseStruct):
    """Per-endpoint metric logging config (endpoints.py:91-124).

    ``endpoint`` may end with '*' for prefix matching; ``metrics`` maps
    variable name -> MetricType.
    """

    endpoint: str
    log_frequency: Optional[float] = None
    metrics: Dict[str, MetricType] = field(default_factory=dict)

    def __post_init__(self):
        self.metrics = {
            k: v if isinstance(v, MetricType) else MetricType(**v)
            for k, v in (self.metrics or {}).items()
        }

    def as_dict(self, remove_null_entries: bool = False) -> Dict[str, Any]:
        d = {
            "endpoint": self.endpoint,
            "log_frequency": self.log_frequency,
            "metrics": {
                k: v.as_dict(remove_null_entries) for k, v in self.metrics.items()
            },
        }
        if remove_null_entries:
            d = {k: v for k, v in d.items() if v is not None}
        return d


@dataclass
class ModelRecord(BaseStruct):
    """A model registry entry -- our local replacement for the ClearML Model
    object (the reference resolves models via clearml Model.query_models,
    model_request_processor.py:404-420)."""

    model_id: str
    name: str
    project: str = ""
    tags: List[str] = field(default_factory=list)
    framework: Optional[str] = None
    uri: Optional[str] = None  # local path (or file:// uri) of weights
    published: bool = False
    created: float = 0.0
