"""Engine registry + per-endpoint request processors.

Mirrors the behavior of the reference engine tier (reference: clearml_serving/
serving/preprocess_service.py): a ``BasePreprocessRequest`` registry keyed by
engine-type string, user ``Preprocess`` code hot-loaded from a sha256-verified
artifact, per-phase sync/async dispatch flags, and a ``send_request`` hook
injected for model pipelines/ensembles.

Engine map (reference engine -> here):
- "sklearn"/"xgboost"/"lightgbm"  in-process CPU predict (preprocess_service.py:449-501)
- "custom"/"custom_async"         user-managed model    (preprocess_service.py:504-616)
- "triton"                        -> in-process HIP engine ("hip" alias kept for
                                  CLI parity; no gRPC hop, no sidecar:
                                  the reference marshals numpy -> protobuf ->
                                  tritonserver, preprocess_service.py:267-446;
                                  here process() enqueues straight into the
                                  dynamic batcher on the local GPU)
- "vllm"                          -> native LLM engine ("llm" alias)
"""

import asyncio
import importlib.util
import os
import sys
import zipfile
from hashlib import sha256
from typing import Dict, Optional

import numpy as np

from ..schemas import ModelEndpoint


class Preprocess:
    """Default no-op user-code contract (reference:
    clearml_serving/preprocess/preprocess_template.py:6-168)."""

    def __init__(self):
        self.model_endpoint: Optional[ModelEndpoint] = None

    def preprocess(self, body, state, collect_custom_statistics_fn=None):
        return body

    def postprocess(self, data, state, collect_custom_statistics_fn=None):
        return data


class BasePreprocessRequest:
    __preprocessing_lookup: Dict[str, type] = {}
    __preprocessing_modules = set()

    is_preprocess_async = False
    is_process_async = False
    is_postprocess_async = False
    # sync process() that may BLOCK (user code, threaded pipeline fan-out):
    # the dispatcher runs it in a worker thread so the event loop keeps
    # serving (the sync send_request path depends on a live loop)
    run_process_off_loop = False

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        """``task`` kept for signature parity; here it is the ServingStore."""
        self.model_endpoint = model_endpoint
        self._store = task
        self._preprocess = None
        self._model = None
        self._artifact_sha = None
        if self.model_endpoint.preprocess_artifact:
            try:
                self._instantiate_custom_preprocess_cls()
            except Exception as ex:
                raise ValueError(
                    "Error: Failed loading preprocess code for '{}': {}".format(
                        self.model_endpoint.preprocess_artifact, ex
                    )
                )

    def _instantiate_custom_preprocess_cls(self) -> None:
        """Load the user's Preprocess class from the stored artifact.

        The artifact is content-addressed; we re-verify sha256 before import
        (reference re-downloads on hash mismatch, preprocess_service.py:68-82).
        """
        artifact = self._store.get_artifact(
            self._session_id(), self.model_endpoint.preprocess_artifact
        ) if self._store else None
        if not artifact:
            raise ValueError(
                "preprocess artifact '{}' not found".format(
                    self.model_endpoint.preprocess_artifact
                )
            )
        path = artifact["path"]
        with open(path, "rb") as f:
            if sha256(f.read()).hexdigest() != artifact["sha256"]:
                raise ValueError("artifact hash mismatch for {}".format(path))
        self._artifact_sha = artifact["sha256"]  # staleness check on reload

        if path.endswith(".zip"):
            # package artifact: unpack beside the archive, import its
            # preprocess.py (reference handles package artifacts the same way)
            target = path[:-4] + "_unpacked"
            if not os.path.isdir(target):
                with zipfile.ZipFile(path) as z:
                    z.extractall(target)
            path = os.path.join(target, "preprocess.py")

        spec = importlib.util.spec_from_file_location(
            "_cmls_preprocess_" + artifact["sha256"][:16], path
        )
        module = importlib.util.module_from_spec(spec)
        sys.modules[spec.name] = module
        spec.loader.exec_module(module)
        cls = getattr(module, "Preprocess", None)
        if cls is None:
            raise ValueError("no class named 'Preprocess' in {}".format(path))
        self._preprocess = cls()
        self._preprocess.model_endpoint = self.model_endpoint
        if callable(getattr(self._preprocess, "load", None)):
            local = self._get_local_model_file()
            self._model = self._preprocess.load(local)

    def _session_id(self) -> Optional[str]:
        return getattr(self._store, "_session_id", None) or getattr(
            self._store, "session_id", None
        )

    def _get_local_model_file(self) -> Optional[str]:
        if not self._store or not self.model_endpoint.model_id:
            return None
        return self._store.get_model_local_path(self.model_endpoint.model_id)

    def __del__(self):
        try:
            if self._preprocess is not None and callable(
                getattr(self._preprocess, "unload", None)
            ):
                self._preprocess.unload()
        except Exception:
            pass

    # ------------------------------------------------------------------ #
    # phase dispatch
    # ------------------------------------------------------------------ #
    def preprocess(self, request, state: dict, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "preprocess"):
            return self._preprocess.preprocess(
                request, state, collect_custom_statistics_fn
            )
        return request

    def postprocess(self, data, state: dict, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "postprocess"):
            return self._preprocess.postprocess(
                data, state, collect_custom_statistics_fn
            )
        return data

    def process(self, data, state: dict, collect_custom_statistics_fn=None):
        raise NotImplementedError

    # ------------------------------------------------------------------ #
    # registry
    # ------------------------------------------------------------------ #
    @classmethod
    def register_engine(cls, engine_name: str, modules: Optional[list] = None):
        def register_f(engine_cls):
            engine_cls._engine_type = engine_name
            cls.__preprocessing_lookup[engine_name] = engine_cls
            if modules:
                cls.__preprocessing_modules |= set(modules)
            return engine_cls

        return register_f

    @classmethod
    def load_modules(cls) -> None:
        """Pre-import heavyweight engine deps before forking workers
        (reference: preprocess_service.py:245-253)."""
        for m in cls.__preprocessing_modules:
            try:
                importlib.import_module(m)
            except Exception:
                pass

    # multi-process front mode: GPU engine types resolve to an SHM proxy
    # that ships tensors to the engine-owner process (serving/front.py)
    _engine_overrides: Dict[str, type] = {}

    @classmethod
    def override_engine(cls, engine_name: str, engine_cls: Optional[type]):
        if engine_cls is None:
            cls._engine_overrides.pop(engine_name, None)
        else:
            cls._engine_overrides[engine_name] = engine_cls

    @classmethod
    def get_engine_cls(cls, engine: str) -> type:
        if engine in cls._engine_overrides:
            return cls._engine_overrides[engine]
        if engine not in cls.__preprocessing_lookup:
            raise ValueError("Engine '{}' not recognized".format(engine))
        return cls.__preprocessing_lookup[engine]

    @classmethod
    def validate_engine_type(cls, engine: str) -> bool:
        return engine in cls.__preprocessing_lookup

    @classmethod
    def list_engines(cls):
        return sorted(cls.__preprocessing_lookup.keys())

    # injected helper for pipelines/ensembles -- replaced at processor init
    # (reference: preprocess_service.py:113, 255-264)
    _default_serving_base_url = os.environ.get(
        "CLEARML_DEFAULT_BASE_SERVE_URL", "http://127.0.0.1:8080/serve"
    )

    @staticmethod
    def _preprocess_send_request(
        self, endpoint: str, version: Optional[str] = None, data: Optional[dict] = None
    ) -> Optional[dict]:
        import requests as _requests

        endpoint = "{}/{}".format(endpoint.strip("/"), version.strip("/")) \
            if version else endpoint.strip("/")
        base_url = BasePreprocessRequest._default_serving_base_url
        url = "{}/{}".format(base_url.rstrip("/"), endpoint)
        try:
            r = _requests.post(url, json=data)
            r.raise_for_status()
            return r.json()
        except Exception:
            return None


@BasePreprocessRequest.register_engine("sklearn", modules=["joblib", "sklearn"])
class SKLearnPreprocessRequest(BasePreprocessRequest):
    """joblib-pickled estimator served in-process
    (reference: preprocess_service.py:449-464)."""

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        super().__init__(model_endpoint, task)
        if self._model is None:
            import joblib

            local = self._get_local_model_file()
            if not local:
                raise ValueError(
                    "sklearn endpoint '{}' has no model file".format(
                        model_endpoint.serving_url
                    )
                )
            self._model = joblib.load(local)

    def process(self, data, state: dict, collect_custom_statistics_fn=None):
        return self._model.predict(np.atleast_2d(data))


@BasePreprocessRequest.register_engine("xgboost", modules=["xgboost"])
class XGBoostPreprocessRequest(BasePreprocessRequest):
    """xgboost Booster served in-process (reference: preprocess_service.py:467-483).

    xgboost is not installed in this image; the engine registers (so configs
    validate) and raises a clear error at first use.
    """

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        super().__init__(model_endpoint, task)
        if self._model is None:
            import xgboost  # noqa -- raises ModuleNotFoundError when absent

            self._model = xgboost.Booster()
            self._model.load_model(self._get_local_model_file())

    def process(self, data, state: dict, collect_custom_statistics_fn=None):
        import xgboost

        return self._model.predict(xgboost.DMatrix(np.atleast_2d(data)))


@BasePreprocessRequest.register_engine("lightgbm", modules=["lightgbm"])
class LightGBMPreprocessRequest(BasePreprocessRequest):
    """lightgbm Booster served in-process (reference: preprocess_service.py:486-501);
    registers even when lightgbm is absent, fails at first use."""

    def __init__(self, model_endpoint: ModelEndpoint, task=None):
        super().__init__(model_endpoint, task)
        if self._model is None:
            import lightgbm  # noqa

            self._model = lightgbm.Booster(model_file=self._get_local_model_file())

    def process(self, data, state: dict, collect_custom_statistics_fn=None):
        return self._model.predict(np.atleast_2d(data))


@BasePreprocessRequest.register_engine("custom")
class CustomPreprocessRequest(BasePreprocessRequest):
    """Fully user-managed model: Preprocess.process() runs synchronously
    (reference: preprocess_service.py:504-517) -- in a worker thread, so user
    code may block (e.g. the reference's ThreadPoolExecutor pipeline,
    examples/pipeline/preprocess.py:18-32, whose sync send_request needs the
    serving loop to stay live)."""

    run_process_off_loop = True

    def process(self, data, state: dict, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "process"):
            return self._preprocess.process(data, state, collect_custom_statistics_fn)
        return None


@BasePreprocessRequest.register_engine("custom_async")
class CustomAsyncPreprocessRequest(BasePreprocessRequest):
    """Fully-async custom engine for pipelines/ensembles
    (reference: preprocess_service.py:520-616)."""

    is_preprocess_async = True
    is_process_async = True
    is_postprocess_async = True

    async def preprocess(self, request, state, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "preprocess"):
            fn = self._preprocess.preprocess
            if asyncio.iscoroutinefunction(fn):
                return await fn(request, state, collect_custom_statistics_fn)
            return fn(request, state, collect_custom_statistics_fn)
        return request

    async def postprocess(self, data, state, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "postprocess"):
            fn = self._preprocess.postprocess
            if asyncio.iscoroutinefunction(fn):
                return await fn(data, state, collect_custom_statistics_fn)
            return fn(data, state, collect_custom_statistics_fn)
        return data

    async def process(self, data, state, collect_custom_statistics_fn=None):
        if self._preprocess is not None and hasattr(self._preprocess, "process"):
            fn = self._preprocess.process
            if asyncio.iscoroutinefunction(fn):
                return await fn(data, state, collect_custom_statistics_fn)
            return fn(data, state, collect_custom_statistics_fn)
        return None

    @staticmethod
    async def _preprocess_send_request(
        self, endpoint: str, version: Optional[str] = None, data: Optional[dict] = None
    ) -> Optional[dict]:
        # async pipeline hop (reference: preprocess_service.py:606-616); the
        # processor rebinds this to an intra-process dispatch at launch.
        return await asyncio.to_thread(
            BasePreprocessRequest._preprocess_send_request,
            self, endpoint, version, data,
        )
