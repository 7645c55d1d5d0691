"""User preprocess contract -- the full duck-typed surface.

Copy this file, keep the class named ``Preprocess``, and register it with
an endpoint via ``model add --preprocess your_file.py``. The class is
hot-loaded from a sha256-verified artifact at endpoint creation, so new
code deploys WITHOUT restarting the serving containers (reference
contract: clearml_serving/preprocess/preprocess_template.py:6-168).

Every method is optional -- implement only what the endpoint needs. All
methods receive:
  * ``state``  -- a per-REQUEST dict. Use it to pass values between
    preprocess/process/postprocess of the same request; it is never
    shared across requests.
  * ``collect_custom_statistics_fn`` -- when metric logging sampled this
    request, a callable taking {name: value}; reported values become
    Prometheus series named "{endpoint}:{name}" (see `metrics add`).
    None when the request was not sampled.
"""

from typing import Any, Callable, Optional


class Preprocess(object):
    """Name must be exactly ``Preprocess``."""

    def __init__(self):
        # Called ONCE when the endpoint instance is created (not per
        # request). self.model_endpoint is injected right after -- the
        # full ModelEndpoint schema of this endpoint.
        self.model_endpoint = None

    def load(self, local_file_name: str) -> Any:
        """Optional. Called once with the endpoint model's local path.

        Whatever this returns is the endpoint's model object:
        * for ``custom``/``custom_async`` engines, YOU own it -- use it in
          process().
        * for library engines (sklearn/xgboost/lightgbm/hip/llm), return
          None to let the engine load the file itself, or return a
          loaded model/config to override its loading.
        """
        return None

    def unload(self) -> None:
        """Optional. Called when the endpoint instance is torn down (e.g.
        superseded by a config reload) -- free external resources here."""
        pass

    def preprocess(
        self,
        body: Any,
        state: dict,
        collect_custom_statistics_fn: Optional[Callable[[dict], None]] = None,
    ) -> Any:
        """Request body (decoded JSON or raw bytes) -> model input."""
        return body

    def process(
        self,
        data: Any,
        state: dict,
        collect_custom_statistics_fn: Optional[Callable[[dict], None]] = None,
    ) -> Any:
        """ONLY used by the ``custom`` / ``custom_async`` engines: run the
        model you loaded in load(). Library engines run their own predict
        and ignore this method.

        Model PIPELINES fan out from here with ``self.send_request`` (it
        is injected by the serving process and dispatches IN PROCESS --
        no HTTP hop):

        sync engines (``custom``) -- call it from a worker thread, e.g.
        a ThreadPoolExecutor (the serving event loop must stay free;
        calling it inline on the loop raises):

            from concurrent.futures import ThreadPoolExecutor
            with ThreadPoolExecutor(max_workers=8) as ex:
                futures = [ex.submit(self.send_request, "other_ep", None,
                                     {"x": v}) for v in data]
                results = [f.result() for f in futures]

        async engines (``custom_async``) -- await it directly:

            a, b = await asyncio.gather(
                self.send_request("ep_a", None, data),
                self.send_request("ep_b", None, data))

        A failed fan-out hop returns None (and logs loudly) -- check for
        it.
        """
        return None

    def postprocess(
        self,
        data: Any,
        state: dict,
        collect_custom_statistics_fn: Optional[Callable[[dict], None]] = None,
    ) -> Any:
        """Model output -> JSON-serializable response body."""
        return data

    def send_request(self, endpoint: str, version: Optional[str] = None,
                     data: Optional[dict] = None) -> Optional[dict]:
        """Placeholder -- REPLACED at runtime by the serving process with
        an in-process dispatcher (see process() docstring)."""
        raise RuntimeError("send_request is injected by the serving "
                           "process at endpoint creation")
