"""FastAPI inference service: REST routes + exception policy.

Route surface mirrors the reference (reference: clearml_serving/serving/
main.py:191-231): ``POST /serve/{model_id}[/{version}]`` for generic
endpoints, ``POST|GET /serve/openai/{endpoint_type}`` for OpenAI-compatible
LLM endpoints; gzip-compressed request bodies are transparently decoded
(main.py:32-50); exceptions map to 404/422/500 and a GPU OOM hard-exits the
worker so the supervisor restarts it with a clean HIP context (main.py:111-123).
"""

import gzip
import json
import os
import traceback
from typing import Optional, Union

from fastapi import FastAPI, HTTPException, Request, Response
from fastapi.routing import APIRoute, APIRouter

from .processor import EndpointNotFoundError, ModelRequestProcessor

SERVE_SUFFIX = os.environ.get("CLEARML_DEFAULT_SERVE_SUFFIX", "serve")


class GzipRequest(Request):
    async def body(self) -> bytes:
        if not hasattr(self, "_body"):
            body = await super().body()
            if "gzip" in self.headers.getlist("Content-Encoding"):
                body = gzip.decompress(body)
            self._body = body
        return self._body


class GzipRoute(APIRoute):
    def get_route_handler(self):
        original = super().get_route_handler()

        async def custom_route_handler(request: Request) -> Response:
            return await original(GzipRequest(request.scope, request.receive))

        return custom_route_handler


def _is_gpu_oom(ex: BaseException) -> bool:
    text = str(ex)
    return ("CUDA out of memory" in text or "HIP out of memory" in text
            or "hipErrorOutOfMemory" in text
            or "Memory access fault" in text)


def create_app(
    session_id: Optional[str] = None,
    store_root: Optional[str] = None,
    poll_frequency_sec: Optional[float] = None,
    processor: Optional[ModelRequestProcessor] = None,
) -> FastAPI:
    from ..store import ServingStore

    app = FastAPI(title="clearml-serving-amd inference service")
    router = APIRouter(
        prefix="/" + SERVE_SUFFIX,
        route_class=GzipRoute,
        responses={404: {"description": "Model endpoint not found"}},
    )

    if poll_frequency_sec is None:
        poll_frequency_sec = float(
            os.environ.get("CLEARML_SERVING_POLL_FREQ", 5.0)) * 60.0

    state = {"processor": processor}

    import contextlib

    @contextlib.asynccontextmanager
    async def lifespan(_app):
        await _startup()
        yield
        if state["processor"] is not None:
            state["processor"].stop()

    app.router.lifespan_context = lifespan

    async def _startup():
        if state["processor"] is None:
            store = ServingStore(store_root)
            sid = session_id or os.environ.get("CLEARML_SERVING_TASK_ID")
            state["processor"] = ModelRequestProcessor(task_id=sid, store=store)
        proc = state["processor"]
        if proc._sync_daemon_thread is None:
            proc.launch(poll_frequency_sec=poll_frequency_sec)
        # install the in-process Prometheus stats sink (statistics parity:
        # the reference ships stats through Kafka to a statistics container;
        # single-node-native is a direct in-process registry -- §5.5 SURVEY)
        try:
            from ..statistics.collector import install_stats_sink
            from ..statistics.kafka_forwarder import maybe_install

            kafka = (proc.store.get_params(proc.get_id())
                     .get("kafka_service_server")
                     or os.environ.get("CLEARML_DEFAULT_KAFKA_SERVE_URL"))
            if not maybe_install(proc, kafka):
                install_stats_sink(proc)  # in-process Prometheus (default)
        except Exception as ex:
            proc._report_text("statistics sink unavailable: {}".format(ex))
        app.state.processor = proc

    async def process_with_exceptions(
        base_url: str, version: Optional[str], request_body, serve_type: str
    ):
        processor = state["processor"]
        if processor is None:
            raise HTTPException(status_code=503, detail="service starting")
        try:
            return await processor.process_request(
                base_url=base_url, version=version or "",
                request_body=request_body, serve_type=serve_type,
            )
        except EndpointNotFoundError as ex:
            raise HTTPException(status_code=404, detail=str(ex))
        except HTTPException:
            raise
        except (ValueError, TypeError) as ex:
            if _is_gpu_oom(ex):
                _handle_gpu_oom(ex)
            raise HTTPException(
                status_code=422,
                detail="Error [{}] processing request: {}".format(type(ex), ex),
            )
        except Exception as ex:
            if _is_gpu_oom(ex):
                _handle_gpu_oom(ex)
            traceback.print_exc()
            raise HTTPException(
                status_code=500,
                detail="Error [{}] processing request: {}".format(type(ex), ex),
            )

    def _handle_gpu_oom(ex: BaseException):
        # GPU OOM leaves the HIP context poisoned: hard-exit so the process
        # supervisor restarts a clean worker (reference policy main.py:111-123)
        if os.environ.get("CLEARML_SERVING_DEV_CUDAEXCEPTION"):
            raise HTTPException(status_code=500, detail="GPU OOM: {}".format(ex))
        print("GPU out of memory, exiting worker for restart: {}".format(ex))
        os._exit(1)

    async def _read_body(request: Request) -> Union[bytes, dict]:
        body = await request.body()
        if not body:
            return {}
        content_type = (request.headers.get("content-type") or "").lower()
        if "json" in content_type or not content_type:
            try:
                return json.loads(body)
            except Exception:
                return body
        return body

    # registered BEFORE the generic /{model_id}/{version} route: two-segment
    # OpenAI paths like /serve/openai/tokenize would otherwise match the
    # generic route (model_id="openai", version="tokenize")
    @router.post("/openai/{endpoint_type:path}")
    @router.get("/openai/{endpoint_type:path}")
    async def openai_serve_model(endpoint_type: str, request: Request):
        # reference parity (main.py:207-215): OpenAI routes are JSON-only
        content_type = (request.headers.get("content-type") or "").lower()
        media_type = content_type.split(";", 1)[0].strip()
        if media_type and media_type != "application/json":
            raise HTTPException(
                status_code=415,
                detail="Unsupported Media Type: Only 'application/json' "
                       "is allowed")
        body = await _read_body(request)
        combined = dict(body) if isinstance(body, dict) else {"body": body}
        combined["request"] = request
        model = combined.get("model") or ""
        if endpoint_type.rstrip("/") == "v1/models" and not model:
            # modelless listing: enumerate LLM endpoints
            proc = state["processor"]
            eps = proc.get_synced_endpoints() if proc else {}
            return {"object": "list", "data": [
                {"id": url, "object": "model", "owned_by": "clearml-serving-amd"}
                for url, ep in eps.items() if ep.engine_type in ("llm", "vllm")
            ]}
        out = await process_with_exceptions(
            base_url=model, version=None,
            request_body=combined, serve_type=endpoint_type,
        )
        if isinstance(out, Response):
            return out
        return _jsonable(out)

    @router.post("/{model_id}/{version}")
    @router.post("/{model_id}/")
    @router.post("/{model_id}")
    async def base_serve_model(
        model_id: str, request: Request, version: Optional[str] = None
    ):
        body = await _read_body(request)
        out = await process_with_exceptions(
            base_url=model_id, version=version,
            request_body=body, serve_type="process",
        )
        if isinstance(out, Response):
            return out
        return _jsonable(out)

    app.include_router(router)

    @app.get("/health")
    async def health():
        proc = state["processor"]
        return {
            "status": "ok",
            "session": proc.get_id() if proc else None,
            "endpoints": sorted(proc.get_synced_endpoints().keys()) if proc else [],
        }

    @app.get("/status")
    async def status():
        """Routing/state overview -- the single-node equivalent of the
        reference's endpoint table + Sankey routing plot posted to the
        ClearML UI (model_request_processor.py:1141-1278)."""
        proc = state["processor"]
        if proc is None:
            return {"status": "starting"}
        endpoints = {}
        for url, ep in proc.get_synced_endpoints().items():
            entry = ep.as_dict(remove_null_entries=True)
            engine = proc._engine_processor_lookup.get(url)
            batcher = getattr(engine, "_batcher", None)
            if batcher is not None:
                s = dict(batcher.stats)
                if s.get("batches"):
                    s["mean_batch_occupancy"] = round(
                        s.pop("occupancy_sum") / s["batches"], 3)
                entry["batcher"] = s
                entry["device"] = str(batcher.device)
            llm = getattr(engine, "_engine", None)
            if llm is not None and hasattr(llm, "stats"):
                entry["llm"] = dict(llm.stats)
            endpoints[url] = entry
        routes = {
            url: route for url, route in proc._canary_route.items()
        }
        return {
            "session": proc.get_id(),
            "revision": proc._last_revision,
            "endpoint_telemetry": proc.endpoint_telemetry_snapshot(),
            "instances": proc.store.list_instances(proc.get_id(),
                                                   max_age_sec=600),
            "endpoints": endpoints,
            "canary_routes": routes,
            "monitoring": {
                k: v.as_dict(remove_null_entries=True)
                for k, v in proc.get_model_monitoring().items()
            },
            "metric_logging": {
                k: v.as_dict()
                for k, v in proc.list_endpoint_logging().items()
            },
        }

    return app


def _jsonable(out):
    """Convert numpy/torch containers to JSON-safe structures."""
    import numpy as np

    if isinstance(out, np.ndarray):
        return out.tolist()
    if isinstance(out, (np.floating, np.integer)):
        return out.item()
    if isinstance(out, dict):
        return {k: _jsonable(v) for k, v in out.items()}
    if isinstance(out, (list, tuple)):
        return [_jsonable(v) for v in out]
    try:
        import torch

        if isinstance(out, torch.Tensor):
            return out.detach().cpu().tolist()
    except Exception:
        pass
    return out


# module-level app for `uvicorn clearml_serving_amd.serving.app:app`
app = create_app()
