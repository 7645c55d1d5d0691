"""HTTP front worker for the multi-process serving topology.

Each front worker is a full serving process (routes, canary, preprocess,
statistics, CPU engines) EXCEPT that GPU engine types resolve to an SHM
proxy: after user preprocess, tensors ship over a shared-memory ring to the
engine-owner process that exclusively owns the GPU, so dynamic batches stay
whole no matter how many HTTP workers parse connections. Front workers bind
the same port with SO_REUSEPORT; the kernel load-balances connections.

This is the MI355X-native answer to the reference's two topologies, both of
which lose: N gunicorn workers with N model copies (entrypoint.sh:56-72,
splits batches) and 1 worker (connection handling caps at ~600 req/s).

Run (one per worker, usually via serving.launch):
    python -m clearml_serving_amd.serving.front --store DIR --session ID \
        --prefix cmls_ab12 --worker 0 --owners 1 --port 8080
"""

import argparse
import asyncio
import os
import socket
import zlib
from typing import Any, Optional

from .preprocess import BasePreprocessRequest
from .shm_transport import ShmClient

GPU_ENGINE_TYPES = ("hip", "triton", "pytorch")
LLM_ENGINE_TYPES = ("llm", "vllm")


def _endpoint_url(ep) -> str:
    # same normalization as ModelRequestProcessor._normalize_endpoint_url
    return "{}/{}".format(ep.serving_url.rstrip("/"),
                          ep.version or "").rstrip("/")


def _placement(ep, n_owners: int) -> int:
    """Deterministic endpoint -> owner routing, identical in every front:
    explicit auxiliary_cfg 'gpu' wins, else a stable hash of the url."""
    aux = ep.auxiliary_cfg or {}
    if "gpu" in aux:
        return int(aux["gpu"]) % n_owners
    if "device" in aux and str(aux["device"]).startswith("cuda:"):
        return int(str(aux["device"]).split(":")[1]) % n_owners
    return zlib.crc32(_endpoint_url(ep).encode()) % n_owners


class ShmProxyRequest(BasePreprocessRequest):
    """Front-side stand-in for a GPU engine: preprocess/postprocess run
    locally (user code), process() ships tensors to the engine owner."""

    is_preprocess_async = False
    is_process_async = True
    is_postprocess_async = False

    client: Optional[ShmClient] = None  # set by setup_front()
    n_owners: int = 1

    def __init__(self, model_endpoint, task=None):
        super().__init__(model_endpoint, task)
        self._url = _endpoint_url(model_endpoint)
        self._owner = _placement(model_endpoint, self.n_owners)

    async def process(self, data: Any, state: dict,
                      collect_custom_statistics_fn=None) -> Any:
        return await self.client.infer(self._url, data, owner=self._owner)


class ShmLlmProxyRequest(ShmProxyRequest):
    """LLM endpoints over the SHM transport: whole-request dispatch to the
    owner's engine; SSE streams relay chunk-by-chunk over the response
    ring."""

    is_preprocess_async = True
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

    @staticmethod
    def _clean_body(body: Any) -> Any:
        if isinstance(body, dict):
            # the route injects the raw starlette Request (unpicklable)
            return {k: v for k, v in body.items() if k != "request"}
        return body

    async def _ship(self, serve_type: str, body: Any) -> Any:
        body = self._clean_body(body)
        payload = {"__serve_type__": serve_type, "__body__": body}
        if isinstance(body, dict) and body.get("stream"):
            # SSE: the owner relays the engine's stream chunk-by-chunk over
            # the response ring; re-expose it as a StreamingResponse here
            from fastapi.responses import StreamingResponse

            async def gen():
                async for chunk in self.client.infer_stream(
                        self._url, payload, owner=self._owner):
                    yield chunk

            return StreamingResponse(gen(), media_type="text/event-stream")
        return await self.client.infer(self._url, payload, owner=self._owner)

    async def process(self, data, state, collect_custom_statistics_fn=None):
        return await self._ship("process", data)

    def __getattr__(self, name):
        # v1_chat_completions / v1_completions / v1_embeddings / ... all
        # ship the same way; generated on demand
        if name.startswith("v1_") or name in (
                "pooling", "tokenize", "detokenize", "v2_rerank",
                "classify", "score", "rerank", "version"):
            async def _method(body, state, collect_fn=None, _st=name):
                # shipped as the method name; the owner resolves it the same
                # way the processor does (serve_type.replace('/', '_'))
                return await self._ship(_st, body)
            return _method
        raise AttributeError(name)


def setup_front(prefix: str, worker_id: int, n_owners: int,
                ring_bytes: int = 32 << 20, attach_retries: int = 100):
    """Install the SHM proxy overrides + client in this process."""
    last = None
    for _ in range(attach_retries):
        try:
            client = ShmClient(prefix, worker_id, n_owners,
                               ring_bytes=ring_bytes)
            break
        except Exception as ex:  # owner hasn't created the rings yet
            last = ex
            import time

            time.sleep(0.2)
    else:
        raise RuntimeError("could not attach SHM rings: {}".format(last))
    ShmProxyRequest.client = client
    ShmProxyRequest.n_owners = n_owners
    for name in GPU_ENGINE_TYPES:
        BasePreprocessRequest.override_engine(name, ShmProxyRequest)
    for name in LLM_ENGINE_TYPES:
        BasePreprocessRequest.override_engine(name, ShmLlmProxyRequest)
    return client


def teardown_front():
    for name in GPU_ENGINE_TYPES + LLM_ENGINE_TYPES:
        BasePreprocessRequest.override_engine(name, None)
    if ShmProxyRequest.client is not None:
        ShmProxyRequest.client.close()
        ShmProxyRequest.client = None


def _reuseport_socket(host: str, port: int) -> socket.socket:
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEPORT, 1)
    sock.bind((host, port))
    sock.listen(2048)
    sock.setblocking(False)
    return sock


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--store", required=True)
    ap.add_argument("--session", required=True)
    ap.add_argument("--prefix", required=True)
    ap.add_argument("--worker", type=int, required=True)
    ap.add_argument("--owners", type=int, default=1)
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=8080)
    ap.add_argument("--ring-mb", type=int, default=32)
    ap.add_argument("--poll-freq-sec", type=float, default=10.0)
    args = ap.parse_args(argv)

    setup_front(args.prefix, args.worker, args.owners,
                ring_bytes=args.ring_mb << 20)

    # per-front Prometheus export: worker w serves metrics on
    # STATS_PORT + w (a single shared port would only expose one front's
    # sample; Kafka mode aggregates across processes like the reference)
    base = int(os.environ.get("CLEARML_SERVING_STATS_PORT", 9999))
    if base > 0:
        os.environ["CLEARML_SERVING_STATS_PORT"] = str(base + args.worker)

    import uvicorn

    from .app import create_app

    app = create_app(session_id=args.session, store_root=args.store,
                     poll_frequency_sec=args.poll_freq_sec)
    sock = _reuseport_socket(args.host, args.port)
    print("[front {}] listening on {}:{} (SO_REUSEPORT)".format(
        args.worker, args.host, args.port), flush=True)
    config = uvicorn.Config(app, log_level="warning", access_log=False)
    server = uvicorn.Server(config)
    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(server.serve(sockets=[sock]))
    finally:
        teardown_front()


if __name__ == "__main__":
    main()
