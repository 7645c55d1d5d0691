"""Engine-owner process: the single owner of one GPU's engines.

Receives preprocessed tensor requests from the HTTP front workers over
shared-memory rings (shm_transport), coalesces them through the per-endpoint
dynamic batcher, and ships the outputs back. One owner per GPU keeps every
dynamic batch whole -- the measured fix for the reference's
N-workers-N-model-copies topology (profiles/README.md §5; reference
entrypoint.sh:56-72 is the topology being replaced).

Run:
    python -m clearml_serving_amd.serving.engine_owner \
        --store DIR --session ID --prefix cmls_ab12 --owner 0 --workers 8
"""

import argparse
import asyncio
import os
import sys
import time
import traceback
from typing import Any, Dict

GPU_ENGINE_TYPES = ("hip", "triton", "pytorch", "llm", "vllm")


class _OwnerTelemetryShim:
    """Processor-shaped view over the owner's engine map for the
    Prometheus collectors (live dict reference; the owner mutates it)."""

    def __init__(self, engines):
        self._engine_processor_lookup = engines

    def set_stats_sink(self, sink):
        pass  # the owner emits no per-request stat dicts

    def list_endpoint_logging(self):
        return {}


class EngineOwner:
    def __init__(self, store_root: str, session_id: str, prefix: str,
                 owner_idx: int, n_workers: int, device: int = 0,
                 ring_bytes: int = 32 << 20,
                 poll_frequency_sec: float = 10.0):
        from ..store import ServingStore
        from .processor import ModelRequestProcessor
        from .shm_transport import make_ring

        self.owner_idx = owner_idx
        self.device = device
        store = ServingStore(store_root)
        self.proc = ModelRequestProcessor(task_id=session_id, store=store)
        self.proc.launch(poll_frequency_sec=poll_frequency_sec)
        # owner creates its rings; fronts attach (retrying) after
        self.req_rings = [
            make_ring("{}_req_{}_{}".format(prefix, owner_idx, w),
                      ring_bytes, True)
            for w in range(n_workers)
        ]
        self.resp_rings = [
            make_ring("{}_resp_{}_{}".format(prefix, owner_idx, w),
                      ring_bytes, True)
            for w in range(n_workers)
        ]
        self._engines: Dict[str, Any] = {}
        self._engine_eps: Dict[str, dict] = {}
        self._inflight: Dict[int, "asyncio.Task"] = {}
        self._stop = False
        self.stats = {"requests": 0, "errors": 0, "aborts": 0}
        # the owner has no HTTP server, so its batcher/LLM/GPU telemetry
        # exports straight to Prometheus (fronts take STATS_PORT+worker;
        # owners take STATS_PORT+100+owner so scrape configs can list
        # both). A shim over the owner's OWN engine map feeds the
        # per-stage collector -- inserting engines into the processor's
        # lookup would let its hot-reload flush tear them down mid-request
        # (the owner manages engine lifecycle itself in _get_engine).
        try:
            from ..statistics.collector import install_stats_sink

            base = int(os.environ.get("CLEARML_SERVING_STATS_PORT", 9999))
            if base > 0:
                install_stats_sink(_OwnerTelemetryShim(self._engines),
                                   port=base + 100 + owner_idx)
        except Exception as ex:
            print("[engine-owner] stats export unavailable: {}".format(ex))

    # ------------------------------------------------------------------ #
    def _get_engine(self, url: str):
        """Engine instance for a normalized endpoint url; rebuilt when the
        endpoint's config changed on a hot reload."""
        from .preprocess import BasePreprocessRequest

        ep = self.proc.get_synced_endpoints().get(url)
        if ep is None:
            raise LookupError(
                "endpoint '{}' not found on engine owner".format(url))
        ep_dict = ep.as_dict()
        cached = self._engines.get(url)
        if cached is not None and self._engine_eps.get(url) == ep_dict:
            return cached
        if cached is not None:
            shutdown = getattr(cached, "shutdown", None)
            if shutdown:
                try:
                    shutdown()
                except Exception:
                    traceback.print_exc()
            self._engines.pop(url, None)
        if ep.engine_type not in GPU_ENGINE_TYPES:
            raise ValueError(
                "engine owner only serves GPU engine types, got '{}'".format(
                    ep.engine_type))
        # this owner's GPU wins over placement heuristics
        import copy

        ep = copy.deepcopy(ep)
        aux = dict(ep.auxiliary_cfg or {})
        try:
            import torch

            if torch.cuda.is_available():
                if ep.engine_type in ("llm", "vllm"):
                    aux.setdefault("device", "cuda:{}".format(self.device))
                else:
                    aux.setdefault("gpu", self.device)
        except ImportError:
            pass
        ep.auxiliary_cfg = aux
        cls = BasePreprocessRequest.get_engine_cls(ep.engine_type)
        engine = cls(model_endpoint=ep, task=self.proc.store)
        self._engines[url] = engine
        self._engine_eps[url] = ep_dict
        return engine

    async def _push(self, worker: int, record: bytes) -> None:
        ring = self.resp_rings[worker]
        while not ring.push(record):
            await asyncio.sleep(0.001)

    async def _relay_stream(self, req_id: int, worker: int, response) -> None:
        """Relay a StreamingResponse body chunk-by-chunk over the response
        ring (FIFO per worker, so chunk order is preserved); the front's
        ShmClient.infer_stream reassembles it into an SSE response."""
        from .shm_transport import (pack_response, pack_stream_chunk,
                                    pack_stream_end)

        try:
            async for chunk in response.body_iterator:
                if isinstance(chunk, str):
                    chunk = chunk.encode()
                if chunk:
                    await self._push(worker, pack_stream_chunk(req_id, chunk))
            await self._push(worker, pack_stream_end(req_id))
            self.stats["requests"] += 1
        except asyncio.CancelledError:
            # abort record: close the generator explicitly -- the cancel may
            # land while suspended in _push, which unwinds the async-for
            # WITHOUT finalizing the generator, so the engine's
            # abort-on-disconnect (generate()'s finally) would otherwise
            # wait for GC while generation runs to max_tokens
            aclose = getattr(response.body_iterator, "aclose", None)
            if aclose is not None:
                try:
                    await aclose()
                except Exception:
                    pass
            raise
        except Exception as ex:
            traceback.print_exc()
            self.stats["errors"] += 1
            await self._push(worker, pack_response(
                req_id, error="{}: {}".format(type(ex).__name__, ex)))

    async def _handle(self, raw: bytes, worker: int) -> None:
        from .shm_transport import pack_response, unpack_request

        req_id = None
        try:
            req_id, url, data = unpack_request(raw)
            if url is None:  # abort record: cancel the in-flight request
                task = self._inflight.get((worker, req_id))
                if task is not None and not task.done():
                    task.cancel()
                    self.stats["aborts"] += 1
                return
            serve_type = "process"
            if isinstance(data, dict) and "__serve_type__" in data:
                serve_type = data["__serve_type__"]
                data = data.get("__body__")
            engine = self._get_engine(url)
            if serve_type == "process":
                if engine.is_process_async:
                    out = await engine.process(data, {})
                else:
                    out = engine.process(data, {})
            else:
                method = getattr(engine, serve_type.replace("/", "_"), None)
                if method is None:
                    raise ValueError(
                        "engine for '{}' does not implement '{}'".format(
                            url, serve_type))
                out = await method(data, {})
            if hasattr(out, "body_iterator"):  # StreamingResponse (SSE)
                await self._relay_stream(req_id, worker, out)
                return
            resp = pack_response(req_id, out)
            self.stats["requests"] += 1
        except asyncio.CancelledError:
            return  # aborted by the client: nothing to send back
        except Exception as ex:
            traceback.print_exc()
            self.stats["errors"] += 1
            if req_id is None:
                return
            resp = pack_response(req_id, error="{}: {}".format(
                type(ex).__name__, ex))
        await self._push(worker, resp)

    async def serve(self) -> None:
        idle_sleep = 0.0002
        while not self._stop:
            got = False
            for w, ring in enumerate(self.req_rings):
                for raw in ring.drain(512):
                    got = True
                    import struct as _struct

                    rid, ulen = _struct.unpack_from("<QH", raw, 0)
                    task = asyncio.ensure_future(self._handle(raw, w))
                    # abort records carry the SAME req_id as the request
                    # they cancel: registering them would clobber (and
                    # self-cancel) the generation task's _inflight entry
                    is_abort = (ulen == 0 and len(raw) > 10
                                and raw[10] == 2)  # KIND_ABORT
                    if not is_abort:
                        # keyed by (worker, id): every front worker counts
                        # req_ids from 1, so bare ids collide across workers
                        # and an abort from one front could cancel another
                        # front's request
                        self._inflight[(w, rid)] = task
                        task.add_done_callback(
                            lambda t, k=(w, rid):
                            self._inflight.pop(k, None))
            if got:
                await asyncio.sleep(0)
            else:
                await asyncio.sleep(idle_sleep)

    def stop(self) -> None:
        self._stop = True

    def close(self) -> None:
        self.proc.stop()
        for r in self.req_rings + self.resp_rings:
            r.close()


def main(argv=None) -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--store", required=True)
    ap.add_argument("--session", required=True)
    ap.add_argument("--prefix", required=True)
    ap.add_argument("--owner", type=int, default=0)
    ap.add_argument("--workers", type=int, required=True)
    ap.add_argument("--device", type=int, default=None)
    ap.add_argument("--ring-mb", type=int, default=32)
    ap.add_argument("--poll-freq-sec", type=float, default=10.0)
    args = ap.parse_args(argv)

    device = args.device if args.device is not None else args.owner
    owner = EngineOwner(
        store_root=args.store, session_id=args.session, prefix=args.prefix,
        owner_idx=args.owner, n_workers=args.workers, device=device,
        ring_bytes=args.ring_mb << 20,
        poll_frequency_sec=args.poll_freq_sec)
    print("[engine-owner {}] serving on device {} for {} workers".format(
        args.owner, device, args.workers), flush=True)
    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(owner.serve())
    except KeyboardInterrupt:
        pass
    finally:
        owner.close()


if __name__ == "__main__":
    main()
