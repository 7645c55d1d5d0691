"""TP serving launcher: `torchrun --nproc-per-node N -m
clearml_serving_amd.parallel.tp_serve [--endpoint URL] [--port 8080]`.

One process per GPU over RCCL/xGMI. Rank 0 owns the serving session (HTTP
front + continuous-batching scheduler); ranks 1..N-1 run the TP worker loop,
executing rank 0's step plans (engine.run_tp_worker). The engine is built
EAGERLY on every rank from the llm endpoint's model card (lazy first-request
construction cannot work for workers -- they never see requests).
"""

import argparse
import os

import torch

from . import tp
from ..engines.llm.adapter import LlmPreprocessRequest
from ..serving.processor import ModelRequestProcessor
from ..store import ServingStore


def find_llm_endpoint(processor: ModelRequestProcessor, endpoint: str = None):
    eps = processor.get_synced_endpoints()
    for url, ep in eps.items():
        if ep.engine_type in ("llm", "vllm") and (
                endpoint is None or url == endpoint):
            return url, ep
    raise ValueError(
        "no llm endpoint found in session (have: {})".format(sorted(eps)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--endpoint", type=str, default=None,
                    help="llm endpoint url (default: first llm endpoint)")
    ap.add_argument("--port", type=int,
                    default=int(os.environ.get("CLEARML_SERVING_PORT", 8080)))
    ap.add_argument("--session-id", type=str,
                    default=os.environ.get("CLEARML_SERVING_TASK_ID"))
    ap.add_argument("--smoke", action="store_true",
                    help="serve ONE in-process chat completion and exit "
                         "(launcher self-test; no HTTP server)")
    args = ap.parse_args()

    # decode hipGraphs under TP capture the RCCL all-reduces; NCCL work
    # objects are only capturable with async error handling off (the
    # watchdog would poll events owned by the graph)
    os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "0")
    os.environ.setdefault("NCCL_ASYNC_ERROR_HANDLING", "0")

    local_rank = tp.init_from_env()
    rank, world = tp.rank(), tp.world_size()
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)

    store = ServingStore()
    processor = ModelRequestProcessor(task_id=args.session_id, store=store)
    processor.deserialize(skip_sync=True)
    url, ep = find_llm_endpoint(processor, args.endpoint)
    print("[tp_serve rank {}/{}] llm endpoint '{}' on cuda:{}".format(
        rank, world, url, local_rank))

    # build the engine eagerly on every rank (identical config)
    adapter = LlmPreprocessRequest(model_endpoint=ep, task=store)
    engine = adapter._engine

    if rank == 0:
        processor._engine_processor_lookup[url] = adapter
        if args.smoke:
            import asyncio

            async def one():
                return await processor.process_request(
                    base_url=url, version=None, serve_type="v1/chat/completions",
                    request_body={"messages": [{"role": "user",
                                                "content": "smoke"}],
                                  "max_tokens": 4, "temperature": 0.0,
                                  "ignore_eos": True})
            out = asyncio.new_event_loop().run_until_complete(one())
            assert out["usage"]["completion_tokens"] == 4, out
            engine.tp_shutdown()
            print("TP-SERVE-OK", flush=True)
            return
        processor.launch(poll_frequency_sec=300)
        import uvicorn

        from ..serving.app import create_app

        app = create_app(processor=processor)
        try:
            uvicorn.run(app, host="0.0.0.0", port=args.port)
        finally:
            engine.tp_shutdown()
    else:
        engine.run_tp_worker()


if __name__ == "__main__":
    main()
