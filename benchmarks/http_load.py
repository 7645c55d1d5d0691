#!/usr/bin/env python3
"""HTTP-level load test: the reference's ApacheBench recipe
(`ab -l -n 8000 -c 128 -p payload.json .../serve/transformer_model`,
examples/huggingface/readme.md:141-144) as a self-contained async client.

Starts uvicorn in-process against a temp store with a BERT-base endpoint,
fires N requests at concurrency C through real HTTP, reports rps + latency
percentiles.

    python benchmarks/http_load.py [-n 8000] [-c 128]
"""

import argparse
import asyncio
import json
import multiprocessing
import os
import statistics
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PORT = int(os.environ.get("CMLS_HTTP_BENCH_PORT", 18080))


def server_proc(store_root, session_id, port, workers):
    import uvicorn

    if workers > 1:
        # multi-worker: app built per worker process from env (the
        # reference's gunicorn topology -- N workers, N engine copies).
        # uvicorn's worker bootstrap fdopens stdin, which the spawned
        # context closed -- give it /dev/null
        import sys

        devnull = os.open(os.devnull, os.O_RDONLY)
        os.dup2(devnull, 0)
        sys.stdin = os.fdopen(0)
        os.environ["CLEARML_SERVING_TASK_ID"] = session_id
        os.environ["CLEARML_SERVING_AMD_STORE"] = store_root
        os.environ["CLEARML_SERVING_POLL_FREQ"] = "60"
        uvicorn.run("clearml_serving_amd.serving.app:app", host="127.0.0.1",
                    port=port, log_level="warning", workers=workers)
    else:
        from clearml_serving_amd.serving.app import create_app

        app = create_app(session_id=session_id, store_root=store_root,
                         poll_frequency_sec=3600)
        uvicorn.run(app, host="127.0.0.1", port=port, log_level="warning")


def setup_store(tmpdir):
    import torch

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store = ServingStore(os.path.join(tmpdir, "store"))
    proc = ModelRequestProcessor(store=store, name="http-bench",
                                 force_create=True)
    card = os.path.join(tmpdir, "bert.json")
    with open(card, "wt") as f:
        json.dump({"arch": "bert-base", "num_labels": 2,
                   "dtype": "bfloat16"}, f)
    rec = store.register_model(name="bert", project="bench", path=card)
    gpu = torch.cuda.is_available()
    proc.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="transformer_model",
        model_id=rec.model_id,
        input_size=[[128], [128]], input_type=["int64", "int32"],
        input_name=["input_ids", "attention_mask"],
        auxiliary_cfg={"max_batch_size": 64, "max_queue_delay_us": 4000,
                       "use_graphs": gpu,
                       "dtype": "bfloat16" if gpu else "float32"}))
    proc.configure(default_metric_log_freq=0.1)
    proc.serialize()
    return os.path.join(tmpdir, "store"), proc.get_id()


async def load(n, c, payload, url):
    import httpx

    latencies = []
    errors = [0]
    sem = asyncio.Semaphore(c)

    async with httpx.AsyncClient(base_url=url, timeout=60.0) as client:
        # wait for readiness
        for _ in range(120):
            try:
                r = await client.get("/health")
                if r.status_code == 200:
                    break
            except Exception:
                pass
            await asyncio.sleep(1.0)

        async def one():
            async with sem:
                t0 = time.perf_counter()
                try:
                    r = await client.post("/serve/transformer_model",
                                          json=payload)
                    if r.status_code != 200:
                        errors[0] += 1
                except Exception:
                    errors[0] += 1
                latencies.append(time.perf_counter() - t0)

        # warmup (graph capture)
        await asyncio.gather(*[one() for _ in range(min(256, n))])
        latencies.clear()
        errors[0] = 0
        t0 = time.perf_counter()
        await asyncio.gather(*[one() for _ in range(n)])
        dt = time.perf_counter() - t0
    lat = sorted(latencies)
    return {
        "metric": "HTTP requests/s, BERT-base, ab-recipe (-n {} -c {})".format(n, c),
        "value": round(n / dt, 2),
        "errors": errors[0],
        "p50_ms": round(statistics.median(lat) * 1000, 2),
        "p95_ms": round(lat[int(len(lat) * 0.95)] * 1000, 2),
        "p99_ms": round(lat[int(len(lat) * 0.99)] * 1000, 2),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", type=int, default=8000)
    ap.add_argument("-c", type=int, default=128)
    ap.add_argument("--workers", type=int, default=1)
    args = ap.parse_args()

    with tempfile.TemporaryDirectory() as tmpdir:
        store_root, session_id = setup_store(tmpdir)
        # spawn: the parent touched the HIP runtime; a forked child cannot
        # re-initialize it
        ctx = multiprocessing.get_context("spawn")
        # not daemonic: uvicorn's multi-worker mode spawns child processes
        proc = ctx.Process(
            target=server_proc,
            args=(store_root, session_id, PORT, args.workers), daemon=False)
        proc.start()
        payload = {"input_ids": list(range(1, 129)),
                   "attention_mask": [1] * 128}
        try:
            result = asyncio.new_event_loop().run_until_complete(
                load(args.n, args.c, payload,
                     "http://127.0.0.1:{}".format(PORT)))
            print(json.dumps(result))
        finally:
            proc.terminate()
            proc.join(timeout=10)


if __name__ == "__main__":
    main()
