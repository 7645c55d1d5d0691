#!/usr/bin/env python3
"""HTTP-level load test: the reference's ApacheBench recipe
(`ab -l -n 8000 -c 128 -p payload.json .../serve/transformer_model`,
examples/huggingface/readme.md:141-144) as a self-contained multi-process
load client against the serving stack.

Topologies under test (serving/launch.py):
  --workers 0   single process owns HTTP + GPU (round-1 topology)
  --workers N   N SO_REUSEPORT HTTP fronts + 1 engine owner per GPU,
                shared-memory tensor handoff (batches stay whole)

The client is multi-process aiohttp (a single Python event loop cannot
saturate a multi-process server); per ab semantics connections are NOT
reused unless --keepalive is given.

    python benchmarks/http_load.py [-n 8000] [-c 128] [--workers 8]
        [--client-procs 4] [--keepalive] [--out FILE]
"""

import argparse
import asyncio
import json
import multiprocessing
import os
import statistics
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PORT = int(os.environ.get("CMLS_HTTP_BENCH_PORT", 18080))
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def setup_store(tmpdir):
    import torch

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store = ServingStore(os.path.join(tmpdir, "store"))
    proc = ModelRequestProcessor(store=store, name="http-bench",
                                 force_create=True)
    card = os.path.join(tmpdir, "bert.json")
    gpu = torch.cuda.is_available()
    with open(card, "wt") as f:
        json.dump({"arch": "bert-base", "num_labels": 2,
                   "dtype": "bfloat16" if gpu else "float32"}, f)
    rec = store.register_model(name="bert", project="bench", path=card)
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


def client_proc(proc_idx, n, c, payload, url, keepalive, warmup, conn):
    """One load-client process: n requests at concurrency c (aiohttp)."""

    async def run():
        import aiohttp

        connector = aiohttp.TCPConnector(limit=c * 2,
                                         force_close=not keepalive)
        timeout = aiohttp.ClientTimeout(total=120)
        latencies = []
        errors = [0]
        sem = asyncio.Semaphore(c)
        body = json.dumps(payload).encode()
        headers = {"Content-Type": "application/json"}

        async with aiohttp.ClientSession(connector=connector,
                                         timeout=timeout) as session:
            async def one():
                async with sem:
                    t0 = time.perf_counter()
                    try:
                        async with session.post(url, data=body,
                                                headers=headers) as r:
                            await r.read()
                            if r.status != 200:
                                errors[0] += 1
                    except Exception:
                        errors[0] += 1
                    latencies.append(time.perf_counter() - t0)

            await asyncio.gather(*[one() for _ in range(warmup)])
            latencies.clear()
            errors[0] = 0
            t0 = time.perf_counter()
            await asyncio.gather(*[one() for _ in range(n)])
            dt = time.perf_counter() - t0
        return {"n": n, "dt": dt, "errors": errors[0],
                "latencies": latencies}

    res = asyncio.new_event_loop().run_until_complete(run())
    conn.send(res)
    conn.close()


def wait_healthy(url, tries=240):
    import urllib.request

    for _ in range(tries):
        try:
            with urllib.request.urlopen(url + "/health", timeout=2) as r:
                if r.status == 200:
                    return True
        except Exception:
            pass
        time.sleep(0.5)
    return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-n", type=int, default=8000)
    ap.add_argument("-c", type=int, default=128)
    ap.add_argument("--workers", type=int, default=0,
                    help="HTTP front workers (0 = single-process topology)")
    ap.add_argument("--client-procs", type=int, default=4)
    ap.add_argument("--keepalive", action="store_true",
                    help="reuse connections (ab default is one per request)")
    ap.add_argument("--warmup", type=int, default=256)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()

    with tempfile.TemporaryDirectory() as tmpdir:
        store_root, session_id = setup_store(tmpdir)
        env = dict(os.environ)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        server = subprocess.Popen(
            [sys.executable, "-m", "clearml_serving_amd.serving.launch",
             "--store", store_root, "--session", session_id,
             "--host", "127.0.0.1", "--port", str(PORT),
             "--workers", str(args.workers)],
            env=env, cwd=REPO)
        base = "http://127.0.0.1:{}".format(PORT)
        payload = {"input_ids": list(range(1, 129)),
                   "attention_mask": [1] * 128}
        try:
            if not wait_healthy(base):
                raise RuntimeError("service did not come up")
            P = max(args.client_procs, 1)
            ctx = multiprocessing.get_context("spawn")
            pipes, procs = [], []
            per_n = args.n // P
            per_c = max(args.c // P, 1)
            per_warm = max(args.warmup // P, 1)
            url = base + "/serve/transformer_model"
            for i in range(P):
                parent, child = ctx.Pipe()
                p = ctx.Process(target=client_proc,
                                args=(i, per_n, per_c, payload, url,
                                      args.keepalive, per_warm, child))
                p.start()
                pipes.append(parent)
                procs.append(p)
            results = [pipe.recv() for pipe in pipes]
            for p in procs:
                p.join(timeout=30)
            total_n = sum(r["n"] for r in results)
            # wall time = max over client procs (they start ~together)
            dt = max(r["dt"] for r in results)
            lat = sorted(x for r in results for x in r["latencies"])
            errors = sum(r["errors"] for r in results)
            out = {
                "metric": "HTTP requests/s, BERT-base, ab-recipe "
                          "(-n {} -c {})".format(args.n, args.c),
                "value": round(total_n / dt, 2),
                "errors": errors,
                "p50_ms": round(statistics.median(lat) * 1000, 2),
                "p95_ms": round(lat[int(len(lat) * 0.95)] * 1000, 2),
                "p99_ms": round(lat[int(len(lat) * 0.99)] * 1000, 2),
                "topology": ("{} fronts + shm engine-owner".format(
                    args.workers) if args.workers else "single process"),
                "keepalive": bool(args.keepalive),
                "client_procs": P,
            }
            print(json.dumps(out))
            if args.out:
                with open(args.out, "wt") as f:
                    json.dump(out, f)
        finally:
            server.terminate()
            try:
                server.wait(timeout=15)
            except subprocess.TimeoutExpired:
                server.kill()


if __name__ == "__main__":
    main()
