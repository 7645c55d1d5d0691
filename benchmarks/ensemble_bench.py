#!/usr/bin/env python3
"""BASELINE config 5: ensemble pipeline (ResNet-50 + BERT-base) with canary
routing, N GPUs (one replica per rank via torchrun, like bench.py).

Each request fans out to a ResNet endpoint and a canary-routed BERT
endpoint (90% v2 / 10% v1) and combines -- the reference's ensemble example
shape at GPU scale.

  torchrun --nproc-per-node N benchmarks/ensemble_bench.py --steps K
"""

import argparse
import asyncio
import json
import os
import statistics
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REQ_PER_STEP = int(os.environ.get("CMLS_BENCH_REQ", 512))


def build(device_idx, tmpdir):
    from clearml_serving_amd.schemas import CanaryEP, ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store = ServingStore(os.path.join(tmpdir, "store"))
    proc = ModelRequestProcessor(store=store, name="ens-bench",
                                 force_create=True)
    proc._metric_log_freq = 0.0
    gpu = torch.cuda.is_available()
    aux = {"max_batch_size": 64, "max_queue_delay_us": 4000,
           "use_graphs": gpu, "gpu": device_idx,
           "dtype": "bfloat16" if gpu else "float32"}

    cards = {
        "resnet_ep": ({"arch": "resnet50", "num_classes": 1000,
                       "dtype": "bfloat16"},
                      dict(input_size=[3, 224, 224], input_type="float32"),
                      ""),
        "bert_ep_v1": ({"arch": "bert-base", "num_labels": 2,
                        "dtype": "bfloat16"},
                       dict(input_size=[[128], [128]],
                            input_type=["int64", "int32"],
                            input_name=["input_ids", "attention_mask"]),
                       "1"),
        "bert_ep_v2": ({"arch": "bert-base", "num_labels": 2,
                        "dtype": "bfloat16"},
                       dict(input_size=[[128], [128]],
                            input_type=["int64", "int32"],
                            input_name=["input_ids", "attention_mask"]),
                       "2"),
    }
    for name, (card, spec, ver) in cards.items():
        p = os.path.join(tmpdir, name + ".json")
        with open(p, "wt") as f:
            json.dump(card, f)
        rec = store.register_model(name=name, project="bench", path=p)
        url = name.split("_v")[0] if "_v" in name else name
        proc.add_endpoint(ModelEndpoint(
            engine_type="hip", serving_url=url, version=ver,
            model_id=rec.model_id, auxiliary_cfg=dict(aux), **spec))
    proc.add_canary_endpoint(CanaryEP(
        endpoint="bert_canary", weights=[0.9, 0.1],
        load_endpoint_prefix="bert_ep"))
    proc._update_canary_lookup()

    code = os.path.join(tmpdir, "ens.py")
    with open(code, "wt") as f:
        f.write(
            "import asyncio\n"
            "class Preprocess(object):\n"
            "    async def process(self, data, state, collect_custom_statistics_fn=None):\n"
            "        img, txt = data\n"
            "        a, b = await asyncio.gather(\n"
            "            self.send_request(endpoint='resnet_ep', data=img),\n"
            "            self.send_request(endpoint='bert_canary', data=txt))\n"
            "        return {'cls': 1, 'ok': a is not None and b is not None}\n")
    proc.add_endpoint(
        __import__("clearml_serving_amd.schemas", fromlist=["ModelEndpoint"])
        .ModelEndpoint(engine_type="custom_async", serving_url="ensemble"),
        preprocess_code=code)
    return proc


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=6)
    ap.add_argument("--warmup", type=int, default=2)
    args = ap.parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    has_gpu = torch.cuda.is_available()
    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        dist.init_process_group("nccl" if has_gpu else "gloo")
        if has_gpu:
            torch.cuda.set_device(local_rank)

    torch.manual_seed(1 + rank)
    dtype = torch.bfloat16 if has_gpu else torch.float32
    img = torch.randn(3, 224, 224, dtype=dtype)
    txt = {"input_ids": torch.randint(0, 30000, (128,)),
           "attention_mask": torch.ones(128, dtype=torch.int32)}

    with tempfile.TemporaryDirectory() as tmpdir:
        proc = build(local_rank if has_gpu else 0, tmpdir)
        loop = asyncio.new_event_loop()

        async def step(latencies=None):
            async def one():
                t0 = time.perf_counter()
                await proc.process_request("ensemble", "", (img, txt))
                if latencies is not None:
                    latencies.append(time.perf_counter() - t0)

            await asyncio.gather(*[one() for _ in range(REQ_PER_STEP)])

        for _ in range(args.warmup):
            loop.run_until_complete(step())
        if dist:
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()
        lat = []
        t0 = time.perf_counter()
        for _ in range(args.steps):
            loop.run_until_complete(step(lat))
        if has_gpu:
            torch.cuda.synchronize()
        if dist:
            dist.barrier()
        dt = time.perf_counter() - t0
        if dist:
            t = torch.tensor([dt], dtype=torch.float64,
                             device="cuda" if has_gpu else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            dt = float(t.item())
        if rank == 0:
            print(json.dumps({
                "metric": "ensemble+canary requests/s (config 5)",
                "value": round(world * args.steps * REQ_PER_STEP / dt, 2),
                "n_gpus": world,
                "p50_latency_ms": round(
                    statistics.median(lat) * 1000, 2) if lat else None,
            }))
        if dist:
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
