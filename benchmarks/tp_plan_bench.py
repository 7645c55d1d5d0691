#!/usr/bin/env python3
"""TP step-plan scheduling overhead microbench (VERDICT round-1 item 2).

Measures the per-decode-step cost of the scheduler->worker coordination
alone (no model): rank 0 encodes a b=64 decode plan, ONE int32 tensor
broadcast, workers decode it -- vs the round-1 broadcast_object_list
(pickle) path. Run at world=8 on CPU/gloo (the GPU path replaces gloo with
RCCL over xGMI, which is faster still, and the buffer lands directly in
device memory).

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 --master-port 29551 benchmarks/tp_plan_bench.py
"""

import json
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from clearml_serving_amd.engines.llm.engine import LlmEngineConfig  # noqa: E402
from clearml_serving_amd.engines.llm.plan_codec import PlanCodec  # noqa: E402


def make_plan(b=64, blocks_per_seq=16):
    return {
        "mode": "decode",
        "tokens": list(range(b)),
        "positions": [200 + i for i in range(b)],
        "slots": [4096 + i for i in range(b)],
        "seq_lens": [201 + i for i in range(b)],
        "blocks": [[j for j in range(blocks_per_seq)] for _ in range(b)],
        "sample": [(0.7, 0, 1.0, 12345)] * b,
    }


def bench_codec(codec, plan, rank, iters=500):
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        if rank == 0:
            buf = codec.encode(plan)
            dist.broadcast(buf, src=0)
        else:
            dist.broadcast(codec.buf, src=0)
            codec.decode(codec.buf)
    dist.barrier()
    return (time.perf_counter() - t0) / iters


def bench_pickle(plan, rank, iters=200):
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(iters):
        if rank == 0:
            dist.broadcast_object_list([plan], src=0)
        else:
            box = [None]
            dist.broadcast_object_list(box, src=0)
    dist.barrier()
    return (time.perf_counter() - t0) / iters


def main():
    dist.init_process_group(backend="gloo")
    rank, world = dist.get_rank(), dist.get_world_size()
    cfg = LlmEngineConfig(preset="llama-3-8b", max_num_seqs=64,
                          max_model_len=4096, block_size=16)
    codec = PlanCodec(cfg, torch.device("cpu"))
    plan = make_plan()

    # warmup both paths
    bench_codec(codec, plan, rank, iters=50)
    bench_pickle(plan, rank, iters=20)

    codec_s = bench_codec(codec, plan, rank)
    pickle_s = bench_pickle(plan, rank)

    # CPU-side costs alone (what persists on a GPU deployment, where the
    # gloo TCP broadcast below is replaced by a ~5 us RCCL/xGMI broadcast;
    # gloo world=8 on loopback has a ~300 us latency floor for ANY payload)
    t0 = time.perf_counter()
    for _ in range(1000):
        codec.encode(plan)
    enc_s = (time.perf_counter() - t0) / 1000
    buf = codec.encode(plan)
    t0 = time.perf_counter()
    for _ in range(1000):
        codec.decode(buf)
    dec_s = (time.perf_counter() - t0) / 1000
    import pickle

    pickled = pickle.dumps(plan, protocol=4)

    if rank == 0:
        out = {
            "metric": "TP decode-step scheduling overhead (plan broadcast "
                      "+ decode), b=64, world={} gloo/CPU".format(world),
            "codec_us_per_step": round(codec_s * 1e6, 1),
            "pickle_us_per_step": round(pickle_s * 1e6, 1),
            "encode_us": round(enc_s * 1e6, 1),
            "decode_us": round(dec_s * 1e6, 1),
            "buffer_kib": round(codec.capacity * 4 / 1024, 1),
            "pickle_payload_kib": round(len(pickled) / 1024, 1),
            "note": "gloo/TCP world=8 broadcast floor ~300us dominates the "
                    "end-to-end numbers here; on MI355X the broadcast is "
                    "one RCCL xGMI collective (us-scale) and the CPU "
                    "encode+decode columns are the real per-step cost",
        }
        print(json.dumps(out), flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
