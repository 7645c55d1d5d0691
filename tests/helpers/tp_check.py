"""TP=2 correctness worker (launched by test_tp_cpu via torchrun, gloo).

Builds a full tp=1 llama-tiny, shards its weights into this rank's tp=2
model, and checks TP logits == full-model logits; then runs the engine's
plan-broadcast protocol end to end (rank 0 schedules, rank 1 follows)."""

import asyncio
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))

from clearml_serving_amd.engines.llm.engine import (  # noqa: E402
    LlmEngine, LlmEngineConfig, SamplingParams)
from clearml_serving_amd.models.llama import (  # noqa: E402
    PRESETS, LlamaForCausalLM)
from clearml_serving_amd.parallel import tp  # noqa: E402


def _tiny_cfg(world):
    """llama-tiny, widened when the TP degree exceeds its kv_heads=2 (the
    sharding contract needs heads and kv_heads divisible by world)."""
    cfg = PRESETS["llama-tiny"]
    if cfg.kv_heads % world or cfg.heads % world:
        from clearml_serving_amd.models.llama import LlamaConfig

        cfg = LlamaConfig(**{**cfg.__dict__, "heads": 2 * world,
                             "kv_heads": world})
    return cfg


def check_tp_math(rank, world):
    cfg = _tiny_cfg(world)
    torch.manual_seed(7)
    full = LlamaForCausalLM(cfg).eval()
    shard_state = tp.shard_llama_weights(full.state_dict(), cfg, rank, world)
    tp_model = LlamaForCausalLM(cfg, tp_rank=rank, tp_size=world).eval()
    tp_model.load_state_dict(shard_state)

    tokens = torch.arange(10, 26, dtype=torch.long)
    positions = torch.arange(16, dtype=torch.int32)
    ctx = {"mode": "prefill", "batch": 1, "seq": 16,
           "seq_lens": torch.tensor([16], dtype=torch.int32),
           "slot_mapping": torch.full((16,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        ref = full(tokens, positions, kv_caches=None, attn_ctx=ctx)
        got = tp_model(tokens, positions, kv_caches=None, attn_ctx=ctx)
    torch.testing.assert_close(got, ref, atol=2e-4, rtol=2e-4)
    if rank == 0:
        print("TP-MATH-OK", flush=True)


def check_sharded_sampling(rank, world):
    """Distributed sampling primitives at world=2: packed-argmax all-reduce
    and rank-0 row gather must equal their single-process counterparts."""
    torch.manual_seed(3)
    full_logits = torch.randn(6, 64)
    # same full matrix on every rank (same seed); take this rank's shard
    w = 64 // world
    shard = full_logits[:, rank * w:(rank + 1) * w].contiguous()
    got = tp.argmax_sharded(shard, rank * w)
    assert torch.equal(got, full_logits.argmax(dim=-1)), (got,)

    gathered = tp.gather_rows_to_rank0(shard)
    if rank == 0:
        assert torch.equal(gathered, full_logits)
    else:
        assert gathered is None

    # gumbel-max across shards: empirical dist ~ softmax on a 4-token vocab
    logits = torch.tensor([[1.5, 0.5, -0.5, -1.5]])
    w2 = 4 // world
    shard2 = logits[:, rank * w2:(rank + 1) * w2].contiguous()
    counts = torch.zeros(4)
    n = 3000
    for i in range(n):
        idx = tp.sample_gumbel_sharded(shard2, rank * w2, 1.0, [i])
        counts[idx.item()] += 1
    probs = torch.softmax(logits[0], dim=-1)
    assert torch.allclose(counts / n, probs, atol=0.04), (counts / n, probs)
    if rank == 0:
        print("TP-SAMPLE-OK", flush=True)


def check_engine_protocol(rank, world):
    mcfg = _tiny_cfg(world)
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                          block_size=16, max_model_len=128, device="cpu",
                          overrides={"heads": mcfg.heads,
                                     "kv_heads": mcfg.kv_heads})
    eng = LlmEngine(cfg)
    eng.start()
    assert eng.tp_size == world
    if rank == 0:
        async def gen():
            params = SamplingParams(temperature=0.0, max_tokens=6,
                                    ignore_eos=True)

            async def one(i):
                toks = []
                async for item in eng.generate("req %d" % i, params):
                    toks.extend(item["token_ids"])
                return toks

            return await asyncio.gather(one(0), one(1), one(2))

        loop = asyncio.new_event_loop()
        outs = loop.run_until_complete(gen())
        assert all(len(o) == 6 for o in outs), outs

        # stochastic + top-k sampling exercise the gumbel and rank-0-gather
        # collective paths through the full engine protocol
        async def gen2():
            toks = []
            params2 = SamplingParams(temperature=0.8, top_k=8, top_p=0.95,
                                     max_tokens=5, ignore_eos=True)
            async for item in eng.generate("topk req", params2):
                toks.extend(item["token_ids"])
            toks3 = []
            params3 = SamplingParams(temperature=0.8, max_tokens=5,
                                     ignore_eos=True)
            async for item in eng.generate("gumbel req", params3):
                toks3.extend(item["token_ids"])
            return toks, toks3

        t2, t3 = loop.run_until_complete(gen2())
        assert len(t2) == 5 and len(t3) == 5
        # embeddings path through the same plan-broadcast protocol
        vecs = loop.run_until_complete(
            eng.embed_batch(["tp embed a", "a longer tp embed text b"]))
        assert len(vecs) == 2
        assert abs(sum(x * x for x in vecs[0]) - 1.0) < 1e-4

        # oversize embed batch: exceeds the fixed plan buffer, so the
        # codec's OBJECT fallback broadcast carries it (plan_codec mode 5)
        assert eng._plan_codec.encode(
            {"mode": "embed",
             "prompts": [[1] * 100 for _ in range(400)]}) is None
        big = ["x" * 100 for _ in range(400)]
        vecs2 = loop.run_until_complete(eng.embed_batch(big))
        assert len(vecs2) == 400
        eng.tp_shutdown()
        print("TP-ENGINE-OK", flush=True)
    else:
        eng.run_tp_worker()


def check_microbatch_pipeline(rank, world):
    """CMLS_TP_MICROBATCH=1 (decode comm/compute overlap via async
    all-reduce pipelining, model.forward_pipelined): greedy generation
    must be token-identical to the plain TP decode path."""
    mcfg = _tiny_cfg(world)
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                          block_size=16, max_model_len=128, device="cpu",
                          overrides={"heads": mcfg.heads,
                                     "kv_heads": mcfg.kv_heads})

    def gen(eng):
        async def go():
            params = SamplingParams(temperature=0.0, max_tokens=8,
                                    ignore_eos=True)

            async def one(i):
                toks = []
                async for item in eng.generate("mb probe %d" % i, params):
                    toks.extend(item["token_ids"])
                return toks

            return await asyncio.gather(*[one(i) for i in range(3)])

        loop = asyncio.new_event_loop()
        try:
            return loop.run_until_complete(go())
        finally:
            loop.close()

    results = {}
    for flag in ("0", "1"):
        os.environ["CMLS_TP_MICROBATCH"] = flag
        dist.barrier()
        eng = LlmEngine(cfg)
        eng.start()
        if rank == 0:
            results[flag] = gen(eng)
            eng.tp_shutdown()
        else:
            eng.run_tp_worker()
    if rank == 0:
        assert results["1"] == results["0"], (results,)
        print("TP-MICROBATCH-OK", flush=True)
    os.environ.pop("CMLS_TP_MICROBATCH", None)


def check_prefix_cache_under_tp(rank, world):
    """Prefix caching under TP: rank 0's allocator schedules, cached
    admission routes the remainder through the CHUNK plan (prefilled>0),
    so workers see a different broadcast mode -- outputs must stay
    identical across the cold and cache-hit passes."""
    mcfg = _tiny_cfg(world)
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                          block_size=16, max_model_len=128, device="cpu",
                          enable_prefix_caching=True,
                          overrides={"heads": mcfg.heads,
                                     "kv_heads": mcfg.kv_heads})
    eng = LlmEngine(cfg)
    eng.start()
    if rank == 0:
        async def gen():
            toks = []
            async for item in eng.generate(
                    "tp cached prefix " * 4,
                    SamplingParams(temperature=0.0, max_tokens=8,
                                   ignore_eos=True)):
                toks.extend(item["token_ids"])
            return toks

        loop = asyncio.new_event_loop()
        try:
            a = loop.run_until_complete(gen())
            b = loop.run_until_complete(gen())  # cache-hit pass
        finally:
            loop.close()
        assert a == b, (a, b)
        assert eng.allocator.hit_tokens >= 16, eng.allocator.hit_tokens
        eng.tp_shutdown()
        print("TP-PREFIXCACHE-OK", flush=True)
    else:
        eng.run_tp_worker()


def main():
    dist.init_process_group(backend="gloo")
    rank, world = dist.get_rank(), dist.get_world_size()
    check_tp_math(rank, world)
    check_sharded_sampling(rank, world)
    check_engine_protocol(rank, world)
    dist.barrier()
    check_microbatch_pipeline(rank, world)
    dist.barrier()
    check_prefix_cache_under_tp(rank, world)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
