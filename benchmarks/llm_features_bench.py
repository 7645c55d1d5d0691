#!/usr/bin/env python3
"""Compute-accounting benchmark for the opt-in LLM engine features.

Reports DETERMINISTIC compute metrics (forward counts, tokens skipped)
plus wall time, so the effect of each feature is visible even where wall
time is noisy. Runs anywhere; on an MI355X use --preset llama-3-1b or
llama-3-8b.

    python benchmarks/llm_features_bench.py [--preset llama-tiny]
        [--requests 16] [--max-tokens 64]
"""

import argparse
import asyncio
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from clearml_serving_amd.engines.llm.engine import (  # noqa: E402
    LlmEngine, LlmEngineConfig, SamplingParams)


def build(preset, **kw):
    torch.manual_seed(1234)
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    blocks = None if torch.cuda.is_available() else 512
    cfg = LlmEngineConfig(preset=preset, device=dev, num_kv_blocks=blocks,
                          max_model_len=1024, **kw)
    eng = LlmEngine(cfg)
    eng.start()
    return eng


def run_prompts(eng, prompts, max_tokens):
    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=max_tokens,
                                ignore_eos=True)

        async def one(p):
            toks = []
            async for item in eng.generate(p, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one(p) for p in prompts])

    loop = asyncio.new_event_loop()
    try:
        t0 = time.perf_counter()
        outs = loop.run_until_complete(main())
        dt = time.perf_counter() - t0
    finally:
        for t in asyncio.all_tasks(loop):
            t.cancel()
        loop.run_until_complete(asyncio.sleep(0))
        loop.close()
    return outs, dt


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--preset", default="llama-tiny")
    ap.add_argument("--requests", type=int, default=16)
    ap.add_argument("--max-tokens", type=int, default=64)
    args = ap.parse_args()

    shared = "You are a helpful assistant. Context: " + "lorem ipsum " * 40
    prompts = [shared + "Q{}".format(i % 4) for i in range(args.requests)]

    # ---- prefix caching: prompt compute skipped ---------------------- #
    # issued in waves: blocks register when their prefill COMPUTES, so a
    # fully concurrent identical burst all misses (same first-arrival
    # semantics as vLLM); steady-state serving sees the reuse
    for pc in (False, True):
        eng = build(args.preset, enable_prefix_caching=pc)
        dt = 0.0
        wave = max(len(prompts) // 3, 1)
        for lo in range(0, len(prompts), wave):
            _, d = run_prompts(eng, prompts[lo:lo + wave], 8)
            dt += d
        computed = eng.stats["prompt_tokens"]
        hit = eng.stats.get("prefix_cache_hit_tokens", 0)
        total = computed + hit
        print(json.dumps({
            "bench": "prefix_caching", "enabled": pc,
            "preset": args.preset, "requests": args.requests,
            "prompt_tokens_total": total,
            "prompt_tokens_computed": computed,
            "prompt_tokens_skipped": hit,
            "skipped_pct": round(100.0 * hit / max(total, 1), 1),
            "wall_s": round(dt, 3)}), flush=True)
        eng.stop()

    # ---- ngram speculation: decode forwards per token ---------------- #
    rep = "def add(a, b):\n    return a + b\n" * 6  # structured/repetitive
    for spec in (None, {"method": "ngram", "num_spec_tokens": 4,
                        "ngram": 2}):
        eng = build(args.preset, speculative=spec)
        outs, dt = run_prompts(eng, [rep] * 4, args.max_tokens)
        gen = eng.stats["generated_tokens"]
        fwd = eng.stats["decode_batches"]
        print(json.dumps({
            "bench": "ngram_speculation", "enabled": bool(spec),
            "preset": args.preset,
            "generated_tokens": gen, "decode_forwards": fwd,
            "forwards_per_token": round(fwd * 4 / max(gen, 1), 3),
            "proposed": eng.stats.get("spec_proposed", 0),
            "accepted": eng.stats.get("spec_accepted", 0),
            "wall_s": round(dt, 3)}), flush=True)
        eng.stop()


if __name__ == "__main__":
    main()
