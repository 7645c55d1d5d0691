"""GPU tests added AFTER the round's last GPU window (speculative
decoding, Qwen2 qkv_bias, extended GQA groups): kept in a file that sorts
LAST so a surprise on fresh hardware cannot halt (-x) the proven suite."""

import asyncio

import pytest
import torch

import clearml_serving_amd.ops as ops
from clearml_serving_amd.engines.llm.engine import (
    LlmEngine,
    LlmEngineConfig,
    SamplingParams,
)
from tests.test_llm_gpu import run  # noqa: F401
from tests.test_llm_gpu import \
    test_attention_decode_numerics as _decode_numerics  # not re-collected

pytestmark = pytest.mark.gpu

DEV = "cuda:0"




def test_spec_decode_gpu_matches_plain_greedy():
    """ngram speculative decoding on the GPU kernel path (multi-token
    verify through attention_prefill_paged_v2): token-identical to plain
    greedy decode, and speculation actually fires."""
    def gen(speculative):
        torch.manual_seed(11)
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=128,
                              block_size=16, max_model_len=256, device=DEV,
                              max_num_seqs=8, speculative=speculative)
        eng = LlmEngine(cfg)
        eng.start()

        async def go():
            # horizon kept short: spec's verify runs the paged-prefill
            # kernel while plain decode runs attention_decode -- greedy
            # equality holds modulo bf16 near-ties between the two
            # reduction orders (chunked-vs-dense exactness passed on this
            # hardware at similar scales)
            prompts = ["abcabcabcabcabc", "the quick brown fox", "zq!7#"]
            outs = []
            for p in prompts:
                toks = []
                async for item in eng.generate(p, SamplingParams(
                        temperature=0.0, max_tokens=12, ignore_eos=True)):
                    toks.extend(item["token_ids"])
                outs.append(toks)
            return outs

        return run(go()), eng.stats

    plain, _ = gen(None)
    spec, stats = gen({"method": "ngram", "num_spec_tokens": 4, "ngram": 2})
    assert plain == spec
    assert stats["spec_proposed"] > 0 and stats["spec_accepted"] > 0


def test_qwen2_qkv_bias_gpu_generation():
    """qkv_bias (Qwen2 family) on the GPU path: skinny projections plus the
    bias add; greedy generation deterministic and bias-sensitive."""
    torch.manual_seed(12)
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=128,
                          block_size=16, max_model_len=256, device=DEV,
                          overrides={"qkv_bias": True})
    eng = LlmEngine(cfg)
    eng.start()
    assert eng.model.layers[0].qkv.bias is not None

    async def go():
        toks = []
        async for item in eng.generate("qwen gpu probe", SamplingParams(
                temperature=0.0, max_tokens=12, ignore_eos=True)):
            toks.extend(item["token_ids"])
        return toks

    a = run(go())
    b = run(go())
    assert len(a) == 12 and a == b
    with torch.no_grad():
        for layer in eng.model.layers:
            layer.qkv.bias.zero_()
    c = run(go())
    assert c != a


# placed at file END deliberately: these GQA groups (3/5/6/7 -- the qwen2
# shapes) were instantiated after the last GPU window of the round, so an
# unexpected failure here must not halt (-x) the proven suite above
@pytest.mark.parametrize("b,h,hkv,d,seqs", [
    (2, 28, 4, 128, [9, 210]),   # GQ=7: qwen2-7b decode shape
    (2, 12, 2, 128, [9, 130]),   # GQ=6: qwen2-1.5b
    (2, 6, 2, 128, [21, 64]),    # GQ=3
    (2, 10, 2, 128, [21, 64]),   # GQ=5
])
def test_attention_decode_numerics_extended_gqa(b, h, hkv, d, seqs):
    _decode_numerics(b, h, hkv, d, seqs)


def test_gpt2_engine_gpu_generation():
    """GPT-2 through the engine on the GPU kernel path (layernorm /
    bias_gelu / attention / paged decode): deterministic greedy."""
    torch.manual_seed(13)
    cfg = LlmEngineConfig(preset="gpt2-tiny", num_kv_blocks=128,
                          block_size=16, max_model_len=256, device=DEV)
    eng = LlmEngine(cfg)
    eng.start()

    async def go():
        toks = []
        async for item in eng.generate("gpt2 gpu probe", SamplingParams(
                temperature=0.0, max_tokens=12, ignore_eos=True)):
            toks.extend(item["token_ids"])
        return toks

    a = run(go())
    b = run(go())
    assert len(a) == 12 and a == b
    assert eng.allocator.available == eng.allocator.num_blocks


def test_attention_prefill_paged_tiny_chunks():
    """Spec decode feeds the paged-prefill kernel 1..K+1-row chunks;
    exercise the q_len=1 and mixed tiny-q edge directly vs the CPU
    reference."""
    torch.manual_seed(21)
    b, h, hkv, d, bs = 3, 8, 2, 128, 16
    hist = [40, 7, 0]
    chunk = [1, 3, 5]
    sq = max(chunk)
    nb = 12
    k_cache = torch.randn(nb, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    v_cache = torch.randn_like(k_cache)
    block_table = torch.tensor([[0, 3, 6, 9], [1, 4, 7, 10], [2, 5, 8, 11]],
                               dtype=torch.int32, device=DEV)
    kv_lens = torch.tensor([hist[i] + chunk[i] for i in range(b)],
                           dtype=torch.int32, device=DEV)
    q_lens = torch.tensor(chunk, dtype=torch.int32, device=DEV)
    q = torch.randn(b, sq, h, d, device=DEV, dtype=torch.bfloat16)

    got = ops.attention_prefill_paged(q, k_cache, v_cache, block_table,
                                      kv_lens, q_lens)
    ref = ops.attention_prefill_paged(
        q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
        block_table.cpu(), kv_lens.cpu(), q_lens.cpu())
    for i in range(b):
        torch.testing.assert_close(got[i, :chunk[i]].float().cpu(),
                                   ref[i, :chunk[i]].float(),
                                   atol=3e-2, rtol=3e-2)


def test_prefix_caching_gpu_exact():
    """Automatic prefix caching on the GPU path: the cache-hit request
    (paged-prefill over reused pages + decode) emits identical greedy
    tokens, and blocks are actually reused."""
    def mk(pc):
        torch.manual_seed(17)
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=128,
                              block_size=16, max_model_len=256, device=DEV,
                              enable_prefix_caching=pc)
        eng = LlmEngine(cfg)
        eng.start()
        return eng

    def gen(eng, prompt):
        async def go():
            toks = []
            async for item in eng.generate(prompt, SamplingParams(
                    temperature=0.0, max_tokens=10, ignore_eos=True)):
                toks.extend(item["token_ids"])
            return toks

        return run(go())

    plain = mk(False)
    pc = mk(True)
    prompt = "gpu shared prefix " * 4
    a = gen(plain, prompt)
    assert gen(pc, prompt) == a
    assert gen(pc, prompt) == a  # hit pass
    assert pc.allocator.hit_tokens >= 16
