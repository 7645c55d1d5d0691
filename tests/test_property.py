"""Property-based tests (hypothesis) for the routing/scheduling math."""

import math

from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=200, deadline=None)
@given(weights=st.lists(st.floats(min_value=0.01, max_value=100.0),
                        min_size=1, max_size=8))
def test_canary_weight_normalization(weights):
    """Canary route weights always normalize to a probability vector."""
    from clearml_serving_amd.schemas import CanaryEP
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    import tempfile

    with tempfile.TemporaryDirectory() as d:
        store = ServingStore(d)
        p = ModelRequestProcessor(store=store, name="prop",
                                  force_create=True)
        eps = ["e/{}".format(i) for i in range(len(weights))]
        p.add_canary_endpoint(CanaryEP(endpoint="c", weights=list(weights),
                                       load_endpoints=eps))
        p._update_canary_lookup()
        route = p._canary_route["c"]
        total = sum(route["weights"])
        assert math.isclose(total, 1.0, rel_tol=1e-9)
        assert all(w >= 0 for w in route["weights"])
        assert len(route["endpoints"]) == len(weights)


@settings(max_examples=200, deadline=None)
@given(n=st.integers(min_value=1, max_value=64),
       buckets=st.lists(st.integers(min_value=1, max_value=64),
                        min_size=1, max_size=7))
def test_batcher_bucket_selection(n, buckets):
    """The chosen bucket is always >= the batch size and is the smallest
    qualifying preferred size."""
    from clearml_serving_amd.serving.batcher import DynamicBatcher

    b = DynamicBatcher(lambda x: x, device="cpu", max_batch_size=64,
                       preferred_batch_sizes=buckets, use_graphs=False)
    chosen = next(bk for bk in b.buckets if bk >= n)
    assert chosen >= n
    assert all(bk < n for bk in b.buckets if bk < chosen)
    assert b.buckets[-1] == 64  # max_batch_size always a bucket


@settings(max_examples=100, deadline=None)
@given(prompt=st.lists(st.integers(min_value=1, max_value=400),
                       min_size=1, max_size=60),
       block=st.sampled_from([4, 8, 16, 32]))
def test_block_allocator_slot_math(prompt, block):
    """Slot mapping is a bijection prompt-position -> (block, offset)."""
    from clearml_serving_amd.engines.llm.engine import BlockAllocator

    alloc = BlockAllocator(64)
    n_blocks = (len(prompt) + block - 1) // block
    blocks = alloc.alloc(n_blocks)
    slots = [blocks[p // block] * block + p % block
             for p in range(len(prompt))]
    assert len(set(slots)) == len(slots)
    for p, s in enumerate(slots):
        assert s // block == blocks[p // block]
    alloc.free(blocks)
    assert alloc.available == 64


# --------------------------------------------------------------------- #
# speculative decoding: exactness as a PROPERTY over random prompts
# --------------------------------------------------------------------- #
@settings(max_examples=12, deadline=None)
@given(prompt=st.lists(st.integers(min_value=1, max_value=500),
                       min_size=2, max_size=40),
       k=st.integers(min_value=1, max_value=6),
       ngram=st.integers(min_value=1, max_value=3),
       max_tokens=st.integers(min_value=1, max_value=20))
def test_spec_decode_exactness_property(prompt, k, ngram, max_tokens):
    """For ANY prompt/config, ngram speculative decoding emits exactly the
    tokens plain greedy decode would (the defining acceptance invariant)."""
    import asyncio

    import torch

    from clearml_serving_amd.engines.llm.engine import (
        LlmEngine, LlmEngineConfig, SamplingParams)

    global _SPEC_ENGINES
    try:
        plain, spec = _SPEC_ENGINES
    except NameError:
        def mk(speculative):
            torch.manual_seed(7)
            cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=128,
                                  block_size=16, max_model_len=128,
                                  device="cpu", speculative=speculative)
            e = LlmEngine(cfg)
            e.start()
            return e

        plain = mk(None)
        spec = mk({"method": "ngram", "num_spec_tokens": 6, "ngram": 2})
        _SPEC_ENGINES = (plain, spec)
    spec.cfg.speculative = {"method": "ngram", "num_spec_tokens": k,
                            "ngram": ngram}

    def gen(eng):
        async def go():
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=max_tokens, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        loop = asyncio.new_event_loop()
        try:
            return loop.run_until_complete(go())
        finally:
            pending = asyncio.all_tasks(loop)
            for t in pending:
                t.cancel()
            if pending:
                loop.run_until_complete(
                    asyncio.gather(*pending, return_exceptions=True))
            loop.close()

    assert gen(spec) == gen(plain)


# --------------------------------------------------------------------- #
# dynamic batcher: response integrity as a PROPERTY over random streams
# --------------------------------------------------------------------- #
@settings(max_examples=10, deadline=None)
@given(sizes=st.lists(st.integers(min_value=1, max_value=9),
                      min_size=1, max_size=40),
       feat=st.integers(min_value=1, max_value=6))
def test_batcher_every_response_matches_direct_call(sizes, feat):
    """Whatever the arrival pattern (random burst sizes, padding, bucket
    boundaries), each request's batched response must equal model(x) run
    directly on its own input -- no cross-request leakage, no pad rows in
    responses."""
    import asyncio

    import torch

    from clearml_serving_amd.serving.batcher import DynamicBatcher

    def model(x):
        # row-wise, input-dependent (leakage across rows would show)
        return x * 2.0 + x.sum(dim=-1, keepdim=True)

    b = DynamicBatcher(model_fn=model, device="cpu", max_batch_size=8,
                       max_queue_delay_us=500, use_graphs=False)

    reqs = []
    for burst in sizes:
        for _ in range(burst):
            reqs.append(torch.randn(feat))

    async def main():
        outs = await asyncio.gather(*[b.submit(r) for r in reqs])
        return outs

    loop = asyncio.new_event_loop()
    try:
        outs = loop.run_until_complete(main())
    finally:
        for t in asyncio.all_tasks(loop):
            t.cancel()
        loop.run_until_complete(asyncio.sleep(0))
        loop.close()

    for r, o in zip(reqs, outs):
        want = model(r[None])[0]
        torch.testing.assert_close(o, want, atol=1e-6, rtol=1e-6)


# --------------------------------------------------------------------- #
# prefix-cache allocator invariants under random op sequences
# --------------------------------------------------------------------- #
@settings(max_examples=30, deadline=None)
@given(ops_seq=st.lists(
    st.tuples(st.sampled_from(["admit", "finish"]),
              st.integers(min_value=0, max_value=5),   # prompt family
              st.integers(min_value=1, max_value=40)), # prompt length
    min_size=1, max_size=60))
def test_prefix_allocator_invariants(ops_seq):
    """Random admit/finish interleavings: no block is ever handed to two
    live sequences, accounting always balances, and matches never exceed
    len(prompt)-1 tokens."""
    from clearml_serving_amd.engines.llm.engine import PrefixCacheAllocator

    BS = 4
    N = 16
    a = PrefixCacheAllocator(N, BS)
    live = []  # (blocks, prompt)
    in_use = {}  # block -> count of live holders (shared cached blocks ok)

    def track(blocks):
        for b in blocks:
            in_use[b] = in_use.get(b, 0) + 1

    def untrack(blocks):
        for b in blocks:
            in_use[b] -= 1
            if not in_use[b]:
                del in_use[b]

    for op, fam, ln in ops_seq:
        if op == "admit":
            prompt = [(fam * 97 + i) % 50 for i in range(ln)]
            cached, ncached, chain = a.match(prompt)
            assert ncached <= max(len(prompt) - 1, 0)
            need = (len(prompt) + BS - 1) // BS - len(cached)
            if need > a.available:
                a.free(cached)
                continue
            try:
                fresh = a.alloc(need)
            except RuntimeError:
                a.free(cached)
                continue
            # a fresh block must not be LIVE anywhere (shared cached
            # blocks may be held by several sequences; owned may not)
            for b in fresh:
                assert b not in in_use, (b, in_use)
            blocks = cached + fresh
            track(blocks)
            h = chain
            for i in range(ncached // BS, len(prompt) // BS):
                h = a.register_block(h, prompt[i * BS:(i + 1) * BS],
                                     blocks[i])
            live.append((blocks, prompt))
        elif live:
            blocks, _ = live.pop(fam % len(live))
            untrack(blocks)
            a.free(blocks)

    for blocks, _ in live:
        a.free(blocks)
    # every block ends up free or evictable
    assert a.available == N


# --------------------------------------------------------------------- #
# request-body fuzz: from_request never escapes ValueError
# --------------------------------------------------------------------- #
_scalar = st.one_of(st.none(), st.booleans(), st.integers(-10**6, 10**6),
                    st.floats(allow_nan=True, allow_infinity=True),
                    st.text(max_size=8))
_value = st.one_of(_scalar, st.lists(_scalar, max_size=4),
                   st.dictionaries(st.text(max_size=6), _scalar, max_size=3))


@settings(max_examples=200, deadline=None)
@given(body=st.dictionaries(
    st.sampled_from(["temperature", "top_k", "top_p", "max_tokens",
                     "max_completion_tokens", "min_tokens", "seed", "n",
                     "presence_penalty", "frequency_penalty",
                     "repetition_penalty", "logprobs", "top_logprobs",
                     "stop", "stop_token_ids", "ignore_eos", "best_of",
                     "response_format", "stream", "echo", "garbage"]),
    _value, max_size=8))
def test_sampling_params_fuzz_never_crashes(body):
    """Arbitrary request bodies either validate (SamplingParams) or raise
    ValueError (-> 422); any other exception class would 500 and is a
    bug in the validation layer."""
    from clearml_serving_amd.engines.llm.engine import SamplingParams

    try:
        p = SamplingParams.from_request(dict(body))
    except ValueError:
        return
    # accepted: the invariants the sampling kernels rely on hold
    assert p.temperature >= 0.0
    assert p.top_k >= 0
    assert 0.0 < p.top_p <= 1.0
    assert 1 <= p.max_tokens
    assert 0 <= p.min_tokens <= p.max_tokens
    assert -2.0 <= p.presence_penalty <= 2.0
    assert -2.0 <= p.frequency_penalty <= 2.0
    assert p.repetition_penalty > 0.0
    assert p.logprobs is None or 0 <= p.logprobs <= 20
