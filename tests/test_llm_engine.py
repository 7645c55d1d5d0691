"""Native LLM engine: block allocator, scheduler, generation (CPU path)."""

import asyncio
import json

import pytest
import torch

from clearml_serving_amd.engines.llm.engine import (
    BlockAllocator,
    LlmEngine,
    LlmEngineConfig,
    SamplingParams,
    SimpleTokenizer,
)


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        # cancel lingering workers (batcher/engine loops) before closing
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()


def tiny_engine(**kw) -> LlmEngine:
    defaults = dict(preset="llama-tiny", num_kv_blocks=128,
                    block_size=16, max_model_len=256, device="cpu")
    cfg = LlmEngineConfig(**{**defaults, **kw})
    eng = LlmEngine(cfg)
    eng.start()
    return eng


def test_block_allocator():
    a = BlockAllocator(8)
    b1 = a.alloc(3)
    assert len(set(b1)) == 3 and a.available == 5
    a.free(b1)
    assert a.available == 8
    with pytest.raises(RuntimeError):
        a.alloc(9)


def test_tokenizer_roundtrip():
    t = SimpleTokenizer()
    assert t.decode(t.encode("hello world")) == "hello world"


def test_engine_greedy_generation_deterministic():
    eng = tiny_engine()

    async def gen():
        out = []
        async for item in eng.generate(
                "hello", SamplingParams(temperature=0.0, max_tokens=8,
                                        ignore_eos=True)):
            out.extend(item["token_ids"])
        return out

    a = run(gen())
    b = run(gen())
    assert len(a) == 8
    assert a == b  # greedy is deterministic


def test_engine_decode_matches_single_shot_prefill():
    """KV-cache correctness: greedy tokens from incremental decode must match
    teacher-forced logits from a single prefill over the full sequence."""
    eng = tiny_engine()
    prompt = [3, 7, 11, 19, 23]

    async def gen():
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=6,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    generated = run(gen())
    assert len(generated) == 6

    # teacher-forced reference: run the whole sequence through one prefill
    # and check each generated token is the argmax at its position
    model = eng.model
    full = prompt + generated
    t = len(full)
    tokens = torch.tensor(full, dtype=torch.long)
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        logits = model(tokens, positions, kv_caches=None, attn_ctx=attn_ctx)
    for step in range(6):
        pos = len(prompt) + step - 1
        expect = int(logits[pos].argmax())
        assert generated[step] == expect, (
            "decode diverged from teacher-forced prefill at step {}".format(step))


def test_engine_concurrent_requests_continuous_batching():
    eng = tiny_engine()

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=5, ignore_eos=True)

        async def one(text):
            toks = []
            async for item in eng.generate(text, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one("req %d" % i) for i in range(6)])

    outs = run(main())
    assert all(len(o) == 5 for o in outs)
    # decode steps were batched (6 seqs in flight -> far fewer than 6*5 steps)
    assert eng.stats["decode_batches"] <= 12
    # all blocks returned after completion
    assert eng.allocator.available == eng.allocator.num_blocks


def test_engine_blocks_freed_and_reused():
    eng = tiny_engine()
    params = SamplingParams(temperature=0.0, max_tokens=40, ignore_eos=True)

    async def one():
        toks = []
        async for item in eng.generate("x" * 100, params):
            toks.extend(item["token_ids"])
        return toks

    for _ in range(3):  # repeated long generations would exhaust 128 blocks
        out = run(one())  # noqa
    assert eng.allocator.available == eng.allocator.num_blocks


def test_generate_simple_and_openai_shapes():
    eng = tiny_engine()
    out = run(eng.generate_simple({"prompt": "hi", "max_tokens": 4,
                                   "temperature": 0.0, "ignore_eos": True}))
    assert out["tokens"] == 4

    resp = run(eng.openai_chat_completions(
        {"messages": [{"role": "user", "content": "hi"}], "max_tokens": 4,
         "temperature": 0.0, "ignore_eos": True}, "test_llm"))
    assert resp["object"] == "chat.completion"
    assert resp["choices"][0]["finish_reason"] in ("length", "stop")
    assert resp["usage"]["completion_tokens"] >= 1

    resp = run(eng.openai_completions(
        {"prompt": "hello", "max_tokens": 3, "temperature": 0.0,
         "ignore_eos": True}, "test_llm"))
    assert resp["object"] == "text_completion"

    models = eng.openai_models("test_llm")
    assert models["data"][0]["id"] == "test_llm"


def test_config_from_aux(tmp_path):
    card = tmp_path / "card.json"
    card.write_text(json.dumps({
        "arch": "llama", "preset": "llama-tiny", "block_size": 32,
        "max_model_len": 512}))
    cfg = LlmEngineConfig.from_aux(str(card), {"max_num_seqs": 8})
    assert cfg.preset == "llama-tiny"
    assert cfg.block_size == 32
    assert cfg.max_num_seqs == 8
    assert cfg.max_model_len == 512


def test_kv_pressure_preemption():
    """With too few KV blocks for all sequences, the engine preempts the
    newest sequence (recompute) instead of failing -- all requests still
    complete."""
    eng = tiny_engine()
    # shrink capacity: 6 blocks = 96 token slots total
    from clearml_serving_amd.engines.llm.engine import BlockAllocator

    eng.allocator = BlockAllocator(6)

    async def main():
        # 30-token prompts + 40 generated = 5 blocks per finished sequence;
        # two admitted sequences need 10 blocks > 6 -> decode growth must
        # preempt (admission control alone cannot absorb it)
        params = SamplingParams(temperature=0.0, max_tokens=40,
                                ignore_eos=True)

        async def one(i):
            toks = []
            async for item in eng.generate("p%d" % i + "x" * 28, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one(i) for i in range(4)])

    outs = run(main())
    assert all(len(o) == 40 for o in outs)
    assert eng.allocator.available == 6
    assert eng.stats["preemptions"] >= 1


def test_engine_loop_restarts_after_step_crash():
    """A crashed step poisons in-flight requests but the engine serves new
    requests afterwards (loop restarts on next add_request)."""
    eng = tiny_engine()
    original = eng.step
    calls = {"n": 0}

    def crashing_step():
        calls["n"] += 1
        if calls["n"] == 1:
            raise RuntimeError("injected step failure")
        return original()

    eng.step = crashing_step

    async def first():
        seq = await eng.add_request([1, 2, 3], SamplingParams(
            temperature=0.0, max_tokens=4, ignore_eos=True))
        item = await seq.stream.get()
        return item

    item = run(first())
    assert "error" in item and item["finished"]

    async def second():
        toks = []
        async for it in eng.generate("ok", SamplingParams(
                temperature=0.0, max_tokens=4, ignore_eos=True)):
            toks.extend(it["token_ids"])
        return toks

    assert len(run(second())) == 4


def test_chunked_prefill_matches_unchunked():
    """Greedy outputs must be identical whether a long prompt prefills in
    one shot or in small chunks (paged attention over cached history)."""
    prompt = [(7 * i + 3) % 200 for i in range(90)]

    def gen(chunk):
        torch.manual_seed(1234)  # identical weights across engines
        eng = tiny_engine(prefill_chunk=chunk)

        async def go():
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=8, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go()), eng

    full, eng_full = gen(chunk=1024)   # single-shot prefill
    chunked, eng_chunk = gen(chunk=16)  # 90 tokens -> 6 chunks
    assert eng_chunk.stats["prefill_batches"] >= 6
    assert full == chunked
    assert eng_chunk.allocator.available == eng_chunk.allocator.num_blocks


def test_chunked_prefill_interleaves_decode():
    """While a long prompt prefills chunk by chunk, an already-running
    sequence keeps decoding between chunks."""
    torch.manual_seed(0)
    eng = tiny_engine(prefill_chunk=16)

    async def go():
        fast = await eng.add_request([1, 2, 3], SamplingParams(
            temperature=0.0, max_tokens=12, ignore_eos=True))
        slow = await eng.add_request(
            [(i * 3) % 200 for i in range(80)],
            SamplingParams(temperature=0.0, max_tokens=4, ignore_eos=True))
        done = {"fast": [], "slow": []}

        async def drain(seq, key):
            while True:
                item = await seq.stream.get()
                done[key].extend(item["token_ids"])
                if item["finished"]:
                    return

        await asyncio.gather(drain(fast, "fast"), drain(slow, "slow"))
        return done

    done = run(go())
    assert len(done["fast"]) == 12 and len(done["slow"]) == 4
    # chunked prefill happened AND decode steps ran during the same window
    assert eng.stats["prefill_batches"] >= 5
    assert eng.stats["decode_batches"] >= 11  # 12 tokens = prefill sample + 11 decodes


def test_client_disconnect_aborts_generation():
    """Cancelling the consumer mid-stream frees the sequence's KV pages and
    stops further generation for it."""
    eng = tiny_engine()

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=200,
                                ignore_eos=True)
        agen = eng.generate("long request", params)
        got = 0
        async for item in agen:
            got += len(item["token_ids"])
            if got >= 3:
                await agen.aclose()  # simulates client disconnect
                break
        # let the engine run a few more steps
        for _ in range(20):
            await asyncio.sleep(0.01)
            if not eng.running and not eng.waiting:
                break
        return got

    got = run(main())
    assert got >= 3
    assert eng.stats["aborts"] == 1
    assert not eng.running and not eng.waiting
    assert eng.allocator.available == eng.allocator.num_blocks
    assert eng.stats["generated_tokens"] < 50  # didn't run to max_tokens


def test_batched_chunked_prefill_multiple_long_prompts():
    """Several long prompts chunk through TOGETHER (one batched paged
    forward per step) and all generate correctly."""
    torch.manual_seed(5)
    eng = tiny_engine(prefill_chunk=32)
    prompts = [[(i * 7 + j * 3) % 200 for j in range(70 + 10 * i)]
               for i in range(3)]

    async def main():
        async def one(p):
            seq = await eng.add_request(list(p), SamplingParams(
                temperature=0.0, max_tokens=5, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return await asyncio.gather(*[one(p) for p in prompts])

    outs = run(main())
    assert all(len(o) == 5 for o in outs)
    assert eng.allocator.available == eng.allocator.num_blocks
    # single-shot engine agrees (chunking is semantically invisible)
    torch.manual_seed(5)
    eng2 = tiny_engine(prefill_chunk=4096)

    async def single(p):
        seq = await eng2.add_request(list(p), SamplingParams(
            temperature=0.0, max_tokens=5, ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    for p, o in zip(prompts, outs):
        assert run(single(p)) == o


def test_two_chunked_prompts_batch_and_match_solo():
    """Two long prompts chunk-prefilling concurrently share batched
    paged-prefill steps, and each still gets the exact greedy output it
    would get alone."""
    p1 = [(7 * i + 3) % 200 for i in range(70)]
    p2 = [(11 * i + 5) % 200 for i in range(70)]

    def solo(prompt):
        torch.manual_seed(77)
        eng = tiny_engine(prefill_chunk=64)

        async def go():
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=6, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go())

    want1, want2 = solo(p1), solo(p2)

    torch.manual_seed(77)
    eng = tiny_engine(prefill_chunk=64)

    async def both():
        out = {}

        async def one(key, prompt):
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=6, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    break
            out[key] = toks

        await asyncio.gather(one("a", p1), one("b", p2))
        return out

    out = run(both())
    assert out["a"] == want1 and out["b"] == want2
    # budget packs across prompt boundaries: 140 tokens / 64-token chunks
    # = 3 batched paged-prefill calls (serial, unpacked would take 4)
    assert eng.stats["prefill_batches"] == 3


def test_embeddings_normalized_and_batch_invariant():
    """v1/embeddings path: unit-norm vectors, identical whether texts are
    embedded together or one at a time (padding must not leak)."""
    torch.manual_seed(3)
    eng = tiny_engine()
    texts = ["hello world", "a much longer sentence for the second row"]

    async def both():
        return await eng.embed_batch(texts)

    async def solo(t):
        return (await eng.embed_batch([t]))[0]

    vb = run(both())
    v0 = run(solo(texts[0]))
    v1 = run(solo(texts[1]))
    for v in (vb[0], vb[1], v0, v1):
        assert abs(sum(x * x for x in v) - 1.0) < 1e-4
    assert max(abs(a - b) for a, b in zip(vb[0], v0)) < 1e-4
    assert max(abs(a - b) for a, b in zip(vb[1], v1)) < 1e-4
    # different texts -> different embeddings
    assert max(abs(a - b) for a, b in zip(vb[0], vb[1])) > 1e-3


def test_embeddings_concurrent_with_generation():
    """embed_batch runs off-loop under the engine's exec lock: concurrent
    generation + embedding requests both complete, embeddings match the
    solo result (no interleaving with step()'s forward), and the event
    loop is never blocked by the dense embed forward."""
    torch.manual_seed(5)
    eng = tiny_engine()
    text = "concurrent embedding probe"

    async def mixed():
        params = SamplingParams(temperature=0.0, max_tokens=24,
                                ignore_eos=True)

        async def gen(i):
            toks = []
            async for item in eng.generate("gen prompt %d" % i, params):
                toks.extend(item["token_ids"])
            return toks

        gens = [asyncio.ensure_future(gen(i)) for i in range(3)]
        # fire embeddings while decode steps are in flight
        embeds = [asyncio.ensure_future(eng.embed_batch([text]))
                  for _ in range(4)]
        toks = await asyncio.gather(*gens)
        vecs = await asyncio.gather(*embeds)
        return toks, vecs

    toks, vecs = run(mixed())
    assert all(len(t) == 24 for t in toks)

    async def solo():
        return await eng.embed_batch([text])

    want = run(solo())[0]
    for v in vecs:
        assert max(abs(a - b) for a, b in zip(v[0], want)) < 1e-4


def test_tokenize_detokenize_roundtrip():
    eng = tiny_engine()
    out = eng.openai_tokenize({"prompt": "round trip!"})
    assert out["count"] == len(out["tokens"]) > 0
    back = eng.openai_detokenize({"tokens": out["tokens"]})
    assert back["prompt"] == "round trip!"


def test_stopped_engine_rejects_requests():
    eng = tiny_engine()

    async def go():
        seq = await eng.add_request([1, 2, 3], SamplingParams(
            temperature=0.0, max_tokens=2, ignore_eos=True))
        while True:
            item = await seq.stream.get()
            if item["finished"]:
                break
        eng.stop()
        with pytest.raises(RuntimeError, match="stopped"):
            await eng.add_request([4, 5], SamplingParams(max_tokens=2))

    run(go())
    assert eng.model is None


def test_engine_loads_safetensors_weights(tmp_path):
    """cfg.weights: a real safetensors checkpoint loads (and TP-shards)
    into the engine's model instead of random init."""
    safetensors = pytest.importorskip("safetensors.torch")

    from clearml_serving_amd.models.llama import PRESETS, LlamaForCausalLM

    torch.manual_seed(99)
    src_model = LlamaForCausalLM(PRESETS["llama-tiny"])
    path = tmp_path / "model.safetensors"
    safetensors.save_file(
        {k: v.clone() for k, v in src_model.state_dict().items()}, str(path))

    eng = tiny_engine(weights=str(path))
    got = eng.model.state_dict()
    for key in ("embed.weight", "layers.0.qkv.weight", "lm_head.weight"):
        assert torch.allclose(got[key].float(),
                              src_model.state_dict()[key].float(),
                              atol=1e-2), key

    async def go():
        toks = []
        async for it in eng.generate("weights ok", SamplingParams(
                temperature=0.0, max_tokens=3, ignore_eos=True)):
            toks.extend(it["token_ids"])
        return toks

    assert len(run(go())) == 3


def test_config_from_model_folder(tmp_path):
    """A model FOLDER (card.json + model.safetensors [+ tokenizer.json])
    resolves weights and settings like the reference's model-dir layout."""
    import json as _json

    from clearml_serving_amd.engines.llm.engine import LlmEngineConfig

    d = tmp_path / "model_dir"
    d.mkdir()
    (d / "card.json").write_text(_json.dumps(
        {"preset": "llama-tiny", "max_model_len": 77}))
    (d / "model.safetensors").write_bytes(b"\0" * 8)
    cfg = LlmEngineConfig.from_aux(str(d), {"block_size": 8})
    assert cfg.preset == "llama-tiny"
    assert cfg.max_model_len == 77
    assert cfg.block_size == 8          # aux overrides layer on top
    assert cfg.weights == str(d / "model.safetensors")


def test_engine_error_surfaces_in_responses():
    """A step crash must surface as an error to BOTH response styles --
    a raised RuntimeError for non-streaming, an error SSE chunk for
    streaming -- never as silently truncated output."""
    eng = tiny_engine()

    def boom():
        raise ValueError("injected kernel failure")

    eng.step = boom

    async def collect():
        with pytest.raises(RuntimeError, match="injected kernel failure"):
            await eng._collect("x", SamplingParams(max_tokens=4,
                                                   ignore_eos=True))

    run(collect())

    eng2 = tiny_engine()
    eng2.step = boom

    async def stream():
        resp = eng2._sse_stream("x", SamplingParams(max_tokens=4,
                                                    ignore_eos=True),
                                "rid", "m", chat=True)
        chunks = []
        async for c in resp.body_iterator:
            chunks.append(c)
        return chunks

    chunks = run(stream())
    assert any("engine_error" in c for c in chunks), chunks


def test_sampling_params_validation_rejects_bad_values():
    """Malformed sampling params 422 at request time instead of crashing the
    shared step() for every in-flight sequence."""
    for bad in ({"temperature": -1}, {"top_k": -1}, {"top_p": 0.0},
                {"top_p": 1.5}, {"max_tokens": 0},
                {"temperature": "hot"}, {"stop": [1, 2]},
                {"stop_token_ids": ["x"]}):
        with pytest.raises(ValueError):
            SamplingParams.from_request(bad)
    # valid edge values pass
    p = SamplingParams.from_request(
        {"temperature": 0, "top_k": 0, "top_p": 1.0, "max_tokens": 1,
         "stop": "END", "stop_token_ids": [7]})
    assert p.temperature == 0.0 and p.stop == ["END"] \
        and p.stop_token_ids == [7]


def test_stop_token_ids_finish_generation():
    eng = tiny_engine()

    async def gen():
        # greedy output is deterministic: find its 3rd token, then re-run
        # with that token as a stop id
        params = SamplingParams(temperature=0.0, max_tokens=8,
                                ignore_eos=True)
        toks = []
        async for item in eng.generate("stop test", params):
            toks.extend(item["token_ids"])
        stop_at = toks[2]
        params2 = SamplingParams(temperature=0.0, max_tokens=8,
                                 ignore_eos=True, stop_token_ids=[stop_at])
        out, reason = [], None
        async for item in eng.generate("stop test", params2):
            out.extend(item["token_ids"])
            reason = item.get("finish_reason") or reason
        return toks, out, reason

    toks, out, reason = run(gen())
    assert out == toks[:3]
    assert reason == "stop"


def test_stop_strings_finish_and_truncate():
    eng = tiny_engine()

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=10,
                                ignore_eos=True)
        full = await eng.generate_simple(
            {"prompt": "abc", "max_tokens": 10, "temperature": 0,
             "ignore_eos": True})
        # pick a substring from the middle of the deterministic output
        text = full["text"]
        assert len(text) >= 3
        stop = text[2]
        res = await eng.generate_simple(
            {"prompt": "abc", "max_tokens": 10, "temperature": 0,
             "ignore_eos": True, "stop": stop})
        return text, stop, res["text"]

    text, stop, stopped = run(main())
    assert stop not in stopped
    assert stopped == text[:text.find(stop)]


def test_abort_of_waiting_seq_does_not_drop_neighbors():
    """Regression (advisor): abort() used to mutate waiting[] from the event
    loop while _admit() popped it from the step thread, which could admit-
    then-drop a DIFFERENT queued sequence. Aborts now drain at the top of
    step() on the step thread."""
    eng = tiny_engine()

    async def main():
        paramsA = SamplingParams(temperature=0.0, max_tokens=4,
                                 ignore_eos=True)
        paramsB = SamplingParams(temperature=0.0, max_tokens=4,
                                 ignore_eos=True)
        # enqueue two, abort the first before the engine runs a step
        seq_a = await eng.add_request([1, 2, 3], paramsA)
        seq_b = await eng.add_request([4, 5, 6], paramsB)
        eng.abort(seq_a)
        out_b = []
        while True:
            item = await asyncio.wait_for(seq_b.stream.get(), timeout=10)
            out_b.extend(item["token_ids"])
            if item.get("finished"):
                break
        return seq_a, out_b

    seq_a, out_b = run(main())
    assert len(out_b) == 4          # b generated fully
    assert seq_a.finished and not seq_a.blocks
    assert not eng.waiting and not eng.running
    assert eng.allocator.available == eng.allocator.num_blocks


def test_llama3_chat_template_detected(tmp_path):
    """A tokenizer carrying llama-3 header-id tokens switches chat formatting
    to the real llama-3 template (VERDICT: _chat_prompt was synthetic)."""
    eng = tiny_engine()

    class FakeL3Tok:
        is_llama3 = True
        eos_id = 0
        vocab_size = 300

        def encode(self, t):
            return [1]

        def decode(self, ids):
            return ""

    eng.tokenizer = FakeL3Tok()
    p = eng._chat_prompt([
        {"role": "system", "content": "be brief"},
        {"role": "user", "content": "hi"},
    ])
    assert p.startswith("<|begin_of_text|>")
    assert "<|start_header_id|>system<|end_header_id|>\n\nbe brief<|eot_id|>" in p
    assert p.endswith("<|start_header_id|>assistant<|end_header_id|>\n\n")

    eng.tokenizer = SimpleTokenizer()
    p2 = eng._chat_prompt([{"role": "user", "content": "hi"}])
    assert p2.endswith("<|assistant|>\n")


def test_engine_cache_key_distinguishes_aux_only_endpoints(tmp_path):
    """Two aux-config-only LLM endpoints (no model_id) with different configs
    must NOT share one cached engine (advisor: '__default__' collision)."""
    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest
    from clearml_serving_amd.schemas import ModelEndpoint

    saved_engines = dict(LlmPreprocessRequest._engines)
    saved_refs = dict(LlmPreprocessRequest._engine_refs)
    LlmPreprocessRequest._engines.clear()
    LlmPreprocessRequest._engine_refs.clear()
    try:
        ep1 = ModelEndpoint(
            engine_type="llm", serving_url="llm_a",
            auxiliary_cfg={"preset": "llama-tiny", "num_kv_blocks": 64,
                           "device": "cpu", "max_model_len": 128})
        ep2 = ModelEndpoint(
            engine_type="llm", serving_url="llm_b",
            auxiliary_cfg={"preset": "llama-tiny", "num_kv_blocks": 32,
                           "device": "cpu", "max_model_len": 64})
        r1 = LlmPreprocessRequest(ep1)
        r2 = LlmPreprocessRequest(ep2)
        assert r1._engine is not r2._engine
        assert r1._engine.cfg.num_kv_blocks == 64
        assert r2._engine.cfg.num_kv_blocks == 32
        # identical config DOES share the engine
        ep3 = ModelEndpoint(
            engine_type="llm", serving_url="llm_c",
            auxiliary_cfg={"preset": "llama-tiny", "num_kv_blocks": 64,
                           "device": "cpu", "max_model_len": 128})
        r3 = LlmPreprocessRequest(ep3)
        assert r3._engine is r1._engine
        r1.shutdown(); r2.shutdown(); r3.shutdown()
    finally:
        LlmPreprocessRequest._engines.clear()
        LlmPreprocessRequest._engines.update(saved_engines)
        LlmPreprocessRequest._engine_refs.clear()
        LlmPreprocessRequest._engine_refs.update(saved_refs)


def test_chunked_prefill_fair_share():
    """Two long prompts chunking together each advance every step (the
    budget splits; round-1 FCFS packing starved the later prompt until the
    earlier finished)."""
    eng = tiny_engine(prefill_chunk=32, max_prefill_tokens=32,
                      max_num_seqs=8)

    progress = []
    orig = eng._exec_chunk

    def spy(plan):
        progress.append([len(t) for t in plan["tokens"]])
        return orig(plan)

    eng._exec_chunk = spy

    async def main():
        pa = [(7 * i + 1) % 500 for i in range(96)]
        pb = [(11 * i + 3) % 500 for i in range(96)]
        params = SamplingParams(temperature=0.0, max_tokens=2,
                                ignore_eos=True)
        sa = await eng.add_request(pa, params)
        sb = await eng.add_request(pb, params)
        for s in (sa, sb):
            while True:
                item = await asyncio.wait_for(s.stream.get(), timeout=30)
                if item.get("finished"):
                    break

    run(main())
    # steps where both prompts were mid-prefill must split the budget
    both = [p for p in progress if len(p) == 2]
    assert both, progress
    for chunks in both:
        assert chunks[0] == chunks[1] == 16, (chunks, progress)


def test_fp8_kv_cache_engine_generates():
    """kv_dtype="fp8": uint8 e4m3 caches + per-token scales; generation is
    deterministic and the paged decode path runs on the quantized cache."""
    eng = tiny_engine(kv_dtype="fp8")
    assert len(eng.kv_caches[0]) == 4
    assert eng.kv_caches[0][0].dtype == torch.uint8
    assert eng.kv_caches[0][2].dtype == torch.float32

    async def gen():
        out = []
        async for item in eng.generate(
                "hello fp8", SamplingParams(temperature=0.0, max_tokens=8,
                                            ignore_eos=True)):
            out.extend(item["token_ids"])
        return out

    a = run(gen())
    b = run(gen())
    assert len(a) == 8 and a == b
    # scales were actually written (not all still at init value 1.0)
    assert (eng.kv_caches[0][2] != 1.0).any()


def test_fp8_kv_close_to_bf16_kv_generation():
    """Greedy tokens from the fp8-KV engine track the full-precision-KV
    engine on a short continuation (e4m3 + per-token-per-head scales keep
    logits within quantization noise)."""
    eng16 = tiny_engine()
    eng8 = tiny_engine(kv_dtype="fp8")
    prompt = [3, 7, 11, 19, 23, 29, 31]

    async def gen(eng):
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=4,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    t16 = run(gen(eng16))
    t8 = run(gen(eng8))
    assert len(t8) == 4
    # the first decoded token attends to a freshly-quantized prompt cache:
    # argmax agreement here is the direct numerics check; later tokens may
    # legitimately diverge once sequences differ
    assert t8[0] == t16[0]


def test_fp8_kv_chunked_prefill_generates():
    """Chunked prefill reads the fp8 cache through the paged-prefill path."""
    eng = tiny_engine(kv_dtype="fp8", prefill_chunk=16)
    prompt = list(range(2, 50))

    async def gen():
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=4,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    assert len(run(gen())) == 4


def test_qwen2_style_engine_generation():
    """qkv_bias (Qwen2 family) through the full engine: prefill + decode
    paths apply the bias (the skinny projection route is bias-free, the
    layer adds it) and greedy generation is deterministic."""
    eng = tiny_engine(overrides={"qkv_bias": True})
    assert eng.model.layers[0].qkv.bias is not None

    async def gen():
        out = []
        async for item in eng.generate(
                "qwen probe", SamplingParams(temperature=0.0, max_tokens=8,
                                             ignore_eos=True)):
            out.extend(item["token_ids"])
        return out

    a = run(gen())
    b = run(gen())
    assert len(a) == 8 and a == b

    # the bias must actually influence the logits: zeroing it changes them
    import torch as _torch

    with _torch.no_grad():
        for layer in eng.model.layers:
            layer.qkv.bias.zero_()
    c = run(gen())
    assert c != a


def test_chatml_template_for_qwen2_tokenizers():
    """Tokenizers carrying <|im_start|>/<|im_end|> get the ChatML chat
    template (and their eos is <|im_end|>)."""
    eng = tiny_engine()

    class FakeChatml:
        is_chatml = True

    eng.tokenizer.is_chatml = True  # simulate a Qwen2 tokenizer
    prompt = eng._chat_prompt([
        {"role": "system", "content": "be brief"},
        {"role": "user", "content": "hi"},
    ])
    assert prompt == ("<|im_start|>system\nbe brief<|im_end|>\n"
                      "<|im_start|>user\nhi<|im_end|>\n"
                      "<|im_start|>assistant\n")


# ------------------------------------------------------------------ #
# speculative decoding (ngram prompt-lookup)
# ------------------------------------------------------------------ #
def spec_engine(**spec_kw):
    return tiny_engine(speculative={"method": "ngram", "num_spec_tokens": 4,
                                    "ngram": 2, **spec_kw})


def _greedy(eng, prompt, max_tokens=24):
    async def gen():
        out = []
        async for item in eng.generate(
                prompt, SamplingParams(temperature=0.0,
                                       max_tokens=max_tokens,
                                       ignore_eos=True)):
            out.extend(item["token_ids"])
        return out

    return run(gen())


def test_ngram_proposer():
    eng = spec_engine()
    from clearml_serving_amd.engines.llm.engine import Sequence

    s = Sequence("t", [1, 2, 3, 4, 1, 2], SamplingParams())
    # trailing 2-gram (1,2) last occurred at the start; the 4 tokens that
    # followed it are the proposal
    assert eng._ngram_propose(s, 4) == [3, 4, 1, 2]
    assert eng._ngram_propose(s, 2) == [3, 4]
    # no earlier occurrence -> no proposal
    s2 = Sequence("t", [5, 6, 7, 8], SamplingParams())
    assert eng._ngram_propose(s2, 4) == []


def test_spec_decode_exactly_matches_plain_greedy():
    """The defining property: with ngram speculation on, greedy output is
    IDENTICAL to plain decode -- across prompts that force both heavy
    acceptance (repetitive) and heavy rejection (random model output)."""
    torch.manual_seed(0)
    plain = tiny_engine()
    torch.manual_seed(0)
    spec = spec_engine()

    prompts = [
        "abcabcabcabcabc",          # repetitive: proposals accepted
        "the quick brown fox",      # generic
        "zq!7#",                    # short, no structure
    ]
    for p in prompts:
        a = _greedy(plain, p)
        b = _greedy(spec, p)
        assert a == b, (p, a, b)
    # speculation actually engaged (random-init models emit repetitive
    # token loops, so ngram proposals fire and some get accepted)
    assert spec.stats["spec_proposed"] > 0
    assert spec.stats["spec_accepted"] > 0
    # and it saved forwards: fewer decode batches than generated tokens
    assert spec.stats["decode_batches"] < plain.stats["decode_batches"]


def test_spec_decode_kv_cache_stays_correct():
    """After speculative steps (including rejected proposals), the KV cache
    must equal a teacher-forced prefill of the final sequence -- greedy
    continuation from the cache matches single-shot logits."""
    eng = spec_engine()
    prompt = [3, 7, 11, 3, 7, 11, 3, 7]

    async def gen():
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=10,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    generated = run(gen())
    assert len(generated) == 10
    model = eng.model
    full = prompt + generated
    t = len(full)
    tokens = torch.tensor(full, dtype=torch.long)
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        logits = model(tokens, positions, kv_caches=None, attn_ctx=attn_ctx)
    for step in range(10):
        pos = len(prompt) + step - 1
        assert generated[step] == int(logits[pos].argmax()), step


def test_spec_decode_mixed_with_sampled_requests():
    """Greedy sequences speculate; sampled sequences run plain decode in
    the same engine -- all complete, blocks all return."""
    eng = spec_engine()

    async def main():
        async def one(i, temp):
            toks = []
            async for item in eng.generate(
                    "req %d abcabcabc" % i,
                    SamplingParams(temperature=temp, max_tokens=8,
                                   ignore_eos=True)):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(
            one(0, 0.0), one(1, 0.8), one(2, 0.0), one(3, 1.2))

    outs = run(main())
    assert all(len(o) == 8 for o in outs)
    assert eng.allocator.available == eng.allocator.num_blocks


def test_spec_decode_respects_max_tokens_and_eos():
    """Accepted speculative tokens never overshoot max_tokens, and a
    mid-proposal eos stops emission."""
    eng = spec_engine(num_spec_tokens=8)
    out = _greedy(eng, "abcabcabcabc", max_tokens=5)
    assert len(out) == 5

    # force an eos mid-stream: stop on the first generated token id
    first = out[0]

    async def gen():
        toks, reason = [], None
        async for item in eng.generate(
                "abcabcabcabc",
                SamplingParams(temperature=0.0, max_tokens=20,
                               stop_token_ids=[first])):
            toks.extend(item["token_ids"])
            reason = item.get("finish_reason") or reason
        return toks, reason

    toks, reason = run(gen())
    assert toks[-1] == first and reason == "stop"
    assert len(toks) <= 20


# ------------------------------------------------------------------ #
# OpenAI sampling controls: n, seed, penalties
# ------------------------------------------------------------------ #
def test_openai_n_choices():
    eng = tiny_engine()
    resp = run(eng.openai_chat_completions(
        {"messages": [{"role": "user", "content": "hi"}], "max_tokens": 4,
         "temperature": 0.0, "ignore_eos": True, "n": 3}, "m"))
    assert len(resp["choices"]) == 3
    assert [c["index"] for c in resp["choices"]] == [0, 1, 2]
    # greedy: all n choices identical (OpenAI semantics)
    assert len({c["message"]["content"] for c in resp["choices"]}) == 1
    assert resp["usage"]["completion_tokens"] == 12

    with pytest.raises(ValueError):
        run(eng.openai_completions({"prompt": "x", "n": 0}, "m"))
    with pytest.raises(ValueError):
        run(eng.openai_completions(
            {"prompt": "x", "n": 2, "stream": True}, "m"))


def test_seed_reproducible_sampling():
    """Same request seed -> same sampled tokens; different seed differs
    (vLLM/OpenAI 'seed' semantics)."""
    eng = tiny_engine()

    def gen(seed):
        async def go():
            toks = []
            async for item in eng.generate("seed probe", SamplingParams(
                    temperature=1.0, max_tokens=12, ignore_eos=True,
                    seed=seed)):
                toks.extend(item["token_ids"])
            return toks

        return run(go())

    a, b, c = gen(1234), gen(1234), gen(99)
    assert a == b
    assert a != c


def test_frequency_penalty_reduces_repetition():
    """Random-init tiny models loop on a few tokens under greedy; a strong
    frequency penalty must strictly increase the distinct-token count."""
    eng = tiny_engine()

    def gen(**kw):
        async def go():
            toks = []
            async for item in eng.generate("rep probe", SamplingParams(
                    temperature=0.0, max_tokens=24, ignore_eos=True, **kw)):
                toks.extend(item["token_ids"])
            return toks

        return run(go())

    plain = gen()
    penalized = gen(frequency_penalty=2.0, presence_penalty=2.0)
    assert len(set(penalized)) > len(set(plain)), (plain, penalized)

    rep = gen(repetition_penalty=5.0)
    assert len(set(rep)) > len(set(plain))


def test_penalty_validation_and_tp_guard():
    eng = tiny_engine()
    with pytest.raises(ValueError):
        SamplingParams.from_request({"presence_penalty": 3.0})
    with pytest.raises(ValueError):
        SamplingParams.from_request({"frequency_penalty": -2.5})
    with pytest.raises(ValueError):
        SamplingParams.from_request({"repetition_penalty": 0.0})
    # max_completion_tokens is honored as the newer max_tokens name
    p = SamplingParams.from_request({"max_completion_tokens": 7})
    assert p.max_tokens == 7

    # TP guard: penalties refuse loudly instead of silently diverging
    eng.tp_size = 2
    with pytest.raises(ValueError):
        run(eng.add_request([1, 2], SamplingParams(presence_penalty=1.0)))
    eng.tp_size = 1


def test_logprobs_returned_and_consistent():
    """OpenAI logprobs: chosen-token logprob + top_logprobs per position;
    under greedy the chosen token IS the top-1 and its logprob matches a
    teacher-forced log_softmax of the model."""
    eng = tiny_engine()
    resp = run(eng.openai_completions(
        {"prompt": "lp probe", "max_tokens": 4, "temperature": 0.0,
         "ignore_eos": True, "logprobs": 3}, "m"))
    lp = resp["choices"][0]["logprobs"]
    assert len(lp["tokens"]) == 4
    assert len(lp["token_logprobs"]) == 4
    # completions format keys by decoded token STRING: the byte-fallback
    # tokenizer folds distinct ids to one string, so <= 3 after dedup
    assert all(1 <= len(t) <= 3 for t in lp["top_logprobs"])
    # greedy: chosen token has the max logprob of its top list
    for chosen, top in zip(lp["token_logprobs"], lp["top_logprobs"]):
        assert abs(chosen - max(top.values())) < 1e-5
        assert chosen <= 0.0

    # chat-style request: logprobs: true + top_logprobs
    resp = run(eng.openai_chat_completions(
        {"messages": [{"role": "user", "content": "hi"}], "max_tokens": 3,
         "temperature": 0.0, "ignore_eos": True, "logprobs": True,
         "top_logprobs": 2}, "m"))
    content = resp["choices"][0]["logprobs"]["content"]
    assert len(content) == 3
    assert all(len(e["top_logprobs"]) == 2 for e in content)
    assert all(e["logprob"] <= 0.0 for e in content)

    # validation
    with pytest.raises(ValueError):
        SamplingParams.from_request({"logprobs": 30})
    # TP guard
    eng.tp_size = 2
    with pytest.raises(ValueError):
        run(eng.add_request([1, 2], SamplingParams(logprobs=1)))
    eng.tp_size = 1


def test_logprobs_match_teacher_forced_reference():
    eng = tiny_engine()
    prompt = [5, 9, 13]
    import math as _math

    async def gen():
        seq = await eng.add_request(list(prompt), SamplingParams(
            temperature=0.0, max_tokens=4, ignore_eos=True, logprobs=0))
        items = []
        while True:
            item = await seq.stream.get()
            items.append(item)
            if item["finished"]:
                return items

    items = run(gen())
    generated = [t for it in items for t in it["token_ids"]]
    full = prompt + generated
    t = len(full)
    tokens = torch.tensor(full, dtype=torch.long)
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        logits = eng.model(tokens, positions, kv_caches=None,
                           attn_ctx=attn_ctx)
    for step, item in enumerate(items):
        pos = len(prompt) + step - 1
        ref = float(torch.log_softmax(logits[pos].float(), -1)[
            generated[step]])
        assert abs(item["logprob"] - ref) < 5e-3, (step, item["logprob"], ref)


def test_decode_microbatch_pipeline_matches_plain():
    """model.forward_pipelined (TP comm/compute overlap schedule) must be
    numerically identical to the plain decode forward -- verified here
    single-process (works are None; pure math identity), world=2 gloo in
    tests/helpers/tp_check.py."""
    import os as _os

    eng = tiny_engine()

    # seed some sequences so the KV cache and block tables are real
    async def seed():
        params = SamplingParams(temperature=0.0, max_tokens=4,
                                ignore_eos=True)

        async def one(i):
            toks = []
            async for item in eng.generate("mb %d" % i, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one(i) for i in range(4)])

    run(seed())

    # craft a decode plan over 4 live-looking sequences
    from clearml_serving_amd.engines.llm.engine import Sequence

    seqs = []
    for i in range(4):
        s = Sequence("mb%d" % i, [3 + i, 7, 11 + i], SamplingParams(
            temperature=0.0, max_tokens=8, ignore_eos=True))
        s.blocks = eng.allocator.alloc(1)
        s.output_ids = [5 + i]
        s.prefilled = len(s.prompt_ids)
        seqs.append(s)
    plan = {
        "mode": "decode",
        "tokens": [s.output_ids[-1] for s in seqs],
        "positions": [len(s) - 1 for s in seqs],
        "slots": [eng._slot(s, len(s) - 1) for s in seqs],
        "seq_lens": [len(s) for s in seqs],
        "blocks": [list(s.blocks) for s in seqs],
    }
    plain = eng._exec_decode(plan)
    eng.tp_size = 2  # force the microbatch route (works stay None on CPU
    _os.environ["CMLS_TP_MICROBATCH"] = "1"  # single-process: world<=1)
    try:
        mb = eng._exec_decode(plan)
    finally:
        eng.tp_size = 1
        del _os.environ["CMLS_TP_MICROBATCH"]
    for s in seqs:
        eng.allocator.free(s.blocks)
    torch.testing.assert_close(mb, plain, atol=1e-5, rtol=1e-5)


def test_min_tokens_and_echo():
    """min_tokens bans eos/stop ids from sampling until the floor (vLLM
    semantics: masked, not post-filtered); echo prefixes the prompt."""
    eng = tiny_engine()

    # find what greedy emits first, then use it as a stop id with a floor
    first = run(eng.generate_simple(
        {"prompt": "mt probe", "max_tokens": 1, "temperature": 0.0,
         "ignore_eos": True}))

    async def gen(min_tokens):
        seq = await eng.add_request(
            eng.tokenizer.encode("mt probe"),
            SamplingParams(temperature=0.0, max_tokens=12,
                           min_tokens=min_tokens,
                           stop_token_ids=[]))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    toks = run(gen(0))
    stop_id = toks[0]

    async def gen_stop(min_tokens):
        seq = await eng.add_request(
            eng.tokenizer.encode("mt probe"),
            SamplingParams(temperature=0.0, max_tokens=12,
                           min_tokens=min_tokens,
                           stop_token_ids=[stop_id]))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    # without the floor the first token stops generation immediately
    assert len(run(gen_stop(0))) == 1
    # with the floor, the banned id cannot be sampled before 6 tokens
    out = run(gen_stop(6))
    assert len(out) >= 6
    assert stop_id not in out[:6]

    with pytest.raises(ValueError):
        SamplingParams.from_request({"min_tokens": 200, "max_tokens": 10})

    resp = run(eng.openai_completions(
        {"prompt": "hello", "max_tokens": 2, "temperature": 0.0,
         "ignore_eos": True, "echo": True}, "m"))
    assert resp["choices"][0]["text"].startswith("hello")


def test_spec_decode_backoff_on_rejections():
    """Sequences whose proposals keep getting rejected stop paying the
    multi-token verify: after 4 consecutive misses the engine proposes
    only on the periodic retry (every 32 tokens)."""
    eng = spec_engine()
    from clearml_serving_amd.engines.llm.engine import Sequence

    s = Sequence("bo", [1, 2, 1, 2], SamplingParams(temperature=0.0))
    s.generated = 5  # not on a retry boundary
    s.spec_misses = 4
    # the proposer would fire (repeating 2-gram)...
    assert eng._ngram_propose(s, 4) != []
    # ...but the backoff decision (as _decode_spec computes it) caps it
    cap = 4
    if s.spec_misses >= 4 and s.generated % 32 != 0:
        cap = 0
    assert cap == 0

    # end-to-end: exactness holds regardless of backoff state
    torch.manual_seed(3)
    e1 = spec_engine()
    torch.manual_seed(3)
    e2 = tiny_engine()
    assert _greedy(e1, "zq!7# unstructured", max_tokens=48) == \
        _greedy(e2, "zq!7# unstructured", max_tokens=48)


def test_spec_decode_with_fp8_kv_cache():
    """Feature interaction: ngram speculation over the fp8 KV cache --
    the multi-token verify reads quantized pages; output must equal the
    plain fp8-KV decode path exactly (same quantization noise)."""
    torch.manual_seed(9)
    plain = tiny_engine(kv_dtype="fp8")
    torch.manual_seed(9)
    spec = tiny_engine(kv_dtype="fp8",
                       speculative={"method": "ngram",
                                    "num_spec_tokens": 4, "ngram": 2})
    for p in ("abcabcabcabc", "mixed 123 text"):
        assert _greedy(plain, p, 20) == _greedy(spec, p, 20)
    assert spec.stats["spec_proposed"] > 0


def test_echo_with_prompt_logprobs():
    """completions echo+logprobs: the logprobs arrays cover prompt AND
    completion tokens; the first prompt token's logprob is null; prompt
    entries match a teacher-forced log_softmax."""
    eng = tiny_engine()
    prompt = "prompt lp"
    resp = run(eng.openai_completions(
        {"prompt": prompt, "max_tokens": 3, "temperature": 0.0,
         "ignore_eos": True, "echo": True, "logprobs": 2}, "m"))
    choice = resp["choices"][0]
    assert choice["text"].startswith(prompt)
    lp = choice["logprobs"]
    ids = eng.tokenizer.encode(prompt)
    assert len(lp["tokens"]) == len(ids) + 3
    assert lp["token_logprobs"][0] is None
    assert all(v is not None for v in lp["token_logprobs"][1:])

    # teacher-forced reference over the prompt
    n = len(ids)
    tokens = torch.tensor(ids, dtype=torch.long)
    positions = torch.arange(n, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": n,
                "seq_lens": torch.tensor([n], dtype=torch.int32),
                "slot_mapping": torch.full((n,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        logits = eng.model(tokens, positions, kv_caches=None,
                           attn_ctx=attn_ctx)
    lsm = torch.log_softmax(logits.float(), -1)
    for i in range(1, n):
        assert abs(lp["token_logprobs"][i] - float(lsm[i - 1][ids[i]])) \
            < 5e-3, i


# ------------------------------------------------------------------ #
# GPT-2 family through the same engine
# ------------------------------------------------------------------ #
def gpt2_engine(**kw):
    cfg = LlmEngineConfig(preset="gpt2-tiny", num_kv_blocks=128,
                          block_size=16, max_model_len=256,
                          device="cpu", **kw)
    eng = LlmEngine(cfg)
    eng.start()
    return eng


def test_gpt2_engine_decode_matches_prefill():
    """GPT-2 (pre-LN, learned positions) through the paged-KV engine:
    incremental decode equals teacher-forced prefill argmax."""
    eng = gpt2_engine()
    from clearml_serving_amd.models.gpt2 import GPT2ForCausalLM

    assert isinstance(eng.model, GPT2ForCausalLM)
    prompt = [3, 7, 11, 19, 23]

    async def gen():
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=6,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    generated = run(gen())
    assert len(generated) == 6
    full = prompt + generated
    t = len(full)
    tokens = torch.tensor(full, dtype=torch.long)
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        logits = eng.model(tokens, positions, kv_caches=None,
                           attn_ctx=attn_ctx)
    for step in range(6):
        pos = len(prompt) + step - 1
        assert generated[step] == int(logits[pos].argmax()), step


def test_gpt2_chunked_prefill_and_speculation():
    """GPT-2 exercises the chunked paged-prefill path and ngram
    speculation identically to llama (shared engine interface)."""
    def gen(eng, prompt, n=10):
        async def go():
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=n, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go())

    long_prompt = [(3 * i + 1) % 500 for i in range(120)]
    torch.manual_seed(2)
    unchunked = gpt2_engine(prefill_chunk=2048)
    torch.manual_seed(2)
    chunked = gpt2_engine(prefill_chunk=48)
    a = gen(unchunked, long_prompt)
    b = gen(chunked, long_prompt)
    assert a == b
    assert chunked.stats["prefill_batches"] >= 3

    torch.manual_seed(2)
    spec = gpt2_engine(prefill_chunk=2048,
                       speculative={"method": "ngram",
                                    "num_spec_tokens": 4, "ngram": 2})
    assert gen(spec, long_prompt) == a


def test_gpt2_fp8_refused_loudly():
    """quantize_llama_fp8 finds no llama projections on GPT-2: the engine
    raises instead of silently serving unquantized (checked via the
    quantizer hook directly -- the GPU check fires first on CPU)."""
    from clearml_serving_amd.models.quant import quantize_llama_fp8

    eng = gpt2_engine()
    assert quantize_llama_fp8(eng.model) == 0


# ------------------------------------------------------------------ #
# automatic prefix caching
# ------------------------------------------------------------------ #
def pc_engine(**kw):
    return tiny_engine(enable_prefix_caching=True, **kw)


def test_prefix_cache_exact_and_hits():
    """Identical prompts: the second request reuses the first's full
    prompt blocks (hit stats move) and produces IDENTICAL greedy output;
    divergent-suffix prompts share only the common full blocks."""
    torch.manual_seed(4)
    plain = tiny_engine()
    torch.manual_seed(4)
    pc = pc_engine()
    prompt = "shared prefix " * 4  # > 2 full blocks of byte tokens

    a1 = _greedy(plain, prompt, 12)
    b1 = _greedy(pc, prompt, 12)
    assert a1 == b1
    assert pc.allocator.hit_tokens == 0  # first request: nothing cached

    a2 = _greedy(plain, prompt, 12)
    b2 = _greedy(pc, prompt, 12)
    assert a2 == b2 == a1
    assert pc.allocator.hit_tokens >= 16  # >= one full block reused
    assert pc.stats["prefix_cache_hit_tokens"] >= 16

    # divergent suffix: still shares the common prefix blocks
    before = pc.allocator.hit_tokens
    c = _greedy(pc, prompt + "DIFFERENT TAIL", 12)
    assert pc.allocator.hit_tokens > before
    d = _greedy(plain, prompt + "DIFFERENT TAIL", 12)
    assert c == d


def test_prefix_cache_under_eviction_pressure():
    """A small pool forces evictions: requests keep completing correctly
    and every block is accounted for at the end."""
    torch.manual_seed(5)
    pc = pc_engine(num_kv_blocks=24)
    torch.manual_seed(5)
    plain = tiny_engine(num_kv_blocks=24)
    prompts = ["aaaa " * 8, "bbbb " * 8, "cccc " * 8, "aaaa " * 8,
               "dddd " * 8, "aaaa " * 8]
    for p in prompts:
        assert _greedy(pc, p, 8) == _greedy(plain, p, 8)
    # full accounting: every block is free or evictable
    assert pc.allocator.available == pc.allocator.num_blocks


def test_prefix_cache_concurrent_and_aborted():
    eng = pc_engine()

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=8,
                                ignore_eos=True)

        async def one(i):
            toks = []
            async for item in eng.generate("common root " * 3 + str(i),
                                           params):
                toks.extend(item["token_ids"])
            return toks

        async def aborted():
            task = asyncio.ensure_future(eng.generate_simple(
                {"prompt": "common root " * 3 + "x", "max_tokens": 60,
                 "temperature": 0.0, "ignore_eos": True}))
            await asyncio.sleep(0.05)
            task.cancel()
            try:
                await task
            except asyncio.CancelledError:
                pass

        outs = await asyncio.gather(*[one(i) for i in range(4)], aborted())
        return outs[:4]

    outs = run(main())
    assert all(len(o) == 8 for o in outs)
    # drain cleanup
    import time as _time

    for _ in range(100):
        if eng.allocator.available == eng.allocator.num_blocks:
            break
        _time.sleep(0.02)
    assert eng.allocator.available == eng.allocator.num_blocks


def test_prefix_cache_with_chunked_prefill_and_spec():
    """Interactions: a cached prefix admits with prefilled>0 and continues
    through the chunked path; speculation still produces exact output."""
    torch.manual_seed(6)
    pc = pc_engine(prefill_chunk=32,
                   speculative={"method": "ngram", "num_spec_tokens": 4,
                                "ngram": 2})
    torch.manual_seed(6)
    plain = tiny_engine(prefill_chunk=32)
    prompt = "abc abc abc " * 6  # long enough to chunk
    a = _greedy(plain, prompt, 16)
    assert _greedy(pc, prompt, 16) == a
    assert _greedy(pc, prompt, 16) == a  # cache-hit pass
    assert pc.allocator.hit_tokens > 0


def test_prefix_cache_never_full_prompt():
    """ncached always leaves >= 1 token to prefill (the last token's
    logits feed first-token sampling)."""
    from clearml_serving_amd.engines.llm.engine import PrefixCacheAllocator

    a = PrefixCacheAllocator(16, 4)
    ids = list(range(8))  # exactly 2 full blocks
    blocks = a.alloc(2)
    h = a.register_block(0, ids[:4], blocks[0])
    a.register_block(h, ids[4:], blocks[1])
    got, n, _ = a.match(list(ids))
    assert len(got) == 1 and n == 4  # NOT both blocks: last token prefills
    a.free(got)
    got2, n2, _ = a.match(ids + [99])  # longer prompt: both reusable
    assert len(got2) == 2 and n2 == 8


def test_prefix_cache_covers_generated_tokens():
    """Multi-turn pattern: turn 2's prompt = turn 1's prompt + the
    assistant's GENERATED reply + new text. Generated full blocks were
    registered as they filled, so the turn-2 hit extends past the
    original prompt into the generated region -- and output stays exact
    vs the plain engine."""
    torch.manual_seed(11)
    pc = pc_engine()
    torch.manual_seed(11)
    plain = tiny_engine()

    turn1_ids = pc.tokenizer.encode("multi turn root " * 3)

    def gen_ids(eng, ids, n):
        async def go():
            seq = await eng.add_request(list(ids), SamplingParams(
                temperature=0.0, max_tokens=n, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go())

    out1_pc = gen_ids(pc, turn1_ids, 40)
    out1_pl = gen_ids(plain, turn1_ids, 40)
    assert out1_pc == out1_pl

    # turn 2: context = turn1 prompt + reply + follow-up
    follow = pc.tokenizer.encode(" and then?")
    turn2_ids = turn1_ids + out1_pc + follow
    before = pc.allocator.hit_tokens
    out2_pc = gen_ids(pc, turn2_ids, 8)
    out2_pl = gen_ids(plain, turn2_ids, 8)
    assert out2_pc == out2_pl
    hit = pc.allocator.hit_tokens - before
    bs = pc.cfg.block_size
    # the hit must reach INTO the generated region: more than the full
    # blocks of the original prompt alone
    assert hit > (len(turn1_ids) // bs) * bs, hit


def test_prefix_cache_with_fp8_kv():
    """Cached blocks carry their per-token fp8 scales with them: reuse
    over the e4m3 cache stays exact vs the plain fp8-KV engine."""
    torch.manual_seed(12)
    pc = pc_engine(kv_dtype="fp8")
    torch.manual_seed(12)
    plain = tiny_engine(kv_dtype="fp8")
    prompt = "fp8 cached prefix " * 4
    a = _greedy(plain, prompt, 12)
    assert _greedy(pc, prompt, 12) == a
    assert _greedy(pc, prompt, 12) == a
    assert pc.allocator.hit_tokens >= 16


def test_best_of_guard():
    eng = tiny_engine()
    with pytest.raises(ValueError):
        run(eng.openai_completions(
            {"prompt": "x", "n": 1, "best_of": 4}, "m"))
    # best_of == n passes through
    resp = run(eng.openai_completions(
        {"prompt": "x", "n": 2, "best_of": 2, "max_tokens": 2,
         "temperature": 0.0, "ignore_eos": True}, "m"))
    assert len(resp["choices"]) == 2


def test_suffix_fim_guard():
    eng = tiny_engine()
    with pytest.raises(ValueError):
        run(eng.openai_completions(
            {"prompt": "a", "suffix": "z", "max_tokens": 2}, "m"))


def test_streamed_text_deltas_reassemble_and_handle_split_utf8():
    """Streamed 'text' fields are cumulative-decode deltas: concatenated
    they equal the full decode, and a multi-byte character split across
    byte tokens streams intact (held back until complete) instead of as
    replacement chars."""
    eng = tiny_engine()

    async def gen(prompt_ids, n):
        seq = await eng.add_request(list(prompt_ids), SamplingParams(
            temperature=0.0, max_tokens=n, ignore_eos=True))
        texts, toks = [], []
        while True:
            item = await seq.stream.get()
            texts.append(item["text"])
            toks.extend(item["token_ids"])
            if item["finished"]:
                return texts, toks

    texts, toks = run(gen([5, 9, 13], 12))
    assert "".join(texts) == eng.tokenizer.decode(toks)

    # force a split multi-byte char through the emission path directly:
    # "é" = 0xC3 0xA9 -> byte-tokenizer ids 0xC3+1, 0xA9+1
    from clearml_serving_amd.engines.llm.engine import Sequence

    s = Sequence("utf", [1], SamplingParams(max_tokens=10, ignore_eos=True))
    eng._emit_tokens(s, [0xC3 + 1])
    first = s.stream.get_nowait()
    assert first["text"] == ""  # incomplete tail held back
    eng._emit_tokens(s, [0xA9 + 1])
    second = s.stream.get_nowait()
    assert second["text"] == "é"


def test_prefix_cache_with_preemption_pressure_randomized():
    """The deepest interaction: a small pool forces decode-growth
    preemption WHILE cached blocks are being shared/evicted. Outputs must
    match the plain engine for every request, and accounting must balance
    -- across randomized overlapping-prefix workloads."""
    import random as _random

    total_preempt = 0
    total_hits = 0
    for trial in range(4):
        rng = _random.Random(100 + trial)
        torch.manual_seed(20 + trial)
        pc = pc_engine(num_kv_blocks=18, max_num_seqs=6)
        torch.manual_seed(20 + trial)
        plain = tiny_engine(num_kv_blocks=18, max_num_seqs=6)

        roots = ["root one ", "root two "]
        prompts = []
        for i in range(8):
            root = roots[rng.randrange(2)] * rng.randint(2, 4)
            prompts.append(root + "tail{}".format(rng.randrange(3)))
        n_new = rng.randint(6, 20)

        def gen_all(eng):
            async def main():
                params = SamplingParams(temperature=0.0, max_tokens=n_new,
                                        ignore_eos=True)

                async def one(p):
                    toks = []
                    async for item in eng.generate(p, params):
                        toks.extend(item["token_ids"])
                    return toks

                return await asyncio.gather(*[one(p) for p in prompts])

            return run(main())

        a = gen_all(plain)
        b = gen_all(pc)
        assert a == b, (trial, a, b)
        assert pc.allocator.available == pc.allocator.num_blocks, trial
        total_preempt += pc.stats["preemptions"]
        total_hits += pc.allocator.hit_tokens
    # the workload genuinely exercised the pressure paths (probed: every
    # seeded config preempts at least once and shares blocks)
    assert total_preempt >= 1
    assert total_hits > 0
