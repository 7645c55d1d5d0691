"""OpenAI-compatible routes through the full HTTP app with the native LLM
engine (llama-tiny on CPU)."""

import json

import pytest
from fastapi.testclient import TestClient

from clearml_serving_amd.schemas import ModelEndpoint
from clearml_serving_amd.serving.app import create_app


@pytest.fixture()
def llm_client(processor, store, tmp_path):
    # reset the per-process engine singleton so each test run is isolated
    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}

    card = tmp_path / "card.json"
    card.write_text(json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu",
    }))
    rec = store.register_model(name="llama-tiny", project="p", path=str(card))
    processor.add_endpoint(ModelEndpoint(
        engine_type="llm", serving_url="test_llm", model_id=rec.model_id))
    processor.serialize()
    app = create_app(processor=processor, poll_frequency_sec=3600)
    with TestClient(app) as c:
        yield c
    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}


def test_chat_completion_route(llm_client):
    r = llm_client.post("/serve/openai/v1/chat/completions", json={
        "model": "test_llm", "max_tokens": 4, "temperature": 0.0,
        "ignore_eos": True,
        "messages": [{"role": "user", "content": "hello"}],
    })
    assert r.status_code == 200, r.text
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["usage"]["completion_tokens"] == 4


def test_completion_route(llm_client):
    r = llm_client.post("/serve/openai/v1/completions", json={
        "model": "test_llm", "prompt": "abc", "max_tokens": 3,
        "temperature": 0.0, "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["object"] == "text_completion"


def test_streaming_sse(llm_client):
    with llm_client.stream("POST", "/serve/openai/v1/chat/completions", json={
        "model": "test_llm", "max_tokens": 3, "temperature": 0.0,
        "ignore_eos": True, "stream": True,
        "messages": [{"role": "user", "content": "hi"}],
    }) as r:
        assert r.status_code == 200
        chunks = list(r.iter_lines())
    data_lines = [c for c in chunks if c.startswith("data: ")]
    assert data_lines[-1] == "data: [DONE]"
    assert len(data_lines) >= 4  # 3 tokens + DONE
    first = json.loads(data_lines[0][len("data: "):])
    assert first["object"] == "chat.completion.chunk"


def test_models_listing(llm_client):
    r = llm_client.get("/serve/openai/v1/models")
    assert r.status_code == 200
    assert any(m["id"] == "test_llm" for m in r.json()["data"])


def test_generic_route_prompt(llm_client):
    r = llm_client.post("/serve/test_llm", json={
        "prompt": "xyz", "max_tokens": 2, "temperature": 0.0,
        "ignore_eos": True})
    assert r.status_code == 200, r.text
    assert r.json()["tokens"] == 2


def test_two_llm_models_one_process(processor, store, tmp_path):
    """Two DIFFERENT llama models serve from one process (the reference's
    vLLM integration is limited to one engine per container)."""
    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_singleton = None
    import asyncio

    for name, vocab in (("llm_a", 300), ("llm_b", 500)):
        card = tmp_path / (name + ".json")
        card.write_text(json.dumps({
            "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 32,
            "block_size": 16, "max_model_len": 64, "device": "cpu",
            "overrides": {"vocab_size": vocab}}))
        rec = store.register_model(name=name, project="p", path=str(card))
        processor.add_endpoint(ModelEndpoint(
            engine_type="llm", serving_url=name, model_id=rec.model_id))

    async def ask(name):
        return await processor.process_request(name, "", {
            "prompt": "x", "max_tokens": 2, "temperature": 0.0,
            "ignore_eos": True})

    loop = asyncio.new_event_loop()
    a = loop.run_until_complete(ask("llm_a"))
    b = loop.run_until_complete(ask("llm_b"))
    loop.close()
    assert a["tokens"] == 2 and b["tokens"] == 2
    assert len(LlmPreprocessRequest._engines) == 2
    vocabs = {e.model_config.vocab_size
              for e in LlmPreprocessRequest._engines.values()}
    assert vocabs == {300, 500}
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_singleton = None


def test_embeddings_route(llm_client):
    r = llm_client.post("/serve/openai/v1/embeddings", json={
        "model": "test_llm", "input": ["alpha", "beta"]})
    assert r.status_code == 200, r.text
    out = r.json()
    assert out["object"] == "list" and len(out["data"]) == 2
    assert len(out["data"][0]["embedding"]) > 0
    assert out["usage"]["prompt_tokens"] > 0


def test_tokenize_route(llm_client):
    r = llm_client.post("/serve/openai/tokenize", json={
        "model": "test_llm", "prompt": "hi there"})
    assert r.status_code == 200, r.text
    toks = r.json()["tokens"]
    r2 = llm_client.post("/serve/openai/detokenize", json={
        "model": "test_llm", "tokens": toks})
    assert r2.json()["prompt"] == "hi there"


def test_score_and_rerank_routes(llm_client):
    r = llm_client.post("/serve/openai/v1/score", json={
        "model": "test_llm", "text_1": "query text",
        "text_2": ["query text", "unrelated words entirely"]})
    assert r.status_code == 200, r.text
    scores = [d["score"] for d in r.json()["data"]]
    assert len(scores) == 2
    # identical text scores (cosine) ~1.0 and above the unrelated text
    assert scores[0] > 0.99 and scores[0] >= scores[1]

    r = llm_client.post("/serve/openai/v1/rerank", json={
        "model": "test_llm", "query": "query text",
        "documents": ["unrelated words entirely", "query text"], "top_n": 1})
    assert r.status_code == 200, r.text
    res = r.json()["results"]
    assert len(res) == 1 and res[0]["index"] == 1


def test_pooling_route_unnormalized(llm_client):
    r = llm_client.post("/serve/openai/pooling", json={
        "model": "test_llm", "input": "some text"})
    assert r.status_code == 200, r.text
    vec = r.json()["data"][0]["data"]
    norm = sum(x * x for x in vec) ** 0.5
    assert norm > 0 and abs(norm - 1.0) > 1e-3  # raw, not unit-norm


def test_removed_llm_endpoint_frees_engine(processor, store, tmp_path):
    """Swapping an LLM endpoint to a new model tears down the superseded
    model's engine (auto-update would otherwise leak one engine's HBM per
    version)."""
    import asyncio

    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_refs = {}

    def card(name):
        p = tmp_path / (name + ".json")
        p.write_text(json.dumps({
            "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 32,
            "block_size": 16, "max_model_len": 64, "device": "cpu"}))
        return store.register_model(name=name, project="p", path=str(p))

    loop = asyncio.new_event_loop()
    try:
        r1 = card("v1")
        processor.add_endpoint(ModelEndpoint(
            engine_type="llm", serving_url="swap_llm", model_id=r1.model_id))
        processor.serialize()
        out = loop.run_until_complete(processor.process_request(
            "swap_llm", None, {"prompt": "x", "max_tokens": 2,
                               "ignore_eos": True, "temperature": 0.0}))
        assert out["tokens"] == 2
        assert len(LlmPreprocessRequest._engines) == 1
        old_engine = LlmPreprocessRequest._engines[r1.model_id]

        r2 = card("v2")
        processor.add_endpoint(ModelEndpoint(
            engine_type="llm", serving_url="swap_llm", model_id=r2.model_id))
        processor.serialize()
        processor.deserialize()  # what the sync daemon does on config change
        out = loop.run_until_complete(processor.process_request(
            "swap_llm", None, {"prompt": "x", "max_tokens": 2,
                               "ignore_eos": True, "temperature": 0.0}))
        assert out["tokens"] == 2
        # old engine evicted and its model/KV references dropped
        assert r1.model_id not in LlmPreprocessRequest._engines
        assert old_engine.model is None and old_engine.kv_caches == []
        assert len(LlmPreprocessRequest._engines) == 1
    finally:
        for t in asyncio.all_tasks(loop):
            t.cancel()
        loop.run_until_complete(asyncio.sleep(0))
        loop.close()
        LlmPreprocessRequest._engine_singleton = None
        LlmPreprocessRequest._engines = {}
        LlmPreprocessRequest._engine_refs = {}


def test_unsupported_vllm_serve_types_explain_themselves(llm_client):
    """classify / audio transcription / translation: the reference's vLLM
    errors when the model's task does not match the handler; here the
    routes 422 with a message naming the alternative (hip-engine
    classification) or the missing model family (audio)."""
    r = llm_client.post("/serve/openai/classify", json={
        "model": "test_llm", "input": "some text"})
    assert r.status_code == 422, r.text
    assert "classification" in r.json()["detail"]

    for route in ("v1/audio/transcriptions", "v1/audio/translations"):
        r = llm_client.post("/serve/openai/" + route, json={
            "model": "test_llm"})
        assert r.status_code == 422, r.text
        assert "audio" in r.json()["detail"]


def test_openai_routes_are_json_only(llm_client):
    """Reference parity (main.py:207-215): non-JSON content types 415 on
    the OpenAI routes."""
    r = llm_client.post("/serve/openai/v1/chat/completions",
                        content=b"model=x",
                        headers={"Content-Type":
                                 "application/x-www-form-urlencoded"})
    assert r.status_code == 415, r.text


def test_sse_stream_options_include_usage(llm_client):
    """OpenAI stream_options.include_usage: a final pre-[DONE] chunk with
    empty choices and the usage totals."""
    r = llm_client.post("/serve/openai/v1/chat/completions", json={
        "model": "test_llm", "max_tokens": 5, "temperature": 0.0,
        "ignore_eos": True, "stream": True,
        "stream_options": {"include_usage": True},
        "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 200
    lines = [ln for ln in r.text.splitlines() if ln.startswith("data: ")]
    assert lines[-1] == "data: [DONE]"
    final = json.loads(lines[-2][len("data: "):])
    assert final["choices"] == []
    assert final["usage"]["completion_tokens"] == 5
    assert final["usage"]["total_tokens"] > 5


def test_response_format_json_mode_refused(llm_client):
    """No guided-decoding backend: response_format json_object 422s with
    a clear message instead of silently returning free text."""
    r = llm_client.post("/serve/openai/v1/chat/completions", json={
        "model": "test_llm", "max_tokens": 4,
        "response_format": {"type": "json_object"},
        "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 422, r.text
    assert "response_format" in r.json()["detail"]
    # explicit text type passes through
    r = llm_client.post("/serve/openai/v1/chat/completions", json={
        "model": "test_llm", "max_tokens": 2, "temperature": 0.0,
        "ignore_eos": True, "response_format": {"type": "text"},
        "messages": [{"role": "user", "content": "hi"}]})
    assert r.status_code == 200, r.text


def test_bare_score_rerank_version_routes(llm_client):
    """Reference serve-type parity: vLLM also mounts bare /score and
    /rerank (preprocess_service.py:1290-1336) and a /version route."""
    r = llm_client.post("/serve/openai/score", json={
        "model": "test_llm", "text_1": "q", "text_2": ["a", "b"]})
    assert r.status_code == 200, r.text
    assert len(r.json()["data"]) == 2

    r = llm_client.post("/serve/openai/rerank", json={
        "model": "test_llm", "query": "q", "documents": ["a", "b"],
        "top_n": 1})
    assert r.status_code == 200, r.text
    assert len(r.json()["results"]) == 1

    r = llm_client.post("/serve/openai/version", json={"model": "test_llm"})
    assert r.status_code == 200, r.text
    from clearml_serving_amd import __version__
    assert r.json()["version"] == __version__


def test_full_sampling_surface_through_http(llm_client):
    """Client-visible contract: n + seed + penalties + logprobs in one
    chat request; response carries every structure."""
    body = {
        "model": "test_llm", "messages": [{"role": "user", "content": "x"}],
        "max_tokens": 6, "temperature": 0.9, "top_p": 0.9, "seed": 42,
        "n": 2, "presence_penalty": 0.5, "frequency_penalty": 0.5,
        "repetition_penalty": 1.1, "logprobs": True, "top_logprobs": 2,
        "min_tokens": 2, "ignore_eos": True,
    }
    r1 = llm_client.post("/serve/openai/v1/chat/completions", json=body)
    assert r1.status_code == 200, r1.text
    out1 = r1.json()
    assert len(out1["choices"]) == 2
    for c in out1["choices"]:
        assert len(c["logprobs"]["content"]) == 6
        assert all(len(e["top_logprobs"]) == 2
                   for e in c["logprobs"]["content"])
    # seeded: the same request reproduces both choices exactly
    r2 = llm_client.post("/serve/openai/v1/chat/completions", json=body)
    out2 = r2.json()
    assert [c["message"]["content"] for c in out1["choices"]] == \
        [c["message"]["content"] for c in out2["choices"]]
    # and the two seeded choices differ from each other (offset seeds)
    assert out1["choices"][0]["message"]["content"] != \
        out1["choices"][1]["message"]["content"]


def test_models_listing_via_get(llm_client):
    """GET /serve/openai/v1/models with no body (the openai SDK's list
    call) returns the llm endpoints."""
    r = llm_client.get("/serve/openai/v1/models")
    assert r.status_code == 200, r.text
    ids = [m["id"] for m in r.json()["data"]]
    assert "test_llm" in ids
