"""Chaos test: concurrent mixed traffic + config hot reloads + stream
aborts against one processor -- the interactions the per-feature tests
don't cover together (stall-swap drain vs in-flight LLM streams vs
engine-cache flush)."""

import asyncio
import json

import pytest

from clearml_serving_amd.schemas import ModelEndpoint
from clearml_serving_amd.serving.processor import ModelRequestProcessor
from clearml_serving_amd.store import ServingStore


@pytest.mark.timeout(360)
@pytest.mark.parametrize("llm_extras", [
    {},
    # every opt-in engine feature at once: speculation + prefix caching
    {"speculative": {"method": "ngram", "num_spec_tokens": 4, "ngram": 2},
     "enable_prefix_caching": True},
], ids=["plain", "spec+prefixcache"])
def test_mixed_traffic_survives_hot_reloads(tmp_path, llm_extras):
    import joblib
    import numpy as np
    from sklearn.linear_model import LinearRegression

    from clearml_serving_amd.engines.llm.adapter import LlmPreprocessRequest

    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_refs = {}

    store = ServingStore(str(tmp_path / "store"))
    proc = ModelRequestProcessor(store=store, name="chaos",
                                 force_create=True)
    proc._metric_log_freq = 1.0  # stats on every request

    X = np.array([[0.0, 0.0], [1.0, 0.0], [0.0, 1.0]])
    model = LinearRegression().fit(X, np.array([0.0, 2.0, 3.0]))
    pkl = tmp_path / "m.pkl"
    joblib.dump(model, str(pkl))
    rec = store.register_model(name="lin", project="p", path=str(pkl))
    proc.add_endpoint(ModelEndpoint(engine_type="sklearn",
                                    serving_url="sk", model_id=rec.model_id))

    card = tmp_path / "card.json"
    card.write_text(json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 96,
        "block_size": 16, "max_model_len": 128, "device": "cpu",
        **llm_extras}))
    lrec = store.register_model(name="tl", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tl",
                                    model_id=lrec.model_id))
    proc.serialize()
    proc.deserialize(skip_sync=True)

    stats_batches = []
    proc.set_stats_sink(stats_batches.append)

    results = {"ok": 0, "aborted": 0, "reloads": 0}

    async def main():
        async def sk_req(i):
            out = await proc.process_request("sk", "", [[i % 3, (i + 1) % 3]])
            assert out is not None
            results["ok"] += 1

        async def llm_req(i):
            out = await proc.process_request(
                "tl", "", {"prompt": "p%d" % i, "max_tokens": 6,
                           "temperature": 0.0, "ignore_eos": True})
            assert out["tokens"] == 6
            results["ok"] += 1

        async def llm_abort(i):
            # start a long generation, cancel the awaiting task mid-way
            task = asyncio.ensure_future(proc.process_request(
                "tl", "", {"prompt": "a%d" % i, "max_tokens": 60,
                           "temperature": 0.0, "ignore_eos": True}))
            await asyncio.sleep(0.05 + (i % 3) * 0.03)
            task.cancel()
            try:
                await task
            except asyncio.CancelledError:
                results["aborted"] += 1

        async def reloader():
            # config churn: 3 hot reloads while traffic is in flight
            for r in range(3):
                await asyncio.sleep(0.15)
                # a config change that flushes nothing critical (metric
                # logging tweak) but runs the full stall-swap drain
                proc.configure(default_metric_log_freq=0.5 + 0.1 * r)
                proc.serialize()
                await asyncio.to_thread(proc.deserialize)
                results["reloads"] += 1

        jobs = []
        for i in range(30):
            jobs.append(sk_req(i))
            if i % 2 == 0:
                jobs.append(llm_req(i))
            if i % 7 == 0:
                jobs.append(llm_abort(i))
        jobs.append(reloader())
        await asyncio.gather(*jobs)

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
        # drain: give aborted sequences a few scheduler steps to clean up
        loop.run_until_complete(asyncio.sleep(0.5))
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()

    assert results["ok"] == 30 + 15, results
    assert results["aborted"] == 5, results
    assert results["reloads"] == 3
    # the llm engine survived with every KV page returned
    eng = None
    for e in LlmPreprocessRequest._engines.values():
        eng = e
    if eng is not None and eng.allocator is not None:
        assert eng.allocator.available == eng.allocator.num_blocks
    LlmPreprocessRequest._engine_singleton = None
    LlmPreprocessRequest._engines = {}
    LlmPreprocessRequest._engine_refs = {}
