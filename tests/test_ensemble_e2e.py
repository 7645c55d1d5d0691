"""BASELINE config 5 shape: multi-model ensemble pipeline with canary
routing (CPU-scale models; same code path as the 8-GPU deployment)."""

import asyncio
import json
import os

import numpy as np
import pytest

from clearml_serving_amd.schemas import CanaryEP, ModelEndpoint


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        # drain lingering batcher workers before closing the loop
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()


@pytest.fixture()
def ensemble(processor, store, tmp_path):
    # two hip model endpoints (tiny bert variants as stand-ins) + an async
    # ensemble that fans out to both and combines
    for ver, labels in (("1", 2), ("2", 3)):
        card = tmp_path / ("card%s.json" % ver)
        card.write_text(json.dumps({
            "arch": "bert-base", "num_labels": labels, "dtype": "float32",
            "vocab_size": 200}))
        rec = store.register_model(name="b" + ver, project="p",
                                   path=str(card))
        processor.add_endpoint(ModelEndpoint(
            engine_type="hip", serving_url="encoder", version=ver,
            model_id=rec.model_id,
            auxiliary_cfg={"max_queue_delay_us": 500, "use_graphs": False}))

    # canary: 70% v2, 30% v1 by prefix (newest first)
    processor.add_canary_endpoint(CanaryEP(
        endpoint="encoder_canary", weights=[0.7, 0.3],
        load_endpoint_prefix="encoder"))
    processor._update_canary_lookup()

    code = tmp_path / "ens.py"
    code.write_text(
        "import asyncio\n"
        "class Preprocess(object):\n"
        "    async def process(self, data, state, collect_custom_statistics_fn=None):\n"
        "        a, b = await asyncio.gather(\n"
        "            self.send_request(endpoint='encoder', version='1', data=data),\n"
        "            self.send_request(endpoint='encoder_canary', data=data))\n"
        "        assert a is not None and b is not None\n"
        "        return {'n_a': len(a), 'n_b': len(b)}\n")
    processor.add_endpoint(
        ModelEndpoint(engine_type="custom_async", serving_url="ensemble"),
        preprocess_code=str(code))
    return processor


def test_ensemble_with_canary(ensemble):
    body = {"input_ids": list(range(1, 17)), "attention_mask": [1] * 16}

    async def many(n):
        return await asyncio.gather(*[
            ensemble.process_request("ensemble", "", dict(body))
            for _ in range(n)])

    outs = run(many(20))
    # branch a always hits v1 (2 labels); canary mixes v1 (2) and v2 (3)
    assert all(o["n_a"] == 2 for o in outs)
    sizes = {o["n_b"] for o in outs}
    assert sizes <= {2, 3} and 3 in sizes  # newest version drawn


def test_canary_route_weights(ensemble):
    route = ensemble._canary_route["encoder_canary"]
    assert route["endpoints"] == ["encoder/2", "encoder/1"]
    assert abs(route["weights"][0] - 0.7) < 1e-9


def test_threaded_sync_pipeline(processor, store, tmp_path):
    """Reference parity: examples/pipeline/preprocess.py:18-32 -- a SYNC
    Preprocess.process() that fans out via ThreadPoolExecutor +
    self.send_request (blocking). The custom engine runs process() off-loop
    and sync send_request schedules onto the live serving loop."""
    import joblib
    from sklearn.linear_model import LinearRegression

    X = np.array([[0.0, 0.0], [1.0, 1.0], [2.0, 2.0]])
    m = LinearRegression().fit(X, np.array([0.0, 2.0, 4.0]))
    mp = tmp_path / "lin.pkl"
    joblib.dump(m, mp)
    rec = store.register_model(name="lin", project="p", path=str(mp))
    sk_code = tmp_path / "sk.py"
    sk_code.write_text(
        "import numpy as np\n"
        "class Preprocess(object):\n"
        "    def preprocess(self, body, state, collect_custom_statistics_fn=None):\n"
        "        return np.array([[body['x0'], body['x1']]])\n"
        "    def postprocess(self, data, state, collect_custom_statistics_fn=None):\n"
        "        return dict(y=data.tolist())\n")
    processor.add_endpoint(ModelEndpoint(
        engine_type="sklearn", serving_url="test_model_sklearn",
        model_id=rec.model_id), preprocess_code=str(sk_code))

    # the threaded pipeline example, verbatim from examples/pipeline
    import shutil
    pipe_code = tmp_path / "pipe.py"
    shutil.copyfile(
        os.path.join(os.path.dirname(__file__), "..", "examples",
                     "pipeline", "preprocess.py"), pipe_code)
    processor.add_endpoint(
        ModelEndpoint(engine_type="custom", serving_url="pipeline"),
        preprocess_code=str(pipe_code))

    async def go():
        return await processor.process_request(
            "pipeline", "", {"x0": 1.0, "x1": 1.0})

    out = run(go())
    # both branches predict 2.0; postprocess averages them
    assert abs(out["y"] - 2.0) < 1e-6


def test_sync_send_request_inline_on_loop_raises(processor, store, tmp_path):
    """Calling sync send_request ON the event loop thread fails loudly
    (it used to silently return None)."""
    code = tmp_path / "bad.py"
    code.write_text(
        "class Preprocess(object):\n"
        "    async def process(self, data, state, collect_custom_statistics_fn=None):\n"
        "        # WRONG: sync send_request is not awaitable; calling the\n"
        "        # custom engine's bound sync variant inline must raise\n"
        "        return self.send_request(endpoint='x', data=data)\n")
    # custom_async engines get the async send_request; build a custom (sync)
    # engine whose user code grabs the sync one and calls it inline
    code2 = tmp_path / "bad2.py"
    code2.write_text(
        "class Preprocess(object):\n"
        "    def process(self, data, state, collect_custom_statistics_fn=None):\n"
        "        import asyncio\n"
        "        loop = asyncio.new_event_loop()\n"
        "        try:\n"
        "            return loop.run_until_complete(self._call())\n"
        "        finally:\n"
        "            loop.close()\n"
        "    async def _call(self):\n"
        "        return self.send_request(endpoint='x', data={})\n")
    processor.add_endpoint(
        ModelEndpoint(engine_type="custom", serving_url="bad"),
        preprocess_code=str(code2))

    async def go():
        return await processor.process_request("bad", "", {})

    with pytest.raises(RuntimeError, match="event loop"):
        run(go())
