"""Multi-process serving topology: SHM rings, framing, front<->owner e2e.

The topology under test (serving/launch.py): N HTTP front workers sharing
one port via SO_REUSEPORT, one engine-owner process per GPU, connected by
SPSC shared-memory rings -- GPU dynamic batches stay whole regardless of
HTTP worker count (replaces the reference's N-workers-N-model-copies
gunicorn mode, entrypoint.sh:56-72)."""

import asyncio
import json
import os
import socket
import subprocess
import sys
import time
import uuid

import numpy as np
import pytest

from clearml_serving_amd.serving import shm_transport as st

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


# --------------------------------------------------------------------- #
# ring semantics (both implementations)
# --------------------------------------------------------------------- #
def _ring_roundtrip(make, unlink):
    name = "/cmls_t_{}".format(uuid.uuid4().hex[:8])
    w = make(name, 1 << 14, True)
    r = make(name, 0, False)
    try:
        assert w.push(b"a")
        assert w.push(b"bb" * 100)
        assert r.drain(10) == [b"a", b"bb" * 100]
        # wrap across the end many times
        for i in range(500):
            payload = bytes([i % 251]) * (i % 900 + 1)
            assert w.push(payload)
            assert r.drain(4) == [payload]
        # backpressure: fill to capacity, then drain everything
        n = 0
        while w.push(b"z" * 1000):
            n += 1
        assert n >= 10
        got = []
        while True:
            batch = r.drain(64)
            if not batch:
                break
            got.extend(batch)
        assert len(got) == n
        # ring usable again after full drain
        assert w.push(b"after")
        assert r.drain(4) == [b"after"]
    finally:
        w.close()
        r.close()
        unlink(name)


def test_python_ring_roundtrip():
    _ring_roundtrip(st.PyShmRing, st.PyShmRing.unlink)


@pytest.mark.skipif(not st.HAVE_NATIVE_RING,
                    reason="_shmring extension not built")
def test_native_ring_roundtrip():
    _ring_roundtrip(st.ShmRing, st.ShmRing.unlink)


@pytest.mark.skipif(not st.HAVE_NATIVE_RING,
                    reason="_shmring extension not built")
def test_native_and_python_rings_interoperate():
    """Both implementations share the record format: a producer of one kind
    must be drainable by the other (fronts may run without the extension)."""
    name = "/cmls_x_{}".format(uuid.uuid4().hex[:8])
    w = st.ShmRing(name, 1 << 14, True)
    r = st.PyShmRing(name, 0, False)
    try:
        for p in (b"one", b"two" * 50, b"\x00" * 999):
            assert w.push(p)
        assert r.drain(10) == [b"one", b"two" * 50, b"\x00" * 999]
        # and the reverse direction on the same ring
        assert r.push(b"back")
        assert w.drain(10) == [b"back"]
    finally:
        w.close()
        r.close()
        st.ShmRing.unlink(name)


# --------------------------------------------------------------------- #
# framing
# --------------------------------------------------------------------- #
def test_request_framing_tensor_dict():
    data = {"input_ids": np.arange(128, dtype=np.int64),
            "attention_mask": np.ones(128, dtype=np.int32)}
    buf = st.pack_request(42, "bert_ep/1", data)
    rid, url, out = st.unpack_request(buf)
    assert rid == 42 and url == "bert_ep/1"
    assert set(out) == {"input_ids", "attention_mask"}
    np.testing.assert_array_equal(out["input_ids"], data["input_ids"])
    assert out["attention_mask"].dtype == np.int32


def test_request_framing_single_array_and_lists():
    a = np.random.rand(3, 4).astype(np.float32)
    rid, url, out = st.unpack_request(st.pack_request(1, "m", a))
    np.testing.assert_array_equal(out, a)
    # plain python list converts (float64 -> float32)
    rid, url, out = st.unpack_request(st.pack_request(2, "m", [1.0, 2.0]))
    assert out.dtype == np.float32
    np.testing.assert_allclose(out, [1.0, 2.0])


def test_request_framing_pickle_fallback():
    body = {"prompt": "hi", "max_tokens": 4, "nested": {"a": [1, "x"]}}
    rid, url, out = st.unpack_request(st.pack_request(3, "llm", body))
    assert out == body


def test_response_framing():
    rid, status, out = st.unpack_response(
        st.pack_response(7, np.float32([1, 2, 3])))
    assert rid == 7 and status == st.STATUS_OK_TENSORS
    np.testing.assert_allclose(out, [1, 2, 3])
    rid, status, out = st.unpack_response(
        st.pack_response(8, {"text": "ok", "tokens": 3}))
    assert status == st.STATUS_OK_PICKLE and out["text"] == "ok"
    rid, status, out = st.unpack_response(st.pack_response(9, error="boom"))
    assert status == st.STATUS_ERROR and out == "boom"


def test_placement_deterministic():
    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.front import _placement

    ep = ModelEndpoint(engine_type="hip", serving_url="m1",
                       auxiliary_cfg={"gpu": 3})
    assert _placement(ep, 8) == 3
    assert _placement(ep, 2) == 1
    ep2 = ModelEndpoint(engine_type="hip", serving_url="m2")
    assert _placement(ep2, 4) == _placement(ep2, 4)  # stable


# --------------------------------------------------------------------- #
# full multi-process e2e over real HTTP (CPU engines + SHM proxy)
# --------------------------------------------------------------------- #
def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture()
def serving_session(tmp_path):
    """Store with a hip (CPU bert) endpoint + an sklearn endpoint."""
    import joblib
    from sklearn.linear_model import LinearRegression

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="shm-e2e",
                                 force_create=True)
    card = tmp_path / "bert.json"
    card.write_text(json.dumps({"arch": "bert-base", "num_labels": 2,
                                "dtype": "float32", "vocab_size": 500}))
    rec = store.register_model(name="bert", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="transformer_model",
        model_id=rec.model_id,
        auxiliary_cfg={"max_queue_delay_us": 2000, "use_graphs": False,
                       "warmup": False}))

    X = np.array([[0.0, 0.0], [1.0, 1.0], [2.0, 2.0]])
    m = LinearRegression().fit(X, np.array([0.0, 2.0, 4.0]))
    mp = tmp_path / "lin.pkl"
    joblib.dump(m, mp)
    rec2 = store.register_model(name="lin", project="p", path=str(mp))
    code = tmp_path / "sk.py"
    code.write_text(
        "import numpy as np\n"
        "class Preprocess(object):\n"
        "    def preprocess(self, body, state, collect_custom_statistics_fn=None):\n"
        "        return np.array([[body['x0'], body['x1']]])\n"
        "    def postprocess(self, data, state, collect_custom_statistics_fn=None):\n"
        "        return dict(y=data.tolist()[0])\n")
    proc.add_endpoint(ModelEndpoint(
        engine_type="sklearn", serving_url="test_model_sklearn",
        model_id=rec2.model_id), preprocess_code=str(code))
    proc.serialize()
    return store_root, proc.get_id()


@pytest.mark.timeout(180)
def test_multiproc_front_owner_http_e2e(serving_session):
    """launch.py with 2 fronts + 1 owner: hip endpoint served through the
    SHM proxy (batching in the owner), sklearn served front-local; both
    answer over real HTTP on a shared SO_REUSEPORT port."""
    store_root, session_id = serving_session
    port = _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    launcher = subprocess.Popen(
        [sys.executable, "-m", "clearml_serving_amd.serving.launch",
         "--store", store_root, "--session", session_id,
         "--host", "127.0.0.1", "--port", str(port),
         "--workers", "2", "--owners", "1", "--ring-mb", "8",
         "--no-restart"],
        env=env, cwd=REPO)
    try:
        import httpx

        base = "http://127.0.0.1:{}".format(port)
        with httpx.Client(base_url=base, timeout=30.0) as client:
            for _ in range(240):
                try:
                    if client.get("/health").status_code == 200:
                        break
                except Exception:
                    pass
                time.sleep(0.5)
                assert launcher.poll() is None, "launcher died"
            else:
                pytest.fail("service did not come up")

            # sklearn (front-local CPU engine)
            r = client.post("/serve/test_model_sklearn",
                            json={"x0": 1.0, "x1": 1.0})
            assert r.status_code == 200, r.text
            assert abs(float(np.asarray(r.json()["y"]).ravel()[0]) - 2.0) < 1e-5

            # hip endpoint through the SHM proxy (dict tensor payload)
            payload = {"input_ids": list(range(1, 65)),
                       "attention_mask": [1] * 64}
            outs = []
            for _ in range(12):
                r = client.post("/serve/transformer_model", json=payload)
                assert r.status_code == 200, r.text
                outs.append(r.json())
            assert all(len(o) == 2 for o in outs)
            # deterministic model: identical inputs -> identical logits
            assert all(np.allclose(o, outs[0], atol=1e-4) for o in outs)

            # 404 still routes correctly through the front
            r = client.post("/serve/nope", json={})
            assert r.status_code == 404
    finally:
        launcher.terminate()
        try:
            launcher.wait(timeout=20)
        except subprocess.TimeoutExpired:
            launcher.kill()


@pytest.mark.timeout(240)
def test_multiproc_llm_and_sse_streaming(tmp_path):
    """LLM endpoints through the multi-process front: whole-request dispatch
    AND SSE streaming relayed chunk-by-chunk over the response ring."""
    import json as _json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="shm-llm",
                                 force_create=True)
    card = tmp_path / "card.json"
    card.write_text(_json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu"}))
    rec = store.register_model(name="tiny", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tinyllm",
                                    model_id=rec.model_id))
    proc.serialize()

    port = _free_port()
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    launcher = subprocess.Popen(
        [sys.executable, "-m", "clearml_serving_amd.serving.launch",
         "--store", store_root, "--session", proc.get_id(),
         "--host", "127.0.0.1", "--port", str(port),
         "--workers", "1", "--owners", "1", "--ring-mb", "8",
         "--no-restart"],
        env=env, cwd=REPO)
    try:
        import httpx

        base = "http://127.0.0.1:{}".format(port)
        with httpx.Client(base_url=base, timeout=60.0) as client:
            for _ in range(240):
                try:
                    if client.get("/health").status_code == 200:
                        break
                except Exception:
                    pass
                time.sleep(0.5)
                assert launcher.poll() is None, "launcher died"
            else:
                pytest.fail("service did not come up")

            # non-streaming chat completion through the SHM hop
            r = client.post("/serve/openai/v1/chat/completions", json={
                "model": "tinyllm",
                "messages": [{"role": "user", "content": "hello"}],
                "max_tokens": 4, "temperature": 0, "ignore_eos": True})
            assert r.status_code == 200, r.text
            body = r.json()
            assert body["usage"]["completion_tokens"] == 4

            # SSE streaming: chunks relayed over the ring
            chunks = []
            with client.stream("POST", "/serve/openai/v1/chat/completions",
                               json={"model": "tinyllm", "stream": True,
                                     "messages": [{"role": "user",
                                                   "content": "hi"}],
                                     "max_tokens": 5, "temperature": 0,
                                     "ignore_eos": True}) as resp:
                assert resp.status_code == 200
                assert "text/event-stream" in resp.headers["content-type"]
                for line in resp.iter_lines():
                    if line.startswith("data: "):
                        chunks.append(line[len("data: "):])
            assert chunks[-1] == "[DONE]"
            deltas = [_json.loads(c) for c in chunks[:-1]]
            toks = [d["choices"][0]["delta"].get("content", "")
                    for d in deltas]
            assert len(toks) == 5
    finally:
        launcher.terminate()
        try:
            launcher.wait(timeout=20)
        except subprocess.TimeoutExpired:
            launcher.kill()


@pytest.mark.timeout(300)
def test_shmring_asan_ubsan_tsan():
    """Sanitizer posture for the native tier: the ring core
    builds and passes its harness (wrap torture + backpressure + a real
    two-thread SPSC run) under ASan+UBSan, and under TSan where available."""
    import shutil
    import tempfile

    gxx = shutil.which("g++")
    if gxx is None:
        pytest.skip("no g++")
    src = os.path.join(REPO, "clearml_serving_amd", "serving", "csrc",
                       "shmring_test.cpp")
    with tempfile.TemporaryDirectory() as td:
        exe = os.path.join(td, "shmring_asan")
        r = subprocess.run(
            [gxx, "-O1", "-g", "-std=c++17",
             "-fsanitize=address,undefined", src, "-o", exe, "-lrt",
             "-lpthread"], capture_output=True, text=True, timeout=180)
        assert r.returncode == 0, r.stderr
        r = subprocess.run([exe], capture_output=True, text=True,
                           timeout=120)
        assert r.returncode == 0, r.stdout + r.stderr
        assert "OK" in r.stdout

        # TSan build (supported by this gcc; the SPSC thread pair is the
        # production access pattern)
        exe2 = os.path.join(td, "shmring_tsan")
        r = subprocess.run(
            [gxx, "-O1", "-g", "-std=c++17", "-fsanitize=thread", src,
             "-o", exe2, "-lrt", "-lpthread"],
            capture_output=True, text=True, timeout=180)
        if r.returncode != 0:
            pytest.skip("tsan runtime unavailable: " + r.stderr[:200])
        r = subprocess.run([exe2], capture_output=True, text=True,
                           timeout=180)
        assert r.returncode == 0, r.stdout + r.stderr


@pytest.mark.timeout(120)
def test_owner_cancels_generation_on_stream_abort(tmp_path):
    """Closing an SSE consumer mid-stream sends an abort record over the
    request ring; the owner cancels the in-flight generation (the engine
    does not run to max_tokens for a dead client)."""
    import json as _json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.engine_owner import EngineOwner
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.serving.shm_transport import ShmClient
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="abrt", force_create=True)
    card = tmp_path / "card.json"
    card.write_text(_json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 512, "device": "cpu"}))
    rec = store.register_model(name="t", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tl",
                                    model_id=rec.model_id))
    proc.serialize()

    prefix = "/cmls_abrt_{}".format(os.getpid())
    owner = EngineOwner(store_root=store_root, session_id=proc.get_id(),
                        prefix=prefix, owner_idx=0, n_workers=1,
                        ring_bytes=1 << 20)
    client = ShmClient(prefix, 0, 1, ring_bytes=1 << 20)

    async def main():
        serve_task = asyncio.get_running_loop().create_task(owner.serve())
        payload = {"__serve_type__": "v1_chat_completions",
                   "__body__": {"messages": [{"role": "user",
                                              "content": "hi"}],
                                "stream": True, "max_tokens": 400,
                                "temperature": 0, "ignore_eos": True}}
        agen = client.infer_stream("tl", payload)
        chunks = 0
        async for _chunk in agen:
            chunks += 1
            if chunks >= 2:
                await agen.aclose()  # client disconnect -> abort record
                break
        # the owner should cancel; its llm engine records the abort
        eng = owner._engines["tl"]._engine
        for _ in range(200):
            await asyncio.sleep(0.05)
            if owner.stats["aborts"] >= 1 and eng.stats["aborts"] >= 1 \
                    and not eng.running and not eng.waiting:
                break
        assert owner.stats["aborts"] >= 1
        assert eng.stats["aborts"] >= 1
        assert not eng.running and not eng.waiting
        assert eng.stats["generated_tokens"] < 300  # nowhere near 400
        serve_task.cancel()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()
        client.close()
        owner.close()


@pytest.mark.timeout(120)
def test_abort_from_one_front_never_cancels_another(tmp_path):
    """Every front worker counts req_ids from 1, so the owner must track
    in-flight requests by (worker, id): worker 0 aborting ITS request 1
    must not cancel worker 1's request 1."""
    import json as _json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.engine_owner import EngineOwner
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.serving.shm_transport import ShmClient
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="xw", force_create=True)
    card = tmp_path / "card.json"
    card.write_text(_json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 512, "device": "cpu"}))
    rec = store.register_model(name="t", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tl",
                                    model_id=rec.model_id))
    proc.serialize()

    prefix = "/cmls_xw_{}".format(os.getpid())
    owner = EngineOwner(store_root=store_root, session_id=proc.get_id(),
                        prefix=prefix, owner_idx=0, n_workers=2,
                        ring_bytes=1 << 20)
    c0 = ShmClient(prefix, 0, 1, ring_bytes=1 << 20)
    c1 = ShmClient(prefix, 1, 1, ring_bytes=1 << 20)

    async def main():
        serve_task = asyncio.get_running_loop().create_task(owner.serve())

        def payload(n):
            return {"__serve_type__": "v1_chat_completions",
                    "__body__": {"messages": [{"role": "user",
                                               "content": "hi"}],
                                 "stream": True, "max_tokens": n,
                                 "temperature": 0, "ignore_eos": True}}

        # worker 1: full stream (40 tokens); worker 0: aborts after 2 chunks.
        # Order matters to catch the collision: worker 0's request registers
        # FIRST, worker 1's same-id request registers second (with a bare-id
        # map it would clobber the key), then worker 0's abort arrives --
        # a bare-id lookup would cancel worker 1's stream.
        async def full_stream():
            chunks = 0
            async for _ in c1.infer_stream("tl", payload(40)):
                chunks += 1
            return chunks

        async def aborting_stream():
            agen = c0.infer_stream("tl", payload(400))
            n = 0
            async for _ in agen:
                n += 1
                if n == 1:
                    # worker 0 in flight: now start worker 1's stream
                    await asyncio.sleep(0.2)
                if n >= 25:
                    await agen.aclose()
                    break

        ab = asyncio.get_running_loop().create_task(aborting_stream())

        async def delayed_full():
            await asyncio.sleep(0.1)  # after worker 0 registered
            return await full_stream()

        full = asyncio.get_running_loop().create_task(delayed_full())
        await asyncio.wait_for(ab, timeout=60)
        chunks = await asyncio.wait_for(full, timeout=60)
        # worker 1's stream survived worker 0's abort: 40 content chunks
        # (+1 final empty-finish chunk depending on timing) and a clean end
        assert chunks >= 40, chunks
        assert owner.stats["aborts"] >= 1
        serve_task.cancel()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()
        c0.close()
        c1.close()
        owner.close()


def test_owner_engine_swaps_on_hot_reload(tmp_path):
    """A config change (new model version for the same endpoint) propagates
    to the engine owner's sync daemon and the owner rebuilds the engine --
    the SHM-topology version of the zero-downtime hot reload."""
    import json as _json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.engine_owner import EngineOwner
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.serving.shm_transport import ShmClient
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="hotswap",
                                 force_create=True)

    def register_card(labels, name):
        card = tmp_path / (name + ".json")
        card.write_text(_json.dumps({
            "arch": "bert-base", "num_labels": labels, "dtype": "float32",
            "vocab_size": 300}))
        return store.register_model(name=name, project="p", path=str(card))

    rec1 = register_card(2, "v1")
    proc.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="enc", model_id=rec1.model_id,
        auxiliary_cfg={"use_graphs": False, "warmup": False,
                       "max_queue_delay_us": 500}))
    proc.serialize()

    prefix = "/cmls_swap_{}".format(os.getpid())
    owner = EngineOwner(store_root=store_root, session_id=proc.get_id(),
                        prefix=prefix, owner_idx=0, n_workers=1,
                        ring_bytes=1 << 20, poll_frequency_sec=0.5)
    client = ShmClient(prefix, 0, 1, ring_bytes=1 << 20)

    async def main():
        serve_task = asyncio.get_running_loop().create_task(owner.serve())
        body = {"input_ids": np.arange(1, 17, dtype=np.int64),
                "attention_mask": np.ones(16, dtype=np.int64)}
        out1 = await client.infer("enc", body)
        assert np.asarray(out1).shape == (2,)
        first_engine = owner._engines["enc"]

        # hot swap: same endpoint, new model (3 labels)
        rec2 = register_card(3, "v2")
        proc.add_endpoint(ModelEndpoint(
            engine_type="hip", serving_url="enc", model_id=rec2.model_id,
            auxiliary_cfg={"use_graphs": False, "warmup": False,
                          "max_queue_delay_us": 500}))
        proc.serialize()

        # owner's sync daemon polls every 0.5 s; the next request after the
        # sync must serve the NEW model
        for _ in range(60):
            await asyncio.sleep(0.25)
            out = await client.infer("enc", body)
            if np.asarray(out).shape == (3,):
                break
        assert np.asarray(out).shape == (3,)
        assert owner._engines["enc"] is not first_engine
        serve_task.cancel()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()
        client.close()
        owner.close()


def test_oversize_record_rejected_not_deadlocked():
    """Records over cap/2 can deadlock a wrap position (at_end + need >
    cap forever): both ring implementations must reject them loudly."""
    from clearml_serving_amd.serving.shm_transport import (
        HAVE_NATIVE_RING, PyShmRing, make_ring, unlink_ring)

    name = "/cmls_big_{}".format(os.getpid())
    ring = make_ring(name, 1 << 16, True)
    try:
        with pytest.raises(RuntimeError):
            ring.push(b"x" * ((1 << 15) + 64))
        # half-capacity records still flow
        assert ring.push(b"y" * ((1 << 14)))
        assert len(ring.drain(4)) == 1
    finally:
        ring.close()
        unlink_ring(name)

    # the pure-python ring enforces the same bound
    name2 = "/cmls_big_py_{}".format(os.getpid())
    ring2 = PyShmRing(name2, 1 << 16, True)
    try:
        with pytest.raises(RuntimeError):
            ring2.push(b"x" * ((1 << 15) + 64))
    finally:
        ring2.close()
        PyShmRing.unlink(name2)


def test_owner_exports_batcher_telemetry(tmp_path):
    """The engine owner (no HTTP server) exports its per-stage batcher
    telemetry to Prometheus via a shim over its OWN engine map -- the
    processor's reload flush must never tear owner engines down, so they
    stay out of its lookup."""
    import json as _json

    from prometheus_client import CollectorRegistry, generate_latest

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.engine_owner import (
        EngineOwner, _OwnerTelemetryShim)
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.serving.shm_transport import ShmClient
    from clearml_serving_amd.statistics.collector import (
        BatcherStagesCollector)
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="tel", force_create=True)
    card = tmp_path / "bert.json"
    card.write_text(_json.dumps({"arch": "bert-base", "num_labels": 2,
                                 "dtype": "float32", "vocab_size": 200}))
    rec = store.register_model(name="b", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(
        engine_type="hip", serving_url="telmodel", model_id=rec.model_id,
        auxiliary_cfg={"use_graphs": False, "warmup": False,
                       "max_queue_delay_us": 1000}))
    proc.serialize()

    prefix = "/cmls_tel_{}".format(os.getpid())
    owner = EngineOwner(store_root=store_root, session_id=proc.get_id(),
                        prefix=prefix, owner_idx=0, n_workers=1,
                        ring_bytes=1 << 20)
    client = ShmClient(prefix, 0, 1, ring_bytes=1 << 20)

    async def main():
        serve = asyncio.get_running_loop().create_task(owner.serve())
        ids = np.arange(16, dtype=np.int64) % 100
        out = await client.infer("telmodel", {
            "input_ids": ids, "attention_mask": np.ones(16, np.int32)})
        assert np.asarray(out).shape[-1] == 2
        serve.cancel()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()
        client.close()

    # the owner's engines stay OUT of the processor lookup (reload-flush
    # isolation) but the shim-backed collector sees their batcher stats
    assert "telmodel" not in owner.proc._engine_processor_lookup
    reg = CollectorRegistry()
    reg.register(BatcherStagesCollector(_OwnerTelemetryShim(owner._engines)))
    text = generate_latest(reg).decode()
    assert "serving_batches_total" in text
    assert 'endpoint="telmodel"' in text
    owner.close()


def test_nonstream_abort_crosses_the_ring(tmp_path):
    """Cancelling an awaiting NON-stream infer (client disconnect) sends
    an abort record: the owner cancels the in-flight generation instead
    of running to max_tokens for a dead client."""
    import json as _json

    from clearml_serving_amd.schemas import ModelEndpoint
    from clearml_serving_amd.serving.engine_owner import EngineOwner
    from clearml_serving_amd.serving.processor import ModelRequestProcessor
    from clearml_serving_amd.serving.shm_transport import ShmClient
    from clearml_serving_amd.store import ServingStore

    store_root = str(tmp_path / "store")
    store = ServingStore(store_root)
    proc = ModelRequestProcessor(store=store, name="nsab",
                                 force_create=True)
    card = tmp_path / "card.json"
    card.write_text(_json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 512, "device": "cpu"}))
    rec = store.register_model(name="t", project="p", path=str(card))
    proc.add_endpoint(ModelEndpoint(engine_type="llm", serving_url="tl",
                                    model_id=rec.model_id))
    proc.serialize()

    prefix = "/cmls_nsab_{}".format(os.getpid())
    owner = EngineOwner(store_root=store_root, session_id=proc.get_id(),
                        prefix=prefix, owner_idx=0, n_workers=1,
                        ring_bytes=1 << 20)
    client = ShmClient(prefix, 0, 1, ring_bytes=1 << 20)

    async def main():
        serve = asyncio.get_running_loop().create_task(owner.serve())
        payload = {"__serve_type__": "v1_chat_completions",
                   "__body__": {"messages": [{"role": "user",
                                              "content": "hi"}],
                                "max_tokens": 400, "temperature": 0,
                                "ignore_eos": True}}
        task = asyncio.ensure_future(client.infer("tl", payload))
        await asyncio.sleep(0.3)  # generation well in flight
        task.cancel()
        try:
            await task
        except asyncio.CancelledError:
            pass
        eng = owner._engines["tl"]._engine
        for _ in range(200):
            await asyncio.sleep(0.05)
            if owner.stats["aborts"] >= 1 and not eng.running \
                    and not eng.waiting:
                break
        assert owner.stats["aborts"] >= 1
        assert eng.stats["generated_tokens"] < 300
        assert not eng.running and not eng.waiting
        serve.cancel()

    loop = asyncio.new_event_loop()
    try:
        loop.run_until_complete(main())
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()
        client.close()
        owner.close()
