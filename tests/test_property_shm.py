"""Property-based tests (hypothesis) for the SHM ring record format and the
TP plan codec -- randomized sequences beyond the example-based tests."""

import uuid

import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from clearml_serving_amd.serving import shm_transport as tr


@settings(max_examples=40, deadline=None)
@given(st.lists(st.binary(min_size=0, max_size=700), min_size=1,
                max_size=60))
def test_ring_preserves_arbitrary_record_sequences(payloads):
    name = "/cmls_prop_{}".format(uuid.uuid4().hex[:10])
    w = tr.make_ring(name, 1 << 14, True)
    r = tr.make_ring(name, 0, False)
    try:
        got = []
        i = 0
        while i < len(payloads):
            # push a few, drain a few -- random interleave shapes
            burst = payloads[i:i + 7]
            for p in burst:
                while not w.push(p):
                    got.extend(r.drain(64))
            i += len(burst)
            got.extend(r.drain(3))
        while True:
            batch = r.drain(64)
            if not batch:
                break
            got.extend(batch)
        assert got == payloads
    finally:
        w.close()
        r.close()
        tr.unlink_ring(name)


@settings(max_examples=30, deadline=None)
@given(
    b=st.integers(min_value=1, max_value=8),
    seed=st.integers(min_value=0, max_value=2**31 - 1),
    data=st.data(),
)
def test_plan_codec_decode_roundtrip_random(b, seed, data):
    from clearml_serving_amd.engines.llm.engine import LlmEngineConfig
    from clearml_serving_amd.engines.llm.plan_codec import PlanCodec

    cfg = LlmEngineConfig(preset="llama-tiny", max_num_seqs=8,
                          max_model_len=256, block_size=16,
                          max_prefill_tokens=128, prefill_chunk=64)
    codec = PlanCodec(cfg, torch.device("cpu"))
    plan = {
        "mode": "decode",
        "tokens": [data.draw(st.integers(0, 511)) for _ in range(b)],
        "positions": [data.draw(st.integers(0, 255)) for _ in range(b)],
        "slots": [data.draw(st.integers(0, 4095)) for _ in range(b)],
        "seq_lens": [data.draw(st.integers(1, 256)) for _ in range(b)],
        "blocks": [[data.draw(st.integers(0, 255))
                    for _ in range(data.draw(st.integers(1, 16)))]
                   for _ in range(b)],
        "sample": [(data.draw(st.floats(0, 4, allow_nan=False,
                                        allow_subnormal=False, width=32)),
                    data.draw(st.integers(0, 100)),
                    data.draw(st.floats(0.0625, 1.0, allow_nan=False,
                                        allow_subnormal=False, width=32)),
                    seed) for _ in range(b)],
    }
    buf = codec.encode(plan)
    assert buf is not None
    out = codec.decode(buf)
    for k in ("tokens", "positions", "slots", "seq_lens", "blocks"):
        assert out[k] == plan[k], k
    for got, want in zip(out["sample"], plan["sample"]):
        assert got[1] == want[1] and got[3] == want[3]
        assert got[0] == pytest.approx(want[0], abs=1e-6)
        assert got[2] == pytest.approx(want[2], abs=1e-6)


@settings(max_examples=60, deadline=None)
@given(
    url=st.text(min_size=0, max_size=40),
    req_id=st.integers(min_value=0, max_value=2**63 - 1),
    data=st.one_of(
        # tensor payloads: dict of arrays with random dtypes/shapes
        st.dictionaries(
            st.text(min_size=1, max_size=10),
            st.tuples(
                st.sampled_from(["float32", "int64", "int32", "uint8",
                                 "bool", "float16"]),
                st.lists(st.integers(1, 5), min_size=0, max_size=3)),
            min_size=1, max_size=4),
        # object payloads (pickle fallback)
        st.recursive(
            st.one_of(st.none(), st.booleans(), st.integers(),
                      st.floats(allow_nan=False), st.text(max_size=10)),
            lambda c: st.lists(c, max_size=3)
            | st.dictionaries(st.text(max_size=5), c, max_size=3),
            max_leaves=10),
    ))
def test_request_framing_roundtrip_fuzz(url, req_id, data):
    """pack_request/unpack_request roundtrip over arbitrary tensor maps
    and JSON-ish objects: ids, urls, names, dtypes, shapes and values all
    survive."""
    import numpy as np

    from clearml_serving_amd.serving import shm_transport as st_mod

    if isinstance(data, dict) and data and all(
            isinstance(v, tuple) for v in data.values()):
        arrays = {k: (np.zeros(shape, dtype=dt) + 1).astype(dt)
                  for k, (dt, shape) in data.items()}
        buf = st_mod.pack_request(req_id, url, arrays)
        rid, u, out = st_mod.unpack_request(buf)
        assert rid == req_id and u == url
        if len(arrays) == 1 and "" in arrays:
            out = {"": out}
        assert set(out) == set(arrays)
        for k in arrays:
            assert out[k].dtype == arrays[k].dtype
            assert out[k].shape == arrays[k].shape
            np.testing.assert_array_equal(out[k], arrays[k])
    else:
        buf = st_mod.pack_request(req_id, url, data)
        rid, u, out = st_mod.unpack_request(buf)
        assert rid == req_id and u == url
        import numpy as _np

        if isinstance(out, _np.ndarray):
            # numeric (possibly empty) lists intentionally convert to
            # arrays on the tensor path; a single ""-keyed dict entry is
            # the positional-array convention
            ref = (data[""] if isinstance(data, dict)
                   and set(data) == {""} else data)
            _np.testing.assert_allclose(
                out.astype(_np.float64),
                _np.asarray(ref, dtype=_np.float64))
        elif isinstance(out, dict) and any(
                isinstance(v, _np.ndarray) for v in out.values()):
            # dicts whose values are all numeric lists ride the tensor
            # path: keys preserved, values as arrays
            assert set(out) == set(data)
            for k in out:
                _np.testing.assert_allclose(
                    _np.asarray(out[k], dtype=_np.float64),
                    _np.asarray(data[k], dtype=_np.float64))
        else:
            assert out == data or (out is None and data is None)
