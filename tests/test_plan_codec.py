"""TP step-plan codec: every plan mode round-trips through the fixed int32
buffer (the single-broadcast replacement for pickle plan dicts)."""

import torch

from clearml_serving_amd.engines.llm.engine import LlmEngineConfig
from clearml_serving_amd.engines.llm.plan_codec import PlanCodec


def codec():
    cfg = LlmEngineConfig(preset="llama-tiny", max_num_seqs=8,
                          max_model_len=128, block_size=16,
                          max_prefill_tokens=256, prefill_chunk=64)
    return PlanCodec(cfg, torch.device("cpu"))


def roundtrip(c, plan):
    buf = c.encode(plan)
    assert buf is not None
    return c.decode(buf)


def test_decode_plan_roundtrip():
    c = codec()
    plan = {"mode": "decode", "tokens": [5, 9, 11], "positions": [3, 7, 2],
            "slots": [48, 112, 33], "seq_lens": [4, 8, 3],
            "blocks": [[3], [7, 1], [2]],
            "sample": [(0.0, 0, 1.0, 123), (0.7, 40, 0.9, 123),
                       (1.0, 0, 1.0, 123)]}
    out = roundtrip(c, plan)
    assert out["mode"] == "decode"
    for k in ("tokens", "positions", "slots", "seq_lens", "blocks"):
        assert out[k] == plan[k], k
    for got, want in zip(out["sample"], plan["sample"]):
        assert got[1] == want[1] and got[3] == want[3]
        assert abs(got[0] - want[0]) < 1e-6 and abs(got[2] - want[2]) < 1e-6


def test_prefill_plan_roundtrip():
    c = codec()
    plan = {"mode": "prefill",
            "prompts": [[1, 2, 3], [4, 5, 6, 7, 8]],
            "slots": [[0, 1, 2], [16, 17, 18, 19, 20]],
            "sample": [(1.0, 0, 1.0, 7), (1.0, 0, 1.0, 7)]}
    out = roundtrip(c, plan)
    assert out["prompts"] == plan["prompts"]
    assert out["slots"] == plan["slots"]
    assert len(out["sample"]) == 2


def test_chunk_plan_roundtrip():
    c = codec()
    plan = {"mode": "chunk",
            "tokens": [[9, 8, 7], [1, 2]],
            "slots": [[32, 33, 34], [48, 49]],
            "starts": [4, 0], "kv_lens": [7, 2],
            "complete": [True, False],
            "blocks": [[2, 3], [3]],
            "sample": [(0.0, 0, 1.0, 42)]}
    out = roundtrip(c, plan)
    for k in ("tokens", "slots", "starts", "kv_lens", "complete", "blocks"):
        assert out[k] == plan[k], k
    assert len(out["sample"]) == 1


def test_embed_and_stop_plans():
    c = codec()
    out = roundtrip(c, {"mode": "embed", "prompts": [[1, 2], [3]],
                        "normalize": False})
    assert out == {"mode": "embed", "prompts": [[1, 2], [3]],
                   "normalize": False}
    assert roundtrip(c, {"mode": "stop"})["mode"] == "stop"
    assert roundtrip(c, None)["mode"] == "stop"


def test_overflow_falls_back_to_object():
    c = codec()
    huge = {"mode": "embed",
            "prompts": [[1] * 120 for _ in range(200)]}  # > capacity
    assert c.encode(huge) is None
    # the marker buffer decodes to the object-fallback sentinel
    assert c.decode(c.mark_object()) is None


def test_argmax_sharded_single_rank_matches_torch():
    from clearml_serving_amd.parallel import tp

    logits = torch.randn(5, 64)
    got = tp.argmax_sharded(logits, 0)
    assert torch.equal(got, logits.argmax(dim=-1))
    # negative-heavy rows (IEEE sortable-key edge)
    logits2 = -torch.rand(4, 32) - 1.0
    assert torch.equal(tp.argmax_sharded(logits2, 0),
                       logits2.argmax(dim=-1))
    # vocab offset applies
    assert torch.equal(tp.argmax_sharded(logits, 100),
                       logits.argmax(dim=-1) + 100)


def test_gumbel_sharded_single_rank_distribution():
    """world=1 Gumbel-max must sample from softmax(logits/T): check the
    empirical distribution against the exact softmax on a tiny vocab."""
    from clearml_serving_amd.parallel import tp

    torch.manual_seed(0)
    logits = torch.tensor([[2.0, 1.0, 0.0, -1.0]])
    counts = torch.zeros(4)
    n = 4000
    for i in range(n):
        idx = tp.sample_gumbel_sharded(logits, 0, 1.0, [i])
        counts[idx.item()] += 1
    probs = torch.softmax(logits[0], dim=-1)
    assert torch.allclose(counts / n, probs, atol=0.03)
