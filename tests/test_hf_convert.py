"""Architecture correctness vs the canonical HuggingFace implementations.

Random-init transformers models (no downloads) are converted into the
native layouts; forward outputs must match to fp32 tolerance -- this pins
the native BERT/Llama architectures to the reference semantics, not just
to themselves.
"""

import pytest
import torch

transformers = pytest.importorskip("transformers")

from clearml_serving_amd.models.bert import BertEncoder  # noqa: E402
from clearml_serving_amd.models.convert import (  # noqa: E402
    convert_hf_bert,
    convert_hf_llama,
)
from clearml_serving_amd.models.llama import (  # noqa: E402
    LlamaConfig,
    LlamaForCausalLM,
)


def test_bert_matches_transformers():
    from transformers import BertConfig, BertForSequenceClassification

    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=211, hidden_size=64, num_hidden_layers=2,
                     num_attention_heads=4, intermediate_size=128,
                     max_position_embeddings=64, num_labels=3,
                     hidden_dropout_prob=0.0,
                     attention_probs_dropout_prob=0.0)
    hf = BertForSequenceClassification(cfg).eval()

    ours = BertEncoder(vocab_size=211, hidden=64, layers=2, heads=4,
                       intermediate=128, max_pos=64, num_labels=3).eval()
    ours.load_state_dict(convert_hf_bert(hf.state_dict(), num_layers=2))

    ids = torch.randint(0, 211, (2, 16))
    mask = torch.ones(2, 16, dtype=torch.int64)
    with torch.inference_mode():
        ref = hf(input_ids=ids, attention_mask=mask).logits
        got = ours({"input_ids": ids, "attention_mask": mask})
    torch.testing.assert_close(got, ref, atol=2e-4, rtol=2e-4)


def test_bert_matches_transformers_with_padding():
    from transformers import BertConfig, BertForSequenceClassification

    torch.manual_seed(1)
    cfg = BertConfig(vocab_size=100, hidden_size=64, num_hidden_layers=1,
                     num_attention_heads=2, intermediate_size=96,
                     max_position_embeddings=32, num_labels=2,
                     hidden_dropout_prob=0.0,
                     attention_probs_dropout_prob=0.0)
    hf = BertForSequenceClassification(cfg).eval()
    ours = BertEncoder(vocab_size=100, hidden=64, layers=1, heads=2,
                       intermediate=96, max_pos=32, num_labels=2).eval()
    ours.load_state_dict(convert_hf_bert(hf.state_dict(), num_layers=1))

    ids = torch.randint(0, 100, (2, 12))
    mask = torch.tensor([[1] * 12, [1] * 5 + [0] * 7], dtype=torch.int64)
    with torch.inference_mode():
        ref = hf(input_ids=ids, attention_mask=mask).logits
        got = ours({"input_ids": ids, "attention_mask": mask})
    torch.testing.assert_close(got, ref, atol=2e-4, rtol=2e-4)


def test_llama_matches_transformers():
    from transformers import LlamaConfig as HfLlamaConfig
    from transformers import LlamaForCausalLM as HfLlama

    torch.manual_seed(2)
    hf_cfg = HfLlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=112,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0, rms_norm_eps=1e-5,
        attention_dropout=0.0, tie_word_embeddings=False,
        attn_implementation="eager")
    hf = HfLlama(hf_cfg).eval()

    cfg = LlamaConfig(vocab_size=128, hidden=64, layers=2, heads=4,
                      kv_heads=2, intermediate=112, rope_theta=10000.0,
                      rms_eps=1e-5, max_position=64)
    ours = LlamaForCausalLM(cfg).eval()
    ours.load_state_dict(convert_hf_llama(hf.state_dict(), num_layers=2))

    t = 10
    ids = torch.randint(0, 128, (t,))
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        ref = hf(input_ids=ids[None]).logits[0]
        got = ours(ids, positions, kv_caches=None, attn_ctx=attn_ctx)
    torch.testing.assert_close(got, ref, atol=3e-4, rtol=3e-4)


def test_qwen2_matches_transformers():
    """Qwen2 = llama architecture + QKV biases + ChatML; the converter and
    the native qkv_bias path must reproduce transformers' logits."""
    from transformers import Qwen2Config as HfQwen2Config
    from transformers import Qwen2ForCausalLM as HfQwen2

    torch.manual_seed(4)
    hf_cfg = HfQwen2Config(
        vocab_size=160, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=10000.0, rms_norm_eps=1e-6,
        attention_dropout=0.0, tie_word_embeddings=False, use_cache=False,
        attn_implementation="eager")
    hf = HfQwen2(hf_cfg).eval()
    # Qwen2's defining deviation from llama: q/k/v projections carry biases
    assert hf.state_dict()["model.layers.0.self_attn.q_proj.bias"] is not None

    cfg = LlamaConfig(vocab_size=160, hidden=64, layers=2, heads=4,
                      kv_heads=2, intermediate=96, rope_theta=10000.0,
                      rms_eps=1e-6, max_position=64, qkv_bias=True)
    ours = LlamaForCausalLM(cfg).eval()
    from clearml_serving_amd.models.convert import convert_hf_qwen2

    ours.load_state_dict(convert_hf_qwen2(hf.state_dict(), num_layers=2))

    t = 12
    ids = torch.randint(0, 160, (t,))
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        ref = hf(input_ids=ids[None]).logits[0]
        got = ours(ids, positions, kv_caches=None, attn_ctx=attn_ctx)
    torch.testing.assert_close(got, ref, atol=3e-4, rtol=3e-4)


def test_qwen2_tp_shard_roundtrip():
    """shard_llama_weights handles the Qwen2 qkv bias: rank shards
    reassemble to the full tensors for every world size that divides the
    head counts."""
    from clearml_serving_amd.parallel.tp import shard_llama_weights

    torch.manual_seed(5)
    cfg = LlamaConfig(vocab_size=64, hidden=64, layers=1, heads=4,
                      kv_heads=2, intermediate=64, qkv_bias=True)
    full = LlamaForCausalLM(cfg)
    state = full.state_dict()
    world = 2
    shards = [shard_llama_weights(state, cfg, r, world)
              for r in range(world)]
    hd = cfg.head_dim
    hpr, kvpr = cfg.heads // world, cfg.kv_heads // world
    for key in ("layers.0.qkv.weight", "layers.0.qkv.bias"):
        qs, ks, vs = [], [], []
        for r in range(world):
            s = shards[r][key]
            qs.append(s[:hpr * hd])
            ks.append(s[hpr * hd:hpr * hd + kvpr * hd])
            vs.append(s[hpr * hd + kvpr * hd:])
        rebuilt = torch.cat(qs + ks + vs, dim=0)
        torch.testing.assert_close(rebuilt, state[key])


def test_gpt2_matches_transformers():
    """GPT-2 through the native pre-LN decoder + converter (HF Conv1D
    weights transpose): logits must match transformers.GPT2LMHeadModel."""
    from transformers import GPT2Config as HfGPT2Config
    from transformers import GPT2LMHeadModel

    from clearml_serving_amd.models.convert import convert_hf_gpt2
    from clearml_serving_amd.models.gpt2 import (GPT2Config,
                                                 GPT2ForCausalLM)

    torch.manual_seed(6)
    hf_cfg = HfGPT2Config(
        vocab_size=160, n_embd=64, n_layer=2, n_head=4, n_inner=96,
        n_positions=64, resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
        attn_implementation="eager")
    hf = GPT2LMHeadModel(hf_cfg).eval()

    cfg = GPT2Config(vocab_size=160, hidden=64, layers=2, heads=4,
                     intermediate=96, max_position=64)
    ours = GPT2ForCausalLM(cfg).eval()
    ours.load_state_dict(convert_hf_gpt2(hf.state_dict(), num_layers=2))

    t = 12
    ids = torch.randint(0, 160, (t,))
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        ref = hf(input_ids=ids[None]).logits[0]
        got = ours(ids, positions, kv_caches=None, attn_ctx=attn_ctx)
    torch.testing.assert_close(got, ref, atol=3e-4, rtol=3e-4)


def test_convert_hf_auto_detection(tmp_path):
    """convert_hf_auto detects llama / qwen2 / gpt2 / bert key layouts
    (and passes native dicts through), so raw HF checkpoints load via
    models.load_weights without a manual conversion step."""
    from transformers import GPT2Config as HfGPT2Config
    from transformers import GPT2LMHeadModel
    from transformers import LlamaConfig as HfLlamaConfig
    from transformers import LlamaForCausalLM as HfLlama

    from clearml_serving_amd.models import load_weights
    from clearml_serving_amd.models.convert import convert_hf_auto
    from clearml_serving_amd.models.gpt2 import (GPT2Config,
                                                 GPT2ForCausalLM)

    torch.manual_seed(8)
    hf_l = HfLlama(HfLlamaConfig(
        vocab_size=96, hidden_size=32, intermediate_size=48,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, tie_word_embeddings=False,
        attn_implementation="eager")).eval()
    ours_l = LlamaForCausalLM(LlamaConfig(
        vocab_size=96, hidden=32, layers=2, heads=4, kv_heads=2,
        intermediate=48, max_position=32, rope_theta=10000.0,
        rms_eps=1e-6)).eval()
    ours_l.load_state_dict(convert_hf_auto(hf_l.state_dict()))

    hf_g = GPT2LMHeadModel(HfGPT2Config(
        vocab_size=96, n_embd=32, n_layer=2, n_head=4, n_inner=48,
        n_positions=32, resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
        attn_implementation="eager")).eval()
    ours_g = GPT2ForCausalLM(GPT2Config(
        vocab_size=96, hidden=32, layers=2, heads=4, intermediate=48,
        max_position=32)).eval()

    # end to end through load_weights on a saved HF state dict
    pt = tmp_path / "hf_gpt2.pt"
    torch.save(hf_g.state_dict(), str(pt))
    load_weights(ours_g, str(pt))

    t = 8
    ids = torch.randint(0, 96, (t,))
    positions = torch.arange(t, dtype=torch.int32)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32)}
    with torch.inference_mode():
        torch.testing.assert_close(
            ours_g(ids, positions, kv_caches=None, attn_ctx=attn_ctx),
            hf_g(input_ids=ids[None]).logits[0], atol=3e-4, rtol=3e-4)
        torch.testing.assert_close(
            ours_l(ids, positions, kv_caches=None, attn_ctx=attn_ctx),
            hf_l(input_ids=ids[None]).logits[0], atol=3e-4, rtol=3e-4)

    # native dict passes through unchanged
    native = ours_g.state_dict()
    assert set(convert_hf_auto(native).keys()) == set(native.keys())


def test_gpt2_hf_checkpoint_served_end_to_end(tmp_path):
    """A saved HF GPT-2 checkpoint (raw key layout) drops into an engine
    model card: weights auto-convert at load, and greedy continuation
    matches transformers.generate token for token."""
    import asyncio
    import json

    from safetensors.torch import save_file
    from transformers import GPT2Config as HfGPT2Config
    from transformers import GPT2LMHeadModel

    from clearml_serving_amd.engines.llm.engine import (LlmEngine,
                                                        LlmEngineConfig,
                                                        SamplingParams)

    torch.manual_seed(10)
    hf = GPT2LMHeadModel(HfGPT2Config(
        vocab_size=128, n_embd=64, n_layer=2, n_head=2, n_inner=128,
        n_positions=128, resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
        attn_implementation="eager")).eval()

    mdir = tmp_path / "gpt2_ckpt"
    mdir.mkdir()
    # tied lm_head: save_file refuses shared tensors -- materialize a copy
    sd = {k: v.clone() for k, v in hf.state_dict().items()}
    save_file(sd, str(mdir / "model.safetensors"))
    (mdir / "model_card.json").write_text(json.dumps({
        "arch": "gpt2", "preset": "gpt2-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu",
        "overrides": {"vocab_size": 128, "hidden": 64, "layers": 2,
                      "heads": 2, "intermediate": 128,
                      "max_position": 128}}))

    cfg = LlmEngineConfig.from_aux(str(mdir), {})
    assert cfg.arch == "gpt2" and cfg.weights
    eng = LlmEngine(cfg)
    eng.start()

    prompt_ids = [5, 17, 31, 44]

    async def gen():
        seq = await eng.add_request(list(prompt_ids), SamplingParams(
            temperature=0.0, max_tokens=8, ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    loop = asyncio.new_event_loop()
    try:
        got = loop.run_until_complete(gen())
    finally:
        loop.close()

    with torch.inference_mode():
        ref = hf.generate(torch.tensor([prompt_ids]), max_new_tokens=8,
                          do_sample=False, pad_token_id=0)[0].tolist()
    assert got == ref[len(prompt_ids):], (got, ref)


def test_qwen2_hf_checkpoint_served_end_to_end(tmp_path):
    """A saved HF Qwen2 checkpoint (QKV biases) through the engine:
    auto-converted at load, greedy continuation == transformers.generate."""
    import asyncio
    import json

    from safetensors.torch import save_file
    from transformers import Qwen2Config as HfQwen2Config
    from transformers import Qwen2ForCausalLM

    from clearml_serving_amd.engines.llm.engine import (LlmEngine,
                                                        LlmEngineConfig,
                                                        SamplingParams)

    torch.manual_seed(14)
    hf = Qwen2ForCausalLM(HfQwen2Config(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rope_theta=10000.0, rms_norm_eps=1e-6,
        attention_dropout=0.0, tie_word_embeddings=False, use_cache=False,
        attn_implementation="eager")).eval()

    mdir = tmp_path / "qwen2_ckpt"
    mdir.mkdir()
    save_file({k: v.clone() for k, v in hf.state_dict().items()},
              str(mdir / "model.safetensors"))
    (mdir / "model_card.json").write_text(json.dumps({
        "arch": "llama", "preset": "llama-tiny", "num_kv_blocks": 64,
        "block_size": 16, "max_model_len": 128, "device": "cpu",
        "overrides": {"vocab_size": 128, "hidden": 64, "layers": 2,
                      "heads": 4, "kv_heads": 2, "intermediate": 96,
                      "rope_theta": 10000.0, "rms_eps": 1e-6,
                      "max_position": 128, "qkv_bias": True}}))

    cfg = LlmEngineConfig.from_aux(str(mdir), {})
    eng = LlmEngine(cfg)
    eng.start()
    assert eng.model.layers[0].qkv.bias is not None

    prompt_ids = [7, 21, 42, 63, 11]

    async def gen():
        seq = await eng.add_request(list(prompt_ids), SamplingParams(
            temperature=0.0, max_tokens=8, ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    loop = asyncio.new_event_loop()
    try:
        got = loop.run_until_complete(gen())
    finally:
        loop.close()

    with torch.inference_mode():
        ref = hf.generate(torch.tensor([prompt_ids]), max_new_tokens=8,
                          do_sample=False, pad_token_id=0)[0].tolist()
    assert got == ref[len(prompt_ids):], (got, ref)
