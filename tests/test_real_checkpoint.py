"""Real-checkpoint serving proof (VERDICT round-1 item 6).

examples/llm/make_tiny_checkpoint.py TRAINS a tiny HF-format llama (via
transformers) until it memorizes a pangram, converts it through
models/convert.py, and packages tokenizer.json + model card. These tests
serve that trained checkpoint through the native engine and demand the
memorized continuation -- random weights cannot pass them.
"""

import asyncio
import json
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
PANGRAM = "the quick brown fox jumps over the lazy dog"


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()


@pytest.fixture(scope="session")
def checkpoint_dir(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("tinyllm"))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable,
         os.path.join(REPO, "examples", "llm", "make_tiny_checkpoint.py"),
         out, "--steps", "220"],
        capture_output=True, text=True, timeout=900, env=env)
    assert r.returncode == 0, r.stdout + r.stderr
    card = json.load(open(os.path.join(out, "model_card.json")))
    assert card["memorized"], "training did not converge; raise --steps"
    return out


def _engine_for(checkpoint_dir, device):
    from clearml_serving_amd.engines.llm.engine import (
        LlmEngine, LlmEngineConfig)

    cfg = LlmEngineConfig.from_aux(checkpoint_dir, {"device": device})
    eng = LlmEngine(cfg)
    eng.start()
    return eng


def _greedy(eng, prompt, n):
    from clearml_serving_amd.engines.llm.engine import SamplingParams

    async def go():
        toks = []
        async for item in eng.generate(
                prompt, SamplingParams(temperature=0.0, max_tokens=n,
                                       ignore_eos=True)):
            toks.extend(item["token_ids"])
        return toks

    return run(go())


def test_trained_checkpoint_serves_memorized_text(checkpoint_dir):
    """Engine (CPU) loads safetensors + tokenizer.json and greedily
    reproduces the TRAINED continuation -- end-to-end proof that the
    HF-convert + weight-loading + tokenizer path carries real checkpoints."""
    eng = _engine_for(checkpoint_dir, "cpu")
    # real tokenizer loaded (not the byte fallback), llama-3 template active
    assert eng.tokenizer.__class__.__name__ == "HfTokenizer"
    assert eng.tokenizer.is_llama3
    prefix = "the quick brown fox"
    want_ids = eng.tokenizer.encode(PANGRAM)
    prefix_ids = eng.tokenizer.encode(prefix)
    out = _greedy(eng, prefix, len(want_ids) - len(prefix_ids))
    assert prefix_ids + out == want_ids
    assert PANGRAM.endswith(eng.tokenizer.decode(out).strip()) or \
        eng.tokenizer.decode(out).strip() in PANGRAM
    eng.stop()


def test_engine_logits_match_transformers(checkpoint_dir):
    """Numerics oracle: the served model's fp32 logits equal transformers'
    LlamaForCausalLM on the SAME trained weights."""
    from safetensors.torch import load_file
    from transformers import LlamaConfig as HfLlamaConfig
    from transformers import LlamaForCausalLM as HfLlama

    eng = _engine_for(checkpoint_dir, "cpu")
    hf_state = load_file(os.path.join(checkpoint_dir, "hf_model.safetensors"))
    vocab = hf_state["model.embed_tokens.weight"].shape[0]
    hf = HfLlama(HfLlamaConfig(
        vocab_size=vocab, hidden_size=256, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, intermediate_size=512,
        max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-5,
        tie_word_embeddings=False, attention_bias=False, mlp_bias=False))
    hf.load_state_dict(hf_state, strict=False)
    hf.eval()

    ids = eng.tokenizer.encode(PANGRAM)
    tokens = torch.tensor(ids, dtype=torch.long)
    positions = torch.arange(len(ids), dtype=torch.int32)
    ctx = {"mode": "prefill", "batch": 1, "seq": len(ids),
           "seq_lens": torch.tensor([len(ids)], dtype=torch.int32),
           "slot_mapping": torch.full((len(ids),), -1, dtype=torch.int32)}
    with torch.inference_mode():
        ours = eng.model(tokens, positions, kv_caches=None, attn_ctx=ctx,
                         last_token_idx=None)
        theirs = hf(input_ids=tokens.unsqueeze(0)).logits[0]
    torch.testing.assert_close(ours.view(-1, vocab), theirs,
                               atol=5e-4, rtol=5e-4)
    eng.stop()


@pytest.mark.gpu
def test_trained_checkpoint_serves_on_gpu(checkpoint_dir):
    """The same trained checkpoint on the HIP kernel path (bf16): memorized
    continuation reproduced greedily, llama-3 chat template active."""
    eng = _engine_for(checkpoint_dir, "cuda:0")
    assert eng.tokenizer.is_llama3
    prefix_ids = eng.tokenizer.encode("the quick brown fox")
    want_ids = eng.tokenizer.encode(PANGRAM)
    out = _greedy(eng, "the quick brown fox", len(want_ids) - len(prefix_ids))
    assert prefix_ids + out == want_ids, (
        "bf16 GPU serving did not reproduce the trained continuation")
    # chat template formats with the real llama-3 header tokens
    p = eng._chat_prompt([{"role": "user", "content": "hi"}])
    assert p.startswith("<|begin_of_text|>")
    eng.stop()
