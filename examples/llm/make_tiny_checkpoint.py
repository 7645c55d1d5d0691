#!/usr/bin/env python3
"""Train a tiny HF-format llama checkpoint IN-REPO and package it for the
native LLM engine (the real-checkpoint serving proof: this environment has
no model hub, so the "real trained model" is produced here).

Produces a model directory the serving stack loads end to end:
    model.safetensors   native-layout weights (converted from the HF
                        checkpoint via models.convert.convert_hf_llama)
    hf_model.safetensors  the original HF-layout weights (parity oracle:
                        tests load them into transformers.LlamaForCausalLM)
    tokenizer.json      BPE tokenizer trained here, with the llama-3
                        special tokens (so the engine selects the real
                        llama-3 chat template)
    model_card.json     arch/config for the engine

The model is TRAINED (not random): it memorizes a pangram continuation,
which the tests then reproduce greedily through the full serving path.

    python examples/llm/make_tiny_checkpoint.py OUTDIR [--steps 300]
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

PANGRAM = "the quick brown fox jumps over the lazy dog"
CORPUS = [
    PANGRAM,
    "pack my box with five dozen liquor jugs",
    "how vexingly quick daft zebras jump",
    "sphinx of black quartz judge my vow",
    "the five boxing wizards jump quickly",
]
SPECIALS = ["<|begin_of_text|>", "<|end_of_text|>", "<|start_header_id|>",
            "<|end_header_id|>", "<|eot_id|>"]


def train_tokenizer(outdir: str):
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.BpeTrainer(
        vocab_size=448, special_tokens=["<unk>"] + SPECIALS)
    tok.train_from_iterator(CORPUS * 4, trainer)
    path = os.path.join(outdir, "tokenizer.json")
    tok.save(path)
    return path


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("outdir")
    ap.add_argument("--steps", type=int, default=300)
    args = ap.parse_args()
    os.makedirs(args.outdir, exist_ok=True)

    import torch
    from safetensors.torch import save_file
    from transformers import LlamaConfig as HfLlamaConfig
    from transformers import LlamaForCausalLM as HfLlama

    from clearml_serving_amd.models.convert import convert_hf_llama

    tok_path = train_tokenizer(args.outdir)
    from tokenizers import Tokenizer

    tok = Tokenizer.from_file(tok_path)
    vocab = tok.get_vocab_size()
    vocab_padded = ((vocab + 15) // 16) * 16  # native lm_head wants %16

    torch.manual_seed(0)
    cfg = HfLlamaConfig(
        vocab_size=vocab_padded, hidden_size=256, num_hidden_layers=2,
        num_attention_heads=4, num_key_value_heads=2, intermediate_size=512,
        max_position_embeddings=256, rope_theta=10000.0, rms_norm_eps=1e-5,
        tie_word_embeddings=False, attention_bias=False, mlp_bias=False)
    model = HfLlama(cfg)

    # train: next-token prediction over the pangram corpus until the model
    # reproduces the memorized continuation greedily
    ids = [torch.tensor(tok.encode(s).ids) for s in CORPUS]
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    model.train()
    for step in range(args.steps):
        loss_total = 0.0
        for seq in ids:
            x = seq[:-1].unsqueeze(0)
            y = seq[1:].unsqueeze(0)
            out = model(input_ids=x).logits
            loss = torch.nn.functional.cross_entropy(
                out.reshape(-1, vocab_padded), y.reshape(-1))
            opt.zero_grad()
            loss.backward()
            opt.step()
            loss_total += float(loss)
        if step % 50 == 0:
            print("step {} loss {:.4f}".format(step, loss_total), flush=True)
    model.eval()

    # verify memorization (greedy continuation of the pangram prefix)
    prefix = tok.encode("the quick brown fox").ids
    want = tok.encode(PANGRAM).ids
    seq = list(prefix)
    with torch.inference_mode():
        for _ in range(len(want) - len(prefix)):
            logits = model(input_ids=torch.tensor([seq])).logits
            seq.append(int(logits[0, -1].argmax()))
    memorized = seq == want
    print("memorized:", memorized)

    hf_state = {k: v.contiguous() for k, v in model.state_dict().items()
                if not k.endswith("rotary_emb.inv_freq")}
    save_file(hf_state, os.path.join(args.outdir, "hf_model.safetensors"))
    native = convert_hf_llama(hf_state, num_layers=2)
    save_file({k: v.contiguous() for k, v in native.items()},
              os.path.join(args.outdir, "model.safetensors"))
    with open(os.path.join(args.outdir, "model_card.json"), "wt") as f:
        json.dump({
            "arch": "llama", "preset": "llama-tiny",
            "overrides": {"vocab_size": vocab_padded, "hidden": 256,
                          "layers": 2, "heads": 4, "kv_heads": 2,
                          "intermediate": 512, "rope_theta": 10000.0,
                          "max_position": 256},
            "max_model_len": 192, "block_size": 16, "num_kv_blocks": 128,
            "memorized": memorized,
        }, f, indent=1)
    print("checkpoint written to", args.outdir)


if __name__ == "__main__":
    main()
