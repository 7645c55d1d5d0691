"""HuggingFace checkpoint conversion into the native model layouts.

The environment has no model hub, but the converters are validated against
random-init ``transformers`` models (same architecture, same numerics) in
tests/test_hf_convert.py. Use for real deployments:

    state = convert_hf_bert(load_file("model.safetensors"), num_layers=12)
    save_file(state, "bert_native.safetensors")   # -> model card "weights"
"""

from typing import Dict

import torch


def convert_hf_bert(hf: Dict[str, torch.Tensor],
                    num_layers: int) -> Dict[str, torch.Tensor]:
    """transformers BertForSequenceClassification -> models.bert.BertEncoder."""

    def g(key):
        # accept both "bert.<...>" and bare "<...>" prefixes
        return hf.get("bert." + key, hf.get(key))

    out = {
        "word_emb.weight": g("embeddings.word_embeddings.weight"),
        "pos_emb.weight": g("embeddings.position_embeddings.weight"),
        "type_emb.weight": g("embeddings.token_type_embeddings.weight"),
        "emb_ln_w": g("embeddings.LayerNorm.weight"),
        "emb_ln_b": g("embeddings.LayerNorm.bias"),
        "pooler.weight": g("pooler.dense.weight"),
        "pooler.bias": g("pooler.dense.bias"),
        "classifier.weight": hf["classifier.weight"],
        "classifier.bias": hf["classifier.bias"],
    }
    for i in range(num_layers):
        p = "encoder.layer.{}.".format(i)
        q_w, k_w, v_w = (g(p + "attention.self.{}.weight".format(n))
                         for n in ("query", "key", "value"))
        q_b, k_b, v_b = (g(p + "attention.self.{}.bias".format(n))
                         for n in ("query", "key", "value"))
        o = "layers.{}.".format(i)
        out[o + "attn.qkv.weight"] = torch.cat([q_w, k_w, v_w], dim=0)
        out[o + "attn.qkv.bias"] = torch.cat([q_b, k_b, v_b], dim=0)
        out[o + "attn.out.weight"] = g(p + "attention.output.dense.weight")
        out[o + "attn.out.bias"] = g(p + "attention.output.dense.bias")
        out[o + "ln1_w"] = g(p + "attention.output.LayerNorm.weight")
        out[o + "ln1_b"] = g(p + "attention.output.LayerNorm.bias")
        out[o + "fc1.weight"] = g(p + "intermediate.dense.weight")
        out[o + "fc1_bias"] = g(p + "intermediate.dense.bias")
        out[o + "fc2.weight"] = g(p + "output.dense.weight")
        out[o + "fc2.bias"] = g(p + "output.dense.bias")
        out[o + "ln2_w"] = g(p + "output.LayerNorm.weight")
        out[o + "ln2_b"] = g(p + "output.LayerNorm.bias")
    return out


def convert_hf_llama(hf: Dict[str, torch.Tensor],
                     num_layers: int) -> Dict[str, torch.Tensor]:
    """transformers LlamaForCausalLM -> models.llama.LlamaForCausalLM.

    Also covers Qwen2ForCausalLM (identical key names; its QKV biases are
    picked up when present -- build the native model with qkv_bias=True)."""
    out = {
        "embed.weight": hf["model.embed_tokens.weight"],
        "final_norm": hf["model.norm.weight"],
        "lm_head.weight": hf.get("lm_head.weight",
                                 hf["model.embed_tokens.weight"]),
    }
    for i in range(num_layers):
        p = "model.layers.{}.".format(i)
        o = "layers.{}.".format(i)
        out[o + "qkv.weight"] = torch.cat(
            [hf[p + "self_attn.q_proj.weight"],
             hf[p + "self_attn.k_proj.weight"],
             hf[p + "self_attn.v_proj.weight"]], dim=0)
        if p + "self_attn.q_proj.bias" in hf:  # Qwen2
            out[o + "qkv.bias"] = torch.cat(
                [hf[p + "self_attn.q_proj.bias"],
                 hf[p + "self_attn.k_proj.bias"],
                 hf[p + "self_attn.v_proj.bias"]], dim=0)
        out[o + "o_proj.weight"] = hf[p + "self_attn.o_proj.weight"]
        out[o + "attn_norm"] = hf[p + "input_layernorm.weight"]
        out[o + "mlp_norm"] = hf[p + "post_attention_layernorm.weight"]
        out[o + "gate_up.weight"] = torch.cat(
            [hf[p + "mlp.gate_proj.weight"], hf[p + "mlp.up_proj.weight"]],
            dim=0)
        out[o + "down.weight"] = hf[p + "mlp.down_proj.weight"]
    return out


# Qwen2 checkpoints use llama key names + QKV biases (handled above)
convert_hf_qwen2 = convert_hf_llama


def convert_hf_gpt2(hf: Dict[str, torch.Tensor],
                    num_layers: int) -> Dict[str, torch.Tensor]:
    """transformers GPT2LMHeadModel -> models.gpt2.GPT2ForCausalLM.

    HF GPT-2 stores projections as Conv1D ([in, out]); nn.Linear wants
    [out, in], so every projection weight transposes."""

    def g(key):
        return hf.get("transformer." + key, hf.get(key))

    out = {
        "embed.weight": g("wte.weight"),
        "wpe.weight": g("wpe.weight"),
        "lnf_w": g("ln_f.weight"),
        "lnf_b": g("ln_f.bias"),
        "lm_head.weight": hf.get("lm_head.weight", g("wte.weight")),
    }
    for i in range(num_layers):
        p = "h.{}.".format(i)
        o = "layers.{}.".format(i)
        out[o + "ln1_w"] = g(p + "ln_1.weight")
        out[o + "ln1_b"] = g(p + "ln_1.bias")
        out[o + "attn_qkv.weight"] = g(p + "attn.c_attn.weight").t().contiguous()
        out[o + "attn_qkv.bias"] = g(p + "attn.c_attn.bias")
        out[o + "attn_out.weight"] = g(p + "attn.c_proj.weight").t().contiguous()
        out[o + "attn_out.bias"] = g(p + "attn.c_proj.bias")
        out[o + "ln2_w"] = g(p + "ln_2.weight")
        out[o + "ln2_b"] = g(p + "ln_2.bias")
        out[o + "fc_in.weight"] = g(p + "mlp.c_fc.weight").t().contiguous()
        out[o + "fc_in_bias"] = g(p + "mlp.c_fc.bias")
        out[o + "fc_out.weight"] = g(p + "mlp.c_proj.weight").t().contiguous()
        out[o + "fc_out.bias"] = g(p + "mlp.c_proj.bias")
    return out


def convert_hf_auto(hf: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
    """Detect a raw HuggingFace state dict by its key layout and convert it
    to the matching native layout (llama/qwen2, gpt2, or bert). Native-
    layout dicts pass through unchanged -- callers can always route
    through this."""
    def n_layers(prefix, probe):
        i = 0
        while any(k.startswith("{}{}{}".format(prefix, i, probe))
                  for k in hf):
            i += 1
        return i

    keys = hf.keys()
    if any(k.startswith("model.embed_tokens.") for k in keys):
        return convert_hf_llama(hf, n_layers("model.layers.", "."))
    if any("wte.weight" in k for k in keys) and \
            any(".attn.c_attn." in k for k in keys):
        pref = ("transformer.h."
                if any(k.startswith("transformer.h.") for k in keys)
                else "h.")
        return convert_hf_gpt2(hf, n_layers(pref, "."))
    if any("embeddings.word_embeddings." in k for k in keys):
        pref = ("bert.encoder.layer."
                if any(k.startswith("bert.encoder.layer.") for k in keys)
                else "encoder.layer.")
        return convert_hf_bert(hf, n_layers(pref, "."))
    return hf  # already native
