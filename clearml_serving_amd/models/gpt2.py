"""GPT-2 family decoder for the native LLM engine.

The classic HF decoder architecture (Radford et al. 2019): pre-LayerNorm
blocks, learned absolute position embeddings (no RoPE), merged QKV with
bias, erf-GELU MLP, tied lm_head. Implements the SAME forward interface as
models.llama.LlamaForCausalLM (flat tokens/positions, paged kv_caches,
attn_ctx modes prefill / prefill_paged / decode, last_token_idx), so the
continuous-batching engine, chunked prefill, decode hipGraphs and ngram
speculation all serve it unchanged (reference parity: vLLM's GPT-2 support,
preprocess_service.py:619-1095 delegation).

Numerics pinned against transformers.GPT2LMHeadModel in
tests/test_hf_convert.py. TP sharding is llama-only for now (the engine
refuses arch=gpt2 at tp_size>1).
"""

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from .. import ops
from . import register_arch


@dataclass
class GPT2Config:
    vocab_size: int = 50257
    hidden: int = 768
    layers: int = 12
    heads: int = 12
    intermediate: int = 3072
    max_position: int = 1024
    ln_eps: float = 1e-5

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads

    @property
    def kv_heads(self) -> int:  # MHA: engine KV sizing reads this
        return self.heads


class GPT2Layer(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        h = cfg.hidden
        self.cfg = cfg
        self.heads = cfg.heads
        self.head_dim = cfg.head_dim
        self.ln1_w = nn.Parameter(torch.ones(h))
        self.ln1_b = nn.Parameter(torch.zeros(h))
        self.attn_qkv = nn.Linear(h, 3 * h)      # bias per GPT-2
        self.attn_out = nn.Linear(h, h)
        self.ln2_w = nn.Parameter(torch.ones(h))
        self.ln2_b = nn.Parameter(torch.zeros(h))
        self.fc_in = nn.Linear(h, cfg.intermediate, bias=False)
        self.fc_in_bias = nn.Parameter(torch.zeros(cfg.intermediate))
        self.fc_out = nn.Linear(cfg.intermediate, h)

    def forward(self, x, kv_cache, attn_ctx):
        cfg = self.cfg
        t, h = x.shape
        # pre-LN attention block: x = x + attn(LN(x))
        xn = ops.layernorm(x, self.ln1_w, self.ln1_b, eps=cfg.ln_eps)
        qkv = self.attn_qkv(xn)
        q, k, v = qkv.split([h, h, h], dim=-1)
        q = q.unflatten(-1, (self.heads, self.head_dim))
        k = k.unflatten(-1, (self.heads, self.head_dim))
        v = v.unflatten(-1, (self.heads, self.head_dim))

        if kv_cache is not None:
            k_cache, v_cache = kv_cache[0], kv_cache[1]
            k_sc = kv_cache[2] if len(kv_cache) > 2 else None
            v_sc = kv_cache[3] if len(kv_cache) > 2 else None
            ops.kv_cache_write(k, v, k_cache, v_cache,
                               attn_ctx["slot_mapping"], k_sc, v_sc)

        scale = 1.0 / math.sqrt(self.head_dim)
        if attn_ctx["mode"] == "prefill":
            b, s = attn_ctx["batch"], attn_ctx["seq"]
            ctx = ops.attention(q.unflatten(0, (b, s)),
                                k.unflatten(0, (b, s)),
                                v.unflatten(0, (b, s)),
                                causal=True, scale=scale,
                                seq_lens=attn_ctx["seq_lens"],
                                layout="bshd")
            ctx = ctx.reshape(t, h)
        elif attn_ctx["mode"] == "prefill_paged":
            b, s = attn_ctx["batch"], attn_ctx["seq"]
            ctx = ops.attention_prefill_paged(
                q.unflatten(0, (b, s)), k_cache, v_cache,
                attn_ctx["block_table"], attn_ctx["kv_lens"],
                attn_ctx["q_lens"], scale=scale, k_scale=k_sc, v_scale=v_sc)
            ctx = ctx.reshape(t, h)
        else:  # decode
            ctx = ops.attention_decode(
                q, k_cache, v_cache, attn_ctx["block_table"],
                attn_ctx["seq_lens"], scale=scale, k_scale=k_sc,
                v_scale=v_sc)
            ctx = ctx.view(t, h)
        x = x + self.attn_out(ctx)
        # pre-LN MLP block: x = x + mlp(LN(x))
        xn = ops.layernorm(x, self.ln2_w, self.ln2_b, eps=cfg.ln_eps)
        mlp = self.fc_out(ops.bias_gelu(self.fc_in(xn), self.fc_in_bias))
        return x + mlp


class GPT2ForCausalLM(nn.Module):
    def __init__(self, cfg: GPT2Config):
        super().__init__()
        self.cfg = cfg
        self.tp_size = 1  # TP sharding not implemented for GPT-2
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.wpe = nn.Embedding(cfg.max_position, cfg.hidden)
        self.layers = nn.ModuleList(
            [GPT2Layer(cfg) for _ in range(cfg.layers)])
        self.lnf_w = nn.Parameter(torch.ones(cfg.hidden))
        self.lnf_b = nn.Parameter(torch.zeros(cfg.hidden))
        self.lm_head = nn.Linear(cfg.hidden, cfg.vocab_size, bias=False)
        self.lm_head.weight = self.embed.weight  # GPT-2 ties embeddings
        self._init_weights()

    def _init_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=0.02)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.02)

    def forward(
        self,
        tokens: torch.Tensor,      # [T] flat token ids
        positions: torch.Tensor,   # [T] int32
        kv_caches: Optional[List] = None,
        attn_ctx: Optional[Dict] = None,
        last_token_idx: Optional[torch.Tensor] = None,
        return_hidden: bool = False,
        gather_logits: bool = True,
    ) -> torch.Tensor:
        # position ids clamp at max_position-1 (padded prefill rows carry
        # position 0; real rows are bounded by the engine's max_model_len)
        pos = positions.long().clamp_(max=self.cfg.max_position - 1)
        x = self.embed(tokens.long()) + self.wpe(pos)
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            x = layer(x, cache, attn_ctx)
        x = ops.layernorm(x, self.lnf_w, self.lnf_b, eps=self.cfg.ln_eps)
        if last_token_idx is not None:
            x = x[last_token_idx.long()]
        if return_hidden:
            return x
        return self.lm_head(x)


PRESETS = {
    "gpt2": GPT2Config(),
    "gpt2-medium": GPT2Config(hidden=1024, layers=24, heads=16,
                              intermediate=4096),
    "gpt2-large": GPT2Config(hidden=1280, layers=36, heads=20,
                             intermediate=5120),
    # head_dim 64: the GPU attention kernels support D in {64, 128}
    # (every real GPT-2 size is 64)
    "gpt2-tiny": GPT2Config(vocab_size=512, hidden=128, layers=2, heads=2,
                            intermediate=256, max_position=512),
}


@register_arch("gpt2")
def gpt2(preset: str = "gpt2", **overrides) -> GPT2ForCausalLM:
    cfg = GPT2Config(**{**PRESETS[preset].__dict__, **overrides})
    return GPT2ForCausalLM(cfg)
