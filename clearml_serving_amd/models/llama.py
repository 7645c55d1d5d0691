"""Llama-family decoder for the native LLM engine.

Llama-3 architecture (RMSNorm, SwiGLU MLP, RoPE, GQA) built directly on the
HIP op library with a paged KV cache (the reference delegates all of this to
vLLM, preprocess_service.py:619-1095):

- prefill: padded [B, S] token batch; fused causal flash attention with
  per-sequence lengths; computed K/V scattered into cache pages
  (ops.kv_cache_write) so decode continues where prefill left off
- decode:  [B] one-token batch; paged decode attention (ops.attention_decode)
- RMSNorm keeps the residual stream in place (one HBM pass per norm);
  SwiGLU uses the fused silu_mul kernel; QKV and gate+up are single merged
  GEMMs (hipBLASLt)

Presets: llama-3-8b (the BASELINE.json TP=8 target), plus scaled-down
variants for tests.
"""

import math
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from .. import ops
from . import register_arch


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden: int = 4096
    layers: int = 32
    heads: int = 32
    kv_heads: int = 8
    intermediate: int = 14336
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_position: int = 8192
    tie_embeddings: bool = False
    qkv_bias: bool = False  # Qwen2-family checkpoints carry QKV biases

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads


class LlamaLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_rank: int = 0, tp_size: int = 1):
        super().__init__()
        self.cfg = cfg
        assert cfg.heads % tp_size == 0 and cfg.kv_heads % tp_size == 0, \
            "TP degree must divide head counts"
        self.heads = cfg.heads // tp_size
        self.kv_heads = cfg.kv_heads // tp_size
        self.head_dim = cfg.head_dim
        self.tp_size = tp_size
        h, hd = cfg.hidden, cfg.head_dim
        q_out = self.heads * hd
        kv_out = self.kv_heads * hd
        inter = cfg.intermediate // tp_size

        self.attn_norm = nn.Parameter(torch.ones(h))
        self.qkv = nn.Linear(h, q_out + 2 * kv_out, bias=cfg.qkv_bias)
        self.o_proj = nn.Linear(q_out, h, bias=False)
        self.mlp_norm = nn.Parameter(torch.ones(h))
        self.gate_up = nn.Linear(h, 2 * inter, bias=False)
        self.down = nn.Linear(inter, h, bias=False)
        self.inter = inter

    @staticmethod
    def _proj(mod, x):
        # decode-shaped rows route to the in-tree skinny MFMA kernel;
        # prefill rows and non-plain modules (e.g. Fp8Linear) fall through
        if type(mod) is nn.Linear:
            y = ops.skinny_linear(x, mod.weight)
            if mod.bias is not None:  # Qwen2 QKV bias (skinny kernel is
                y = y + mod.bias      # bias-free; one fused add per call)
            return y
        return mod(x)

    def forward(self, x, residual, positions, kv_cache, attn_ctx):
        attn_out = self.attn_half(x, residual, positions, kv_cache, attn_ctx)
        if self.tp_size > 1:
            from ..parallel import tp as tp_mod

            tp_mod.maybe_all_reduce(attn_out)  # row-parallel o_proj
        mlp_out = self.mlp_half(attn_out, residual)
        if self.tp_size > 1:
            from ..parallel import tp as tp_mod

            tp_mod.maybe_all_reduce(mlp_out)  # row-parallel down projection
        return mlp_out  # residual carries the stream

    def attn_half(self, x, residual, positions, kv_cache, attn_ctx):
        """norm -> qkv -> rope -> KV write -> attention -> o_proj, WITHOUT
        the row-parallel all-reduce -- the TP decode microbatch pipeline
        (LlamaForCausalLM.forward_pipelined) issues that reduce async so it
        overlaps the other microbatch's compute."""
        cfg = self.cfg
        # fp8 serving path (models/quant.py): activation quantization is
        # FUSED into the producing kernels -- rmsnorm/silu_mul emit fp8
        # bytes + per-row scales in the same HBM pass
        fp8 = not isinstance(self.qkv, nn.Linear)
        if fp8:
            x8, xs = ops.rmsnorm_fp8(x, self.attn_norm, cfg.rms_eps,
                                     residual=residual)
            t = x8.shape[0]
            qkv = self.qkv.forward_q(x8, xs)
        else:
            # fused: residual += x_prev; x = rmsnorm(residual)
            x = ops.rmsnorm(x, self.attn_norm, cfg.rms_eps, residual=residual)
            t = x.shape[0]
            qkv = self._proj(self.qkv, x)
        # token-strided views into the merged projection -- the rope /
        # kv_cache_write / decode-attention kernels are stride-aware, so no
        # .contiguous() copies on the decode hot path
        q, k, v = qkv.split(
            [self.heads * self.head_dim, self.kv_heads * self.head_dim,
             self.kv_heads * self.head_dim], dim=-1)
        q = q.unflatten(-1, (self.heads, self.head_dim))
        k = k.unflatten(-1, (self.kv_heads, self.head_dim))
        v = v.unflatten(-1, (self.kv_heads, self.head_dim))
        q, k = ops.rope_inplace(q, k, positions, theta=cfg.rope_theta)

        if kv_cache is not None:
            # 2-tuple (k, v) bf16 cache, or 4-tuple (k8, v8, k_scale,
            # v_scale) fp8 cache (engine cfg kv_dtype="fp8"): e4m3 bytes +
            # per-token-per-head scales, halving KV reads on long decodes
            k_cache, v_cache = kv_cache[0], kv_cache[1]
            k_sc = kv_cache[2] if len(kv_cache) > 2 else None
            v_sc = kv_cache[3] if len(kv_cache) > 2 else None
            ops.kv_cache_write(k, v, k_cache, v_cache,
                               attn_ctx["slot_mapping"], k_sc, v_sc)

        scale = 1.0 / math.sqrt(self.head_dim)
        if attn_ctx["mode"] == "prefill":
            b, s = attn_ctx["batch"], attn_ctx["seq"]
            qb = q.unflatten(0, (b, s))  # [B, S, H, D] strided views
            kb = k.unflatten(0, (b, s))
            vb = v.unflatten(0, (b, s))
            ctx = ops.attention(qb, kb, vb, causal=True, scale=scale,
                                seq_lens=attn_ctx["seq_lens"], layout="bshd")
            ctx = ctx.reshape(t, self.heads * self.head_dim)
        elif attn_ctx["mode"] == "prefill_paged":
            # chunked prefill: this chunk's queries attend to the full paged
            # history (K/V of the chunk were just scattered into the cache)
            b, s = attn_ctx["batch"], attn_ctx["seq"]
            ctx = ops.attention_prefill_paged(
                q.unflatten(0, (b, s)), k_cache, v_cache,
                attn_ctx["block_table"], attn_ctx["kv_lens"],
                attn_ctx["q_lens"], scale=scale, k_scale=k_sc, v_scale=v_sc)
            ctx = ctx.reshape(t, self.heads * self.head_dim)
        else:  # decode: one token per sequence
            ctx = ops.attention_decode(
                q, k_cache, v_cache, attn_ctx["block_table"],
                attn_ctx["seq_lens"], scale=scale, k_scale=k_sc, v_scale=v_sc)
            ctx = ctx.view(t, self.heads * self.head_dim)
        if fp8:
            # attention context has no fused producer: one-pass dynamic quant
            c8, cs = ops.quant_fp8(ctx)
            attn_out = self.o_proj.forward_q(c8, cs)
        else:
            attn_out = self._proj(self.o_proj, ctx)
        return attn_out

    def mlp_half(self, attn_out, residual):
        """norm -> gate_up -> silu*up -> down, WITHOUT the row-parallel
        all-reduce (see attn_half)."""
        cfg = self.cfg
        fp8 = not isinstance(self.qkv, nn.Linear)
        if fp8:
            x8, xs = ops.rmsnorm_fp8(attn_out, self.mlp_norm, cfg.rms_eps,
                                     residual=residual)
            gate_up = self.gate_up.forward_q(x8, xs)
            gate, up = gate_up.split([self.inter, self.inter], dim=-1)
            g8, gs = ops.silu_mul_fp8(gate, up)
            mlp_out = self.down.forward_q(g8, gs)
        else:
            x = ops.rmsnorm(attn_out, self.mlp_norm, cfg.rms_eps,
                            residual=residual)
            gate_up = self.gate_up(x)
            gate, up = gate_up.split([self.inter, self.inter], dim=-1)
            mlp_out = self._proj(self.down, ops.silu_mul(gate, up))
        return mlp_out


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: LlamaConfig, tp_rank: int = 0, tp_size: int = 1):
        super().__init__()
        self.cfg = cfg
        self.tp_rank, self.tp_size = tp_rank, tp_size
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.layers = nn.ModuleList(
            [LlamaLayer(cfg, tp_rank, tp_size) for _ in range(cfg.layers)])
        self.final_norm = nn.Parameter(torch.ones(cfg.hidden))
        # TP: lm_head column-parallel over vocab (shards all-gathered)
        assert cfg.vocab_size % tp_size == 0
        self.lm_head = nn.Linear(cfg.hidden, cfg.vocab_size // tp_size,
                                 bias=False)
        if cfg.tie_embeddings and tp_size == 1:
            self.lm_head.weight = self.embed.weight
        self._init_weights()

    def _init_weights(self):
        std = 0.02
        for m in self.modules():
            if isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, std=std)
            elif isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=std)

    def forward(
        self,
        tokens: torch.Tensor,      # [T] flat token ids
        positions: torch.Tensor,   # [T] int32
        kv_caches: Optional[List] = None,  # per layer (k_cache, v_cache)
        attn_ctx: Optional[Dict] = None,
        last_token_idx: Optional[torch.Tensor] = None,
        return_hidden: bool = False,
        gather_logits: bool = True,
    ) -> torch.Tensor:
        """Returns logits [n_seqs, vocab] at the selected token positions
        (or the final-norm hidden states [T, hidden] when ``return_hidden``
        -- the embeddings serve path)."""
        x = self.embed(tokens.long())
        residual = torch.zeros_like(x)
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            x = layer(x, residual, positions, cache, attn_ctx)
        x = ops.rmsnorm(x, self.final_norm, self.cfg.rms_eps,
                        residual=residual)
        if last_token_idx is not None:
            x = x[last_token_idx.long()]
        if return_hidden:
            return x
        logits = self.lm_head(x)
        if self.tp_size > 1 and gather_logits:
            return self._gathered(logits)
        return logits

    def _gathered(self, logits):
        # lm_head is column-sharded. The serving engine passes
        # gather_logits=False and samples ON the shards (packed-argmax /
        # Gumbel-max all-reduce of O(batch) words, parallel/tp.py) --
        # the full-vocab all-gather (b x 128k x 2B per decode step) is
        # only for offline full-logits consumers like the TP math tests.
        from ..parallel import tp as tp_mod

        return tp_mod.gather_logits(logits)

    def forward_pipelined(self, parts, kv_caches):
        """TP decode comm/compute overlap: run N microbatches through the
        layer stack in a 2-stage pipeline -- each half's row-parallel
        all-reduce is issued async and waited only when ITS next consumer
        runs, so the collective overlaps the other half's GEMMs (xGMI comm
        under compute; ROADMAP 'decode microbatch pipelining').

        parts: list of (tokens, positions, attn_ctx); returns concatenated
        logits rows in part order (gather_logits=False semantics). Issue
        order per layer is identical on every rank (attn_0, attn_1, mlp_0,
        mlp_1), so the collectives line up."""
        from ..parallel import tp as tp_mod

        n = len(parts)
        xs, residuals = [], []
        for tokens, positions, _ctx in parts:
            x = self.embed(tokens.long())
            residuals.append(torch.zeros_like(x))
            xs.append(x)
        works = [None] * n
        for li, layer in enumerate(self.layers):
            cache = kv_caches[li] if kv_caches is not None else None
            attn = [None] * n
            for j in range(n):
                if works[j] is not None:
                    works[j].wait()  # previous mlp reduce of THIS half
                attn[j] = layer.attn_half(xs[j], residuals[j], parts[j][1],
                                          cache, parts[j][2])
                works[j] = tp_mod.all_reduce_async(attn[j])
            for j in range(n):
                if works[j] is not None:
                    works[j].wait()  # attn reduce of this half
                xs[j] = layer.mlp_half(attn[j], residuals[j])
                works[j] = tp_mod.all_reduce_async(xs[j])
        outs = []
        for j in range(n):
            if works[j] is not None:
                works[j].wait()
            h = ops.rmsnorm(xs[j], self.final_norm, self.cfg.rms_eps,
                            residual=residuals[j])
            outs.append(self.lm_head(h))
        return torch.cat(outs, dim=0)


PRESETS = {
    "llama-3-8b": LlamaConfig(),
    "llama-3-1b": LlamaConfig(hidden=2048, layers=16, heads=32, kv_heads=8,
                              intermediate=8192, tie_embeddings=True),
    "llama-tiny": LlamaConfig(vocab_size=512, hidden=256, layers=2, heads=4,
                              kv_heads=2, intermediate=512, rope_theta=10000.0,
                              max_position=512),
    # Qwen2 family: llama architecture + QKV biases + ChatML template
    # (models/convert.py convert_hf_qwen2; numerics pinned vs transformers
    # in tests/test_hf_convert.py)
    "qwen2-7b": LlamaConfig(vocab_size=152064, hidden=3584, layers=28,
                            heads=28, kv_heads=4, intermediate=18944,
                            rope_theta=1e6, rms_eps=1e-6, max_position=32768,
                            qkv_bias=True),
    "qwen2-1.5b": LlamaConfig(vocab_size=151936, hidden=1536, layers=28,
                              heads=12, kv_heads=2, intermediate=8960,
                              rope_theta=1e6, rms_eps=1e-6,
                              max_position=32768, qkv_bias=True,
                              tie_embeddings=True),
}


@register_arch("llama")
def llama(preset: str = "llama-3-8b", **overrides) -> LlamaForCausalLM:
    cfg_base = PRESETS[preset]
    cfg = LlamaConfig(**{**cfg_base.__dict__, **overrides})
    return LlamaForCausalLM(cfg)
