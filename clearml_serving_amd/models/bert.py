"""BERT-base encoder for bf16 inference on MI355X.

Post-LN transformer encoder (Devlin et al. 2018) structured around the HIP
op library (the reference serves BERT through Triton with an external
tokenizer preprocess, examples/huggingface -- here the encoder itself is
in-process):

- QKV projection is ONE [H, 3H] GEMM (hipBLASLt), split per head
- attention runs in the fused CDNA4 flash-style kernel (ops.attention):
  QK^T -> online softmax -> PV in one kernel, padding handled via seq_lens
- residual add + LayerNorm is the fused HIP ``layernorm(residual=...)``
  (one HBM pass over the hidden states)
- MLP fc1 keeps its bias in the fused ``bias_gelu`` epilogue kernel

Forward takes a dict (dynamic-batcher friendly): input_ids [B, S] int,
optional attention_mask [B, S], optional token_type_ids -- output is
classification logits [B, num_labels] (sequence output available via
``return_hidden``). The graph contains no data-dependent python branches,
so every batch bucket is hipGraph-capturable.
"""

from typing import Dict, Optional

import torch
import torch.nn as nn

from .. import ops
from . import register_arch


class BertSelfAttention(nn.Module):
    def __init__(self, hidden: int, heads: int):
        super().__init__()
        self.heads = heads
        self.head_dim = hidden // heads
        self.qkv = nn.Linear(hidden, 3 * hidden)
        # bias rides hipBLASLt's fused epilogue (the round-1 split
        # out/out_bias forced a separate elementwise add per layer --
        # measured 8.1% of BERT GPU time across the two adds)
        self.out = nn.Linear(hidden, hidden)

    def forward(self, x: torch.Tensor, seq_lens: Optional[torch.Tensor]):
        b, s, h = x.shape
        qkv = self.qkv(x)  # [B, S, 3H]
        # [B, S, H, D] strided views: the attention kernel is
        # layout-agnostic, so no transpose copies
        hd = self.head_dim
        q, k, v = qkv.split([h, h, h], dim=-1)
        q = q.unflatten(-1, (self.heads, hd))
        k = k.unflatten(-1, (self.heads, hd))
        v = v.unflatten(-1, (self.heads, hd))
        ctx = ops.attention(q, k, v, causal=False, seq_lens=seq_lens,
                            layout="bshd")
        ctx = ctx.reshape(b, s, h)
        return self.out(ctx)


class BertLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, intermediate: int,
                 ln_eps: float = 1e-12):
        super().__init__()
        self.ln_eps = ln_eps
        self.attn = BertSelfAttention(hidden, heads)
        self.ln1_w = nn.Parameter(torch.ones(hidden))
        self.ln1_b = nn.Parameter(torch.zeros(hidden))
        self.fc1 = nn.Linear(hidden, intermediate, bias=False)
        self.fc1_bias = nn.Parameter(torch.zeros(intermediate))  # bias_gelu
        self.fc2 = nn.Linear(intermediate, hidden)  # bias in GEMM epilogue
        self.ln2_w = nn.Parameter(torch.ones(hidden))
        self.ln2_b = nn.Parameter(torch.zeros(hidden))

    def forward(self, x: torch.Tensor, seq_lens: Optional[torch.Tensor]):
        attn_out = self.attn(x, seq_lens)  # bias fused in the GEMM
        # fused: LN(attn_out + x)
        x = ops.layernorm(attn_out, self.ln1_w, self.ln1_b,
                          eps=self.ln_eps, residual=x)
        mlp = self.fc2(ops.bias_gelu(self.fc1(x), self.fc1_bias))
        return ops.layernorm(mlp, self.ln2_w, self.ln2_b,
                             eps=self.ln_eps, residual=x)


class BertEncoder(nn.Module):
    def __init__(
        self, vocab_size: int = 30522, hidden: int = 768, layers: int = 12,
        heads: int = 12, intermediate: int = 3072, max_pos: int = 512,
        type_vocab: int = 2, num_labels: int = 2, ln_eps: float = 1e-12,
    ):
        super().__init__()
        self.ln_eps = ln_eps  # BERT's canonical LayerNorm eps
        self.word_emb = nn.Embedding(vocab_size, hidden)
        self.pos_emb = nn.Embedding(max_pos, hidden)
        self.type_emb = nn.Embedding(type_vocab, hidden)
        self.emb_ln_w = nn.Parameter(torch.ones(hidden))
        self.emb_ln_b = nn.Parameter(torch.zeros(hidden))
        self.layers = nn.ModuleList(
            [BertLayer(hidden, heads, intermediate, ln_eps)
             for _ in range(layers)]
        )
        self.pooler = nn.Linear(hidden, hidden)
        self.classifier = nn.Linear(hidden, num_labels)
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
        self, inputs: Dict[str, torch.Tensor], return_hidden: bool = False
    ) -> torch.Tensor:
        ids = inputs["input_ids"].long()
        b, s = ids.shape
        mask = inputs.get("attention_mask")
        seq_lens = mask.to(torch.int32).sum(dim=-1) if mask is not None else None
        positions = torch.arange(s, device=ids.device)[None, :].expand(b, s)
        # token_type defaults to segment 0 (BERT always adds the segment
        # embedding; omitting it diverges from the canonical model)
        types = inputs.get("token_type_ids")
        emb = self.word_emb(ids) + self.pos_emb(positions)
        if types is not None:
            emb = emb + self.type_emb(types.long())
        else:
            emb = emb + self.type_emb.weight[0]
        x = ops.layernorm(emb, self.emb_ln_w, self.emb_ln_b, eps=self.ln_eps)

        for layer in self.layers:
            x = layer(x, seq_lens)
        if return_hidden:
            return x
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return self.classifier(pooled)


@register_arch("bert-base")
def bert_base(num_labels: int = 2, vocab_size: int = 30522,
              max_pos: int = 512) -> BertEncoder:
    return BertEncoder(vocab_size=vocab_size, hidden=768, layers=12, heads=12,
                       intermediate=3072, max_pos=max_pos,
                       num_labels=num_labels)


@register_arch("bert-large")
def bert_large(num_labels: int = 2, vocab_size: int = 30522,
               max_pos: int = 512) -> BertEncoder:
    return BertEncoder(vocab_size=vocab_size, hidden=1024, layers=24, heads=16,
                       intermediate=4096, max_pos=max_pos,
                       num_labels=num_labels)
