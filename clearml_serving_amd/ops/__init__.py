"""HIP/CDNA4 kernel library: python dispatch layer.

The compute tier the reference outsources to Triton/vLLM
lives here as hand-written gfx950 kernels (``csrc/*.hip``), compiled in-tree
to ``_hip_ops.so`` by ``__graft_entry__.build()`` (hipcc --offload-arch=gfx950
via torch.utils.cpp_extension).

Dispatch policy:
- tensor on GPU  -> the HIP extension is REQUIRED; a missing .so raises
  instead of silently falling back to eager PyTorch (so a GPU run can never
  "pass" on the fallback path).
- tensor on CPU  -> plain PyTorch reference implementations (used by the CPU
  test suite and as the fp32 numerics reference for the GPU kernels).

Set CLEARML_SERVING_AMD_FORCE_EAGER=1 to bypass the extension on GPU
(debug/ablation only).
"""

import importlib.machinery
import importlib.util
import math
import os
from typing import Optional, Tuple

import torch

_ext = None
_ext_error: Optional[str] = None
_FORCE_EAGER = os.environ.get("CLEARML_SERVING_AMD_FORCE_EAGER") == "1"


def _try_load_ext() -> None:
    global _ext, _ext_error
    if _ext is not None or _ext_error is not None:
        return
    so_path = os.path.join(os.path.dirname(__file__), "_hip_ops.so")
    if not os.path.exists(so_path):
        _ext_error = "extension not built ({} missing)".format(so_path)
        return
    try:
        loader = importlib.machinery.ExtensionFileLoader("_hip_ops", so_path)
        spec = importlib.util.spec_from_loader("_hip_ops", loader)
        mod = importlib.util.module_from_spec(spec)
        loader.exec_module(mod)
        _ext = mod
    except Exception as ex:  # loud, but deferred to first GPU use
        _ext_error = "failed loading {}: {}".format(so_path, ex)


_try_load_ext()


def has_extension() -> bool:
    return _ext is not None


def _require_ext(opname: str):
    if _FORCE_EAGER:
        return None
    if _ext is None:
        raise RuntimeError(
            "clearml_serving_amd HIP extension required for GPU op '{}' but "
            "not loaded: {}. Build it with __graft_entry__.build() "
            "(hipcc --offload-arch=gfx950).".format(opname, _ext_error)
        )
    return _ext


def extension_or_none():
    return _ext


# --------------------------------------------------------------------- #
# LayerNorm (+ fused residual add)
# --------------------------------------------------------------------- #
def layernorm(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
    eps: float = 1e-5, residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """out = LN(x + residual) over the last dim. bf16/fp32 in, same dtype out."""
    if x.is_cuda:
        ext = _require_ext("layernorm")
        if ext is not None:
            return ext.layernorm(x, weight, bias, eps, residual)
    if residual is not None:
        x = x + residual
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), weight.float(), bias.float(), eps
    ).to(x.dtype)


def rmsnorm(
    x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6,
    residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """out = x / rms(x) * weight (llama-style), optional fused residual add.

    When ``residual`` is given the op ALSO writes x+residual back into
    ``residual`` in-place on the GPU path (the running residual stream),
    matching the CPU reference below.
    """
    if x.is_cuda:
        ext = _require_ext("rmsnorm")
        if ext is not None:
            return ext.rmsnorm(x, weight, eps, residual)
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
        residual.copy_(xf.to(residual.dtype))
    out = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (out * weight.float()).to(x.dtype)


# --------------------------------------------------------------------- #
# Activations / epilogues
# --------------------------------------------------------------------- #
def bias_gelu(x: torch.Tensor, bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """out = gelu(x + bias), erf formulation (BERT)."""
    if x.is_cuda:
        ext = _require_ext("bias_gelu")
        if ext is not None:
            return ext.bias_gelu(x, bias)
    if bias is not None:
        x = x + bias
    return torch.nn.functional.gelu(x.float()).to(x.dtype)


def bias_relu_add(
    x: torch.Tensor, bias: Optional[torch.Tensor] = None,
    residual: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """out = relu(x + bias + residual) -- ResNet bottleneck epilogue."""
    if x.is_cuda:
        ext = _require_ext("bias_relu_add")
        if ext is not None:
            return ext.bias_relu_add(x, bias, residual)
    if bias is not None:
        x = x + bias
    if residual is not None:
        x = x + residual
    return torch.relu(x)


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """out = silu(gate) * up -- llama MLP."""
    if gate.is_cuda:
        ext = _require_ext("silu_mul")
        if ext is not None:
            return ext.silu_mul(gate, up)
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(gate.dtype)


# --------------------------------------------------------------------- #
# fp8 (OCP e4m3) decode path: fused activation quant + skinny fp8 GEMM
# (kernels: csrc/fp8_path.hip; replaces the round-1 un-fused quantization
# that made fp8 slower than bf16 end to end)
# --------------------------------------------------------------------- #
F8_MAX = 448.0


def _quant_fp8_ref(y: torch.Tensor):
    """Reference per-row quantization (CPU path / numerics oracle)."""
    y2 = y.float().reshape(-1, y.shape[-1])
    amax = y2.abs().amax(dim=1).clamp(min=1e-8)
    scale = (amax / F8_MAX).to(torch.float32)
    q = (y2 / scale[:, None]).clamp(-F8_MAX, F8_MAX).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale


def rmsnorm_fp8(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6,
                residual: Optional[torch.Tensor] = None):
    """rmsnorm (+fused residual update, same contract as rmsnorm()) emitting
    fp8-e4m3 bytes + per-row scales in ONE pass over HBM."""
    if x.is_cuda:
        ext = _require_ext("rmsnorm_fp8")
        if ext is not None:
            out8, scales = ext.rmsnorm_fp8(x, weight, eps, residual)
            return out8, scales
    return _quant_fp8_ref(rmsnorm(x, weight, eps, residual=residual))


def silu_mul_fp8(gate: torch.Tensor, up: torch.Tensor):
    """silu(gate) * up -> fp8-e4m3 bytes + per-row scales, one fused pass."""
    if gate.is_cuda:
        ext = _require_ext("silu_mul_fp8")
        if ext is not None:
            out8, scales = ext.silu_mul_fp8(gate, up)
            return out8, scales
    return _quant_fp8_ref(silu_mul(gate, up))


def quant_fp8(x: torch.Tensor):
    """Per-row dynamic fp8 quantization (for activations whose producer
    isn't one of the fused kernels, e.g. attention context)."""
    if x.is_cuda:
        ext = _require_ext("quant_fp8")
        if ext is not None:
            out8, scales = ext.quant_fp8(x)
            return out8, scales
    return _quant_fp8_ref(x)


def swizzle_fp8_weight(w8: torch.Tensor) -> torch.Tensor:
    """Reorder an fp8 [N, K] weight for skinny_gemm_fp8: within each
    64-byte k window-pair, bytes become [lane-group][window][8] so one
    16-byte lane load yields both MFMA operands of the pair (un-swizzled
    weights need two 8-byte loads -- measured request-rate bound)."""
    n, k = w8.shape
    assert k % 64 == 0
    v = w8.view(torch.uint8).reshape(n, k // 64, 2, 4, 8)
    return v.permute(0, 1, 3, 2, 4).contiguous().reshape(n, k)


def unswizzle_fp8_weight(w8s: torch.Tensor) -> torch.Tensor:
    n, k = w8s.shape
    v = w8s.view(torch.uint8).reshape(n, k // 64, 4, 2, 8)
    return v.permute(0, 1, 3, 2, 4).contiguous().reshape(n, k)


def skinny_gemm_fp8(a8: torch.Tensor, a_scale: torch.Tensor,
                    w8: torch.Tensor, w_scale: torch.Tensor,
                    swizzled: bool = False) -> torch.Tensor:
    """C[M,N] = (Aq*s_a) @ (Wq*s_w)^T -> bf16, decode-shaped M <= 64.

    One workgroup per 16 W columns (fills the 256-CU chip at llama
    projection N), fp8 operands halve the weight-streaming bytes vs bf16.
    ``swizzled``: w8 is already in swizzle_fp8_weight layout (Fp8Linear
    prepares it once at quantization time).
    """
    if a8.is_cuda:
        ext = _require_ext("skinny_gemm_fp8")
        if ext is not None:
            ws = w8 if swizzled else swizzle_fp8_weight(w8)
            return ext.skinny_gemm_fp8(
                a8.view(torch.uint8), a_scale, ws.view(torch.uint8), w_scale)
    if swizzled:
        w8 = unswizzle_fp8_weight(w8)
    a = a8.view(torch.float8_e4m3fn).float() * a_scale[:, None]
    w = w8.view(torch.float8_e4m3fn).float() * w_scale[:, None]
    return (a @ w.t()).to(torch.bfloat16)


def skinny_gemm_fp8_v2(a8: torch.Tensor, a_scale: torch.Tensor,
                       w8: torch.Tensor, w_scale: torch.Tensor
                       ) -> torch.Tensor:
    """LDS-staged skinny fp8 GEMM for M 17..64 (plain fp8 weight layout):
    4 waves share one k progression over register-pipelined 128-byte
    windows -- covers the mid-batch decode regime where v1's direct
    fragment loads collapse and hipBLASLt underfills the chip."""
    if a8.is_cuda:
        ext = _require_ext("skinny_gemm_fp8_v2")
        if ext is not None:
            return ext.skinny_gemm_fp8_v2(
                a8.view(torch.uint8), a_scale, w8.view(torch.uint8), w_scale)
    a = a8.view(torch.float8_e4m3fn).float() * a_scale[:, None]
    w = w8.view(torch.float8_e4m3fn).float() * w_scale[:, None]
    return (a @ w.t()).to(torch.bfloat16)


def conv3x3_supported(x: torch.Tensor, w: torch.Tensor) -> bool:
    """True when the in-tree conv kernel serves this shape (3x3 s1 p1,
    ResNet-50 bottleneck widths, bf16 GPU)."""
    return (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 4
            and x.shape[2] == x.shape[3] and x.shape[2] in (56, 28, 14)
            and x.shape[1] % 32 == 0 and w.shape[0] % 64 == 0
            and w.shape[2] == 3 and w.shape[3] == 3
            and os.environ.get("CMLS_CONV3", "1") != "0"
            and _ext is not None)


def conv3x3_nhwc(x: torch.Tensor, w: torch.Tensor,
                 bias: Optional[torch.Tensor] = None, relu: bool = False,
                 residual: Optional[torch.Tensor] = None) -> torch.Tensor:
    """3x3 stride-1 pad-1 NHWC bf16 conv (implicit GEMM, MFMA) for the
    ResNet-50 bottleneck widths 56/28/14; optional fused bias+residual+relu
    epilogue. Kernel: csrc/conv3x3.hip. CPU/odd shapes: torch reference."""
    if (x.is_cuda and x.dtype == torch.bfloat16
            and x.shape[2] == x.shape[3] and x.shape[2] in (56, 28, 14)
            and x.shape[1] % 32 == 0 and w.shape[0] % 64 == 0):
        ext = _require_ext("conv3x3_nhwc")
        if ext is not None:
            xc = x.contiguous(memory_format=torch.channels_last)
            wc = w.contiguous(memory_format=torch.channels_last)
            return ext.conv3x3_nhwc(xc, wc, bias, relu, residual)
    y = torch.nn.functional.conv2d(x.float(), w.float(),
                                   bias.float() if bias is not None else None,
                                   stride=1, padding=1)
    if residual is not None:
        y = y + residual.float()
    if relu:
        y = torch.relu(y)
    return y.to(x.dtype)


def softmax(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    """Numerically-stable softmax over the last dim."""
    if x.is_cuda and dim in (-1, x.dim() - 1):
        ext = _require_ext("softmax")
        if ext is not None:
            return ext.softmax_lastdim(x)
    return torch.softmax(x.float(), dim=dim).to(x.dtype)


# --------------------------------------------------------------------- #
# Attention
# --------------------------------------------------------------------- #
def attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    causal: bool = False, scale: Optional[float] = None,
    seq_lens: Optional[torch.Tensor] = None,
    layout: str = "bhsd",
) -> torch.Tensor:
    """Fused scaled-dot-product attention (prefill / encoder).

    layout "bhsd": q,k,v = [B, H, S, D]; layout "bshd": [B, S, H, D]
    (strided views into a merged QKV projection are accepted -- no
    transpose copies). GQA via H_kv dividing H; ``seq_lens`` (int32 [B])
    masks keys >= len. Output has the input layout.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    bshd = layout == "bshd"
    if q.is_cuda:
        ext = _require_ext("attention")
        if ext is not None:
            # v2 (swapped-QK^T in-register softmax) measures 1.10-1.18x v1
            # across BERT/llama shapes (profiles/attn_v2_ab.txt);
            # CMLS_ATTN_V2=0 falls back to v1
            if os.environ.get("CMLS_ATTN_V2", "1") != "0":
                return ext.attention_prefill_v2(q, k, v, bool(causal),
                                                float(scale), seq_lens, bshd)
            return ext.attention_prefill(q, k, v, bool(causal), float(scale),
                                         seq_lens, bshd)
    if bshd:
        out = attention(q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3),
                        v.permute(0, 2, 1, 3), causal=causal, scale=scale,
                        seq_lens=seq_lens)
        return out.permute(0, 2, 1, 3).contiguous()
    # reference path (fp32 math)
    qf, kf, vf = q.float(), k.float(), v.float()
    hq, hkv = q.shape[1], k.shape[1]
    if hkv != hq:
        rep = hq // hkv
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    sq, sk = q.shape[2], k.shape[2]
    if causal:
        mask = torch.ones(sq, sk, dtype=torch.bool, device=q.device).tril(
            diagonal=sk - sq)
        scores = scores.masked_fill(~mask, float("-inf"))
    if seq_lens is not None:
        key_idx = torch.arange(sk, device=q.device)
        pad = key_idx[None, None, None, :] >= seq_lens[:, None, None, None]
        scores = scores.masked_fill(pad, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    return torch.matmul(probs, vf).to(q.dtype)


def _dequant_kv_cpu(cache: torch.Tensor, scale: torch.Tensor) -> torch.Tensor:
    """uint8 e4m3 paged cache [NB, Hkv, BS, D] + scales [NB, Hkv, BS] ->
    float cache (CPU reference path for the fp8 KV cache)."""
    f = cache.view(torch.float8_e4m3fn).float()
    return f * scale[..., None]


def attention_prefill_paged(
    q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
    block_table: torch.Tensor, kv_lens: torch.Tensor, q_lens: torch.Tensor,
    scale: Optional[float] = None,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Chunked-prefill attention: chunk queries [B, Sq, H, D] (bshd) attend
    causally to the full PAGED history (kv_lens keys per seq, already in the
    cache); q_lens masks per-seq chunk padding. Output [B, Sq, H, D].

    fp8 KV: uint8 e4m3 caches + per-token-per-head ``k_scale``/``v_scale``
    float32 [NB, Hkv, BS]; the gfx950 kernel dequantizes while staging tiles
    to LDS (attention_v2.hip FP8KV)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        ext = _require_ext("attention_prefill_paged")
        if ext is not None:
            if os.environ.get("CMLS_ATTN_V2", "1") != "0":
                return ext.attention_prefill_paged_v2(
                    q, k_cache, v_cache, block_table, kv_lens, q_lens,
                    float(scale), k_scale, v_scale)
            if k_scale is not None:
                raise RuntimeError("fp8 KV prefill needs the v2 kernel "
                                   "(CMLS_ATTN_V2=0 set?)")
            return ext.attention_prefill_paged(
                q, k_cache, v_cache, block_table, kv_lens, q_lens,
                float(scale))
    if k_scale is not None:
        return attention_prefill_paged(
            q, _dequant_kv_cpu(k_cache, k_scale),
            _dequant_kv_cpu(v_cache, v_scale), block_table, kv_lens, q_lens,
            scale)
    # reference path: gather pages -> dense causal attention with history
    b, sq, h, d = q.shape
    hkv = k_cache.shape[1]
    bs = k_cache.shape[2]
    out = torch.zeros_like(q)
    for i in range(b):
        n = int(kv_lens[i])
        ql = int(q_lens[i])
        nb = (n + bs - 1) // bs
        blocks = block_table[i, :nb].long()
        k = k_cache[blocks].float().permute(1, 0, 2, 3).reshape(
            hkv, nb * bs, d)[:, :n]
        v = v_cache[blocks].float().permute(1, 0, 2, 3).reshape(
            hkv, nb * bs, d)[:, :n]
        o = attention(
            q[i, :ql].permute(1, 0, 2)[None].float(),
            k[None], v[None], causal=True, scale=scale)
        out[i, :ql] = o[0].permute(1, 0, 2).to(q.dtype)
    return out


def attention_decode(
    q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
    block_table: torch.Tensor, seq_lens: torch.Tensor,
    scale: Optional[float] = None,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Paged-KV decode attention: one new token per sequence.

    q:        [B, H, D]         (current-step queries)
    k_cache:  [num_blocks, H_kv, block_size, D]
    v_cache:  [num_blocks, H_kv, block_size, D]
    block_table: int32 [B, max_blocks] physical block ids per sequence
    seq_lens: int32 [B] total keys per sequence (including current token)

    fp8 KV: uint8 e4m3 caches + float32 [NB, Hkv, BS] per-token scales;
    halves the decode KV-read bytes (the long-context bandwidth bound).
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        ext = _require_ext("attention_decode")
        if ext is not None:
            return ext.attention_decode(q, k_cache, v_cache, block_table,
                                        seq_lens, float(scale),
                                        k_scale, v_scale)
    if k_scale is not None:
        return attention_decode(
            q, _dequant_kv_cpu(k_cache, k_scale),
            _dequant_kv_cpu(v_cache, v_scale), block_table, seq_lens, scale)
    # reference path: gather pages then dense attention per sequence
    bsz, hq, d = q.shape
    hkv = k_cache.shape[1]
    block_size = k_cache.shape[2]
    rep = hq // hkv
    out = torch.empty_like(q, dtype=q.dtype)
    qf = q.float()
    for b in range(bsz):
        n = int(seq_lens[b])
        nblocks = (n + block_size - 1) // block_size
        blocks = block_table[b, :nblocks].long()
        k = k_cache[blocks].float()  # [nb, hkv, bs, d]
        v = v_cache[blocks].float()
        k = k.permute(1, 0, 2, 3).reshape(hkv, nblocks * block_size, d)[:, :n]
        v = v.permute(1, 0, 2, 3).reshape(hkv, nblocks * block_size, d)[:, :n]
        if rep > 1:
            k = k.repeat_interleave(rep, dim=0)
            v = v.repeat_interleave(rep, dim=0)
        scores = torch.einsum("hd,hnd->hn", qf[b], k) * scale
        probs = torch.softmax(scores, dim=-1)
        out[b] = torch.einsum("hn,hnd->hd", probs, v).to(q.dtype)
    return out


def kv_cache_write(
    k_new: torch.Tensor, v_new: torch.Tensor, k_cache: torch.Tensor,
    v_cache: torch.Tensor, slot_mapping: torch.Tensor,
    k_scale: Optional[torch.Tensor] = None,
    v_scale: Optional[torch.Tensor] = None,
) -> None:
    """Scatter [T, H_kv, D] new keys/values into the paged caches at flat
    slots (block_id * block_size + offset); slot -1 skips the token.

    fp8 KV: pass uint8 caches + float32 [NB, Hkv, BS] scale tensors; each
    (token, head) row is absmax-quantized to e4m3 and its scale recorded."""
    if k_new.is_cuda:
        ext = _require_ext("kv_cache_write")
        if ext is not None:
            ext.kv_cache_write(k_new, v_new, k_cache, v_cache, slot_mapping,
                               k_scale, v_scale)
            return
    block_size = k_cache.shape[2]
    fp8 = k_scale is not None
    for t in range(k_new.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        blk, off = slot // block_size, slot % block_size
        if fp8:
            for cache, scales, new in ((k_cache, k_scale, k_new),
                                       (v_cache, v_scale, v_new)):
                row = new[t].float()  # [H_kv, D]
                amax = row.abs().amax(dim=-1).clamp_min(1e-8)
                # pow2 scales match the GPU quantizer (scalef32 dequant
                # applies only the scale exponent -- MX semantics)
                sc = torch.exp2(torch.ceil(torch.log2(amax / 448.0)))
                q8 = (row / sc[:, None]).to(torch.float8_e4m3fn)
                cache[blk, :, off] = q8.view(torch.uint8)
                scales[blk, :, off] = sc
        else:
            k_cache[blk, :, off] = k_new[t]
            v_cache[blk, :, off] = v_new[t]


# --------------------------------------------------------------------- #
# Rotary embedding
# --------------------------------------------------------------------- #
def rope_inplace(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
    theta: float = 10000.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Apply rotary position embedding in place (NeoX interleaving: rotate
    pairs (i, i+D/2)). q: [T, H, D], k: [T, H_kv, D], positions: int32 [T]."""
    if q.is_cuda:
        ext = _require_ext("rope")
        if ext is not None:
            ext.rope_inplace(q, k, positions, float(theta))
            return q, k
    d = q.shape[-1]
    half = d // 2
    inv_freq = 1.0 / (theta ** (torch.arange(half, dtype=torch.float32,
                                             device=q.device) / half))
    angles = positions.float()[:, None] * inv_freq[None, :]  # [T, half]
    cos = angles.cos()[:, None, :]
    sin = angles.sin()[:, None, :]
    for t in (q, k):
        # clone: .float() on an fp32 tensor returns SELF, and lo/hi must not
        # alias the destination (in-place write of :half corrupted hi's input)
        tf = t.float().clone()
        lo, hi = tf[..., :half], tf[..., half:]
        t[..., :half] = (lo * cos - hi * sin).to(t.dtype)
        t[..., half:] = (hi * cos + lo * sin).to(t.dtype)
    return q, k


# --------------------------------------------------------------------- #
# Sampling
# --------------------------------------------------------------------- #
def sample_top_k_top_p(
    logits: torch.Tensor, temperature: float = 1.0, top_k: int = 0,
    top_p: float = 1.0, generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Sample token ids from [B, V] logits with temperature/top-k/top-p.

    Greedy when temperature == 0. Returns int64 [B].
    """
    if temperature == 0.0:
        return logits.argmax(dim=-1)
    if logits.is_cuda:
        ext = _require_ext("sample")
        if ext is not None:
            seed = int(torch.randint(0, 2**31 - 1, (1,),
                                     generator=generator).item())
            return ext.sample_top_k_top_p(
                logits, float(temperature), int(top_k), float(top_p), seed)
    lf = logits.float() / temperature
    if top_k and top_k < lf.shape[-1]:
        kth = torch.topk(lf, top_k, dim=-1).values[..., -1:]
        lf = lf.masked_fill(lf < kth, float("-inf"))
    if top_p < 1.0:
        sorted_logits, idx = torch.sort(lf, descending=True, dim=-1)
        probs = torch.softmax(sorted_logits, dim=-1)
        # drop tokens whose inclusive cumulative prob exceeds p; always keep
        # the top token
        cutoff = probs.cumsum(dim=-1) > top_p
        cutoff[..., 0] = False
        sorted_logits = sorted_logits.masked_fill(cutoff, float("-inf"))
        lf = torch.full_like(lf, float("-inf")).scatter(-1, idx, sorted_logits)
    probs = torch.softmax(lf, dim=-1)
    return torch.multinomial(probs, 1, generator=generator).squeeze(-1)


# --------------------------------------------------------------------- #
# GEMM (bf16 MFMA)
# --------------------------------------------------------------------- #
def gemm_bf16(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """C[M,N] = A[M,K] @ B[N,K]^T in bf16 with fp32 accumulation on MFMA
    (nn.Linear orientation: b is a [out, in] weight matrix).

    STATUS: test fixture / benchmark exhibit, NOT routed in any serving
    model. Measured 867 TF vs hipBLASLt's 1362 TF on large dense shapes
    (profiles/README.md §4), so production GEMMs go through torch.matmul
    (hipBLASLt); the decode-shaped regime where hand-written kernels DO win
    is covered by skinny_gemm / skinny_gemm_fp8 / conv3x3_nhwc, all routed.
    Kept (with tests + kernel_bench entry) as the dense-tile reference the
    conv kernel's tiling derives from. Shapes off the 128/128/32 tile grid
    are zero-padded here.
    """
    if a.is_cuda:
        ext = _require_ext("gemm_bf16")
        if ext is not None:
            m, k = a.shape
            n = b.shape[0]
            pm, pn, pk = (-m) % 128, (-n) % 128, (-k) % 32
            if pm or pn or pk:
                a2 = torch.nn.functional.pad(a, (0, pk, 0, pm))
                b2 = torch.nn.functional.pad(b, (0, pk, 0, pn))
                return ext.gemm_bf16(a2.contiguous(),
                                     b2.contiguous())[:m, :n].contiguous()
            return ext.gemm_bf16(a.contiguous(), b.contiguous())
    return (a.float() @ b.float().T).to(a.dtype)


def skinny_linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """y = x @ weight.T for decode-shaped activations (rows <= 128).

    Routes to the in-tree skinny MFMA kernel on GPU exactly where it
    MEASURES faster than hipBLASLt (profiles/skinny_bench.txt): the
    small-N llama projections at M <= 16 (qkv 1.2x, o_proj 2.0-2.8x --
    hipBLASLt's tiles under-fill the 256-CU chip there) and the large-K
    down projection at M <= 2. Everything else falls back to
    torch.nn.functional.linear (hipBLASLt).
    """
    if (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 2
            and x.shape[1] % 32 == 0 and weight.shape[0] % 16 == 0
            and os.environ.get("CMLS_SKINNY", "1") != "0"):
        m, k = x.shape
        n = weight.shape[0]
        if ((1 <= m <= 16 and n <= 8192 and k <= 8192)
                or (m <= 2 and n <= 8192)):
            ext = _require_ext("skinny_gemm")
            if ext is not None:
                return ext.skinny_gemm(x.contiguous(), weight, 1)
    return torch.nn.functional.linear(x, weight)
