"""GPU numerics: every HIP kernel vs a plain PyTorch fp32 reference.

The reference implementations live in ops/__init__.py (CPU dispatch path);
each test computes the op in fp32 on the same data and compares within
bf16-accumulation tolerances.
"""

import math

import pytest
import torch

import clearml_serving_amd.ops as ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def ref_f32(fn, *tensors, **kw):
    cpu = [t.detach().float().cpu() if isinstance(t, torch.Tensor) else t
           for t in tensors]
    return fn(*cpu, **kw)


def assert_close_bf16(got, ref, atol=2e-2, rtol=2e-2):
    got = got.detach().float().cpu()
    ref = ref.detach().float().cpu()
    torch.testing.assert_close(got, ref, atol=atol, rtol=rtol)


def test_extension_is_loaded():
    # on a GPU box the extension must be present -- no silent eager fallback
    assert ops.has_extension(), "HIP extension missing on GPU box"


@pytest.mark.parametrize("shape", [(8, 64, 56, 56), (2, 2048, 7, 7)])
def test_relu_add(shape):
    x = torch.randn(shape, device=DEV, dtype=torch.bfloat16)
    r = torch.randn(shape, device=DEV, dtype=torch.bfloat16)
    got = ops.bias_relu_add(x, residual=r)
    ref = torch.relu(x.float() + r.float())
    assert_close_bf16(got, ref)


def test_bias_relu_add_channel_bias():
    x = torch.randn(4, 64, 14, 14, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(64, device=DEV, dtype=torch.bfloat16)
    got = ops.bias_relu_add(x, bias=b)
    ref = torch.relu(x.float() + b.float()[None, :, None, None])
    assert_close_bf16(got, ref)


@pytest.mark.parametrize("rows,cols", [(128, 3072), (1000, 768)])
def test_bias_gelu(rows, cols):
    x = torch.randn(rows, cols, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(cols, device=DEV, dtype=torch.bfloat16)
    got = ops.bias_gelu(x, b)
    ref = torch.nn.functional.gelu(x.float() + b.float())
    assert_close_bf16(got, ref)


def test_silu_mul():
    g = torch.randn(64, 11008, device=DEV, dtype=torch.bfloat16)
    u = torch.randn(64, 11008, device=DEV, dtype=torch.bfloat16)
    got = ops.silu_mul(g, u)
    ref = torch.nn.functional.silu(g.float()) * u.float()
    assert_close_bf16(got, ref)


@pytest.mark.parametrize("h", [768, 1024, 4096, 8192])
def test_layernorm(h):
    x = torch.randn(64, h, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(h, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(h, device=DEV, dtype=torch.bfloat16)
    got = ops.layernorm(x, w, b)
    ref = torch.nn.functional.layer_norm(
        x.float(), (h,), w.float(), b.float(), 1e-5)
    assert_close_bf16(got, ref)


def test_layernorm_fused_residual():
    x = torch.randn(32, 768, device=DEV, dtype=torch.bfloat16)
    r = torch.randn(32, 768, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(768, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(768, device=DEV, dtype=torch.bfloat16)
    got = ops.layernorm(x, w, b, residual=r)
    ref = torch.nn.functional.layer_norm(
        (x.float() + r.float()), (768,), w.float(), b.float(), 1e-5)
    assert_close_bf16(got, ref)


@pytest.mark.parametrize("h", [4096, 8192])
def test_rmsnorm(h):
    x = torch.randn(16, h, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(h, device=DEV, dtype=torch.bfloat16)
    got = ops.rmsnorm(x, w)
    xf = x.float()
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    assert_close_bf16(got, ref)


def test_rmsnorm_residual_update():
    x = torch.randn(8, 4096, device=DEV, dtype=torch.bfloat16)
    r = torch.randn(8, 4096, device=DEV, dtype=torch.bfloat16)
    r_ref = (x.float() + r.float())
    w = torch.ones(4096, device=DEV, dtype=torch.bfloat16)
    got = ops.rmsnorm(x, w, residual=r)
    ref = r_ref * torch.rsqrt(r_ref.pow(2).mean(-1, keepdim=True) + 1e-6)
    assert_close_bf16(got, ref)
    # residual stream updated in place to x + r
    assert_close_bf16(r, r_ref)


@pytest.mark.parametrize("shape", [(32, 1000), (12, 384, 384)])
def test_softmax(shape):
    x = torch.randn(shape, device=DEV, dtype=torch.bfloat16) * 4
    got = ops.softmax(x)
    ref = torch.softmax(x.float(), dim=-1)
    assert_close_bf16(got, ref, atol=1e-2)


# ------------------------------------------------------------------ #
# attention
# ------------------------------------------------------------------ #
def _attn_ref(q, k, v, causal=False, seq_lens=None):
    return ops.attention(q.float().cpu(), k.float().cpu(), v.float().cpu(),
                         causal=causal,
                         seq_lens=None if seq_lens is None else seq_lens.cpu())


@pytest.mark.parametrize("b,h,s,d", [(2, 12, 128, 64), (1, 8, 384, 64),
                                     (2, 4, 512, 128), (1, 2, 77, 64)])
def test_attention_prefill(b, h, s, d):
    torch.manual_seed(0)
    q = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    got = ops.attention(q, k, v)
    ref = _attn_ref(q, k, v)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


def test_attention_transpose_detecting():
    # asymmetric, non-random pattern that catches swapped row/col in the
    # C-write or operand maps (guide §5.4 rule 16)
    b, h, s, d = 1, 1, 64, 64
    q = torch.zeros(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.zeros(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.zeros(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    for i in range(s):
        q[0, 0, i, i % d] = 1.0
        k[0, 0, i, (i * 7) % d] = 1.0
        v[0, 0, i, :] = (i % 13) * 0.25 - 1.0
    got = ops.attention(q, k, v)
    ref = _attn_ref(q, k, v)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


def test_attention_causal():
    b, h, s, d = 2, 4, 256, 128
    torch.manual_seed(1)
    q = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    got = ops.attention(q, k, v, causal=True)
    ref = _attn_ref(q, k, v, causal=True)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


def test_attention_seq_lens_padding():
    b, h, s, d = 3, 2, 128, 64
    torch.manual_seed(2)
    q = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    seq_lens = torch.tensor([128, 60, 7], device=DEV, dtype=torch.int32)
    got = ops.attention(q, k, v, seq_lens=seq_lens)
    ref = _attn_ref(q, k, v, seq_lens=seq_lens)
    # only compare valid query rows (padded-query outputs are don't-care)
    for i, n in enumerate([128, 60, 7]):
        assert_close_bf16(got[i, :, :n], ref[i, :, :n], atol=3e-2, rtol=3e-2)


def test_attention_gqa():
    b, hq, hkv, s, d = 2, 8, 2, 128, 128
    torch.manual_seed(3)
    q = torch.randn(b, hq, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, hkv, s, d, device=DEV, dtype=torch.bfloat16)
    got = ops.attention(q, k, v)
    ref = _attn_ref(q, k, v)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


def test_attention_outlier_key_forces_rescale():
    # spike one K row so the running max jumps mid-sequence (forces the
    # online-softmax rescale branch, guide §5.4 rule 26)
    b, h, s, d = 1, 1, 256, 64
    torch.manual_seed(4)
    q = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    v = torch.randn(b, h, s, d, device=DEV, dtype=torch.bfloat16)
    k[0, 0, 200] = q[0, 0, 10] * 8  # huge score at tile 4 for query 10
    got = ops.attention(q, k, v)
    ref = _attn_ref(q, k, v)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


# ------------------------------------------------------------------ #
# rope / sampling
# ------------------------------------------------------------------ #
def test_rope():
    t, h, hkv, d = 64, 8, 2, 128
    torch.manual_seed(5)
    q = torch.randn(t, h, d, device=DEV, dtype=torch.bfloat16)
    k = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    pos = torch.arange(100, 100 + t, device=DEV, dtype=torch.int32)
    q_ref = q.float().cpu().clone()
    k_ref = k.float().cpu().clone()
    ops.rope_inplace(q, k, pos)
    ops.rope_inplace(q_ref, k_ref, pos.cpu())
    assert_close_bf16(q, q_ref, atol=3e-2, rtol=3e-2)
    assert_close_bf16(k, k_ref, atol=3e-2, rtol=3e-2)


def test_sample_greedy_matches_argmax():
    torch.manual_seed(6)
    logits = torch.randn(8, 32000, device=DEV, dtype=torch.bfloat16)
    got = ops.sample_top_k_top_p(logits, temperature=0.0)
    assert torch.equal(got, logits.argmax(dim=-1))


def test_sample_respects_top_k():
    torch.manual_seed(7)
    logits = torch.randn(16, 1000, device=DEV, dtype=torch.float32)
    topk_sets = logits.topk(10, dim=-1).indices
    for _ in range(5):
        s = ops.sample_top_k_top_p(logits, temperature=1.0, top_k=10)
        for b in range(16):
            assert s[b].item() in topk_sets[b].tolist()


def test_sample_distribution_sanity():
    # two tokens with 9:1 odds: frequency must track the distribution
    logits = torch.full((512, 4), -1e9, device=DEV, dtype=torch.float32)
    logits[:, 0] = math.log(0.9)
    logits[:, 1] = math.log(0.1)
    counts = torch.zeros(4)
    for trial in range(20):
        s = ops.sample_top_k_top_p(logits, temperature=1.0)
        counts += torch.bincount(s.cpu(), minlength=4).float()
    frac = counts / counts.sum()
    assert abs(frac[0].item() - 0.9) < 0.03
    assert counts[2] == 0 and counts[3] == 0


# ------------------------------------------------------------------ #
# MFMA GEMM
# ------------------------------------------------------------------ #
@pytest.mark.parametrize("m,n,k", [(128, 128, 32), (256, 512, 1024),
                                   (100, 200, 300)])
def test_gemm_bf16(m, n, k):
    torch.manual_seed(0)
    a = torch.randn(m, k, device=DEV, dtype=torch.bfloat16)
    b = torch.randn(n, k, device=DEV, dtype=torch.bfloat16)
    got = ops.gemm_bf16(a, b)
    ref = a.float() @ b.float().T
    assert_close_bf16(got, ref, atol=5e-2, rtol=5e-2)


def test_gemm_bf16_transpose_detecting():
    # asymmetric B catches swapped C-write / operand maps (guide §3)
    m = n = k = 128
    a = torch.zeros(m, k, device=DEV, dtype=torch.bfloat16)
    b = torch.zeros(n, k, device=DEV, dtype=torch.bfloat16)
    for i in range(m):
        a[i, (i * 3) % k] = 1.0
    for j in range(n):
        b[j, (j * 7 + 1) % k] = float(j % 5 + 1) * 0.25
    got = ops.gemm_bf16(a, b)
    ref = a.float() @ b.float().T
    assert_close_bf16(got, ref, atol=1e-2, rtol=1e-2)


def test_attention_bshd_layout_strided_views():
    # [B, S, H, D] strided views into a merged projection (the serving path)
    b, s, h, d = 2, 128, 8, 64
    torch.manual_seed(9)
    qkv = torch.randn(b, s, 3 * h * d, device=DEV, dtype=torch.bfloat16)
    q, k, v = qkv.split([h * d, h * d, h * d], dim=-1)
    q = q.unflatten(-1, (h, d))
    k = k.unflatten(-1, (h, d))
    v = v.unflatten(-1, (h, d))
    got = ops.attention(q, k, v, layout="bshd")
    ref = ops.attention(
        q.permute(0, 2, 1, 3).float().cpu(),
        k.permute(0, 2, 1, 3).float().cpu(),
        v.permute(0, 2, 1, 3).float().cpu()).permute(0, 2, 1, 3)
    assert_close_bf16(got, ref, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("variant", [1, 2])
@pytest.mark.parametrize("m", [1, 7, 16, 64, 100, 128])
def test_skinny_gemm_matches_fp32(m, variant):
    torch.manual_seed(m)
    k, n = 256, 320
    x = (torch.randn(m, k, device=DEV) / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device=DEV) / 8).to(torch.bfloat16)
    ext = ops._require_ext("skinny_gemm")
    got = ext.skinny_gemm(x, w, variant).float().cpu()
    ref = x.float().cpu() @ w.float().cpu().T
    assert torch.allclose(got, ref, atol=5e-2, rtol=5e-2), \
        (got - ref).abs().max()


def test_skinny_linear_llama_shapes():
    """The production shapes route through the kernel and agree with
    hipBLASLt."""
    torch.manual_seed(0)
    for k, n in ((4096, 6144), (4096, 4096), (14336, 4096)):
        x = (torch.randn(64, k, device=DEV) / 16).to(torch.bfloat16)
        w = (torch.randn(n, k, device=DEV) / 16).to(torch.bfloat16)
        got = ops.skinny_linear(x, w).float()
        ref = torch.nn.functional.linear(x, w).float()
        assert torch.allclose(got, ref, atol=8e-2, rtol=8e-2), \
            (got - ref).abs().max()


def test_attention_causal_unequal_seq_lens():
    """Padded causal batch with per-sequence lengths: every valid query of a
    SHORT sequence must attend its full history (regression: the kernel's
    causal offset went negative for padded rows and silently truncated
    short sequences' attention -- caught by the embeddings parity test)."""
    torch.manual_seed(0)
    b, s, h, d = 3, 33, 4, 64
    q = (torch.randn(b, s, h, d, device=DEV) / 8).to(torch.bfloat16)
    k = (torch.randn(b, s, h, d, device=DEV) / 8).to(torch.bfloat16)
    v = (torch.randn(b, s, h, d, device=DEV) / 8).to(torch.bfloat16)
    lens = torch.tensor([22, 33, 5], dtype=torch.int32, device=DEV)
    got = ops.attention(q, k, v, causal=True, seq_lens=lens, layout="bshd")
    ref = ops.attention(q.cpu().float(), k.cpu().float(), v.cpu().float(),
                        causal=True, seq_lens=lens.cpu(), layout="bshd")
    for i in range(b):
        n = int(lens[i])
        gi = got[i, :n].float().cpu()
        ri = ref[i, :n].float()
        assert torch.allclose(gi, ri, atol=2e-2, rtol=2e-2), \
            (i, (gi - ri).abs().max())


# ------------------------------------------------------------------ #
# fp8 decode path (kernels: csrc/fp8_path.hip)
# ------------------------------------------------------------------ #
def _dequant(q8, s):
    return q8.view(torch.float8_e4m3fn).float() * s[:, None]


def test_quant_fp8_gpu_matches_reference():
    torch.manual_seed(0)
    x = (torch.randn(5, 512, device=DEV) * 3).to(torch.bfloat16)
    q, s = ops.quant_fp8(x)
    q_ref, s_ref = ops._quant_fp8_ref(x.cpu())
    torch.testing.assert_close(s.cpu(), s_ref, atol=1e-6, rtol=1e-4)
    deq = _dequant(q.cpu(), s.cpu())
    deq_ref = _dequant(q_ref, s_ref)
    torch.testing.assert_close(deq, deq_ref, atol=0.05, rtol=0.05)


def test_rmsnorm_fp8_gpu_matches_reference():
    torch.manual_seed(1)
    x = torch.randn(4, 4096, device=DEV).to(torch.bfloat16)
    w = (torch.rand(4096, device=DEV) + 0.5).to(torch.bfloat16)
    res = torch.randn(4, 4096, device=DEV).to(torch.bfloat16)
    res_cpu = res.float().cpu()
    q, s = ops.rmsnorm_fp8(x, w, 1e-5, residual=res)
    q_ref, s_ref = ops.rmsnorm_fp8(x.float().cpu(), w.float().cpu(), 1e-5,
                                   residual=res_cpu)
    torch.testing.assert_close(s.cpu(), s_ref, atol=2e-3, rtol=2e-2)
    torch.testing.assert_close(_dequant(q.cpu(), s.cpu()),
                               _dequant(q_ref, s_ref), atol=0.08, rtol=0.08)
    # fused residual update matches
    torch.testing.assert_close(res.float().cpu(), res_cpu,
                               atol=2e-2, rtol=2e-2)


def test_silu_mul_fp8_gpu_strided_views():
    torch.manual_seed(2)
    inter = 1024
    merged = torch.randn(6, 2 * inter, device=DEV).to(torch.bfloat16)
    gate, up = merged.split([inter, inter], dim=-1)
    q, s = ops.silu_mul_fp8(gate, up)
    ref = (torch.nn.functional.silu(gate.float()) * up.float()).cpu()
    deq = _dequant(q.cpu(), s.cpu())
    torch.testing.assert_close(deq, ref, atol=0.08, rtol=0.08)


@pytest.mark.parametrize("m", [1, 3, 16, 33, 64])
def test_skinny_gemm_fp8_matches_dequant_matmul(m):
    torch.manual_seed(m)
    k, n = 256, 320
    a = (torch.randn(m, k, device=DEV) / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device=DEV) / 8).to(torch.bfloat16)
    a8, as_ = ops.quant_fp8(a)
    w8, ws = ops.quant_fp8(w)
    got = ops.skinny_gemm_fp8(a8, as_, w8, ws).float().cpu()
    ref = _dequant(a8.cpu(), as_.cpu()) @ _dequant(w8.cpu(), ws.cpu()).t()
    assert torch.allclose(got, ref, atol=5e-2, rtol=5e-2), \
        (got - ref).abs().max()


def test_skinny_gemm_fp8_asymmetric_identity():
    """A=I with asymmetric W catches any row/col-swapped fragment map."""
    k = 64
    a = torch.eye(16, k, device=DEV).to(torch.bfloat16)
    w = torch.arange(32 * k, device=DEV, dtype=torch.float32) \
        .reshape(32, k).to(torch.bfloat16) / (32 * k)
    a8, as_ = ops.quant_fp8(a)
    w8, ws = ops.quant_fp8(w)
    got = ops.skinny_gemm_fp8(a8, as_, w8, ws).float().cpu()
    ref = _dequant(a8.cpu(), as_.cpu()) @ _dequant(w8.cpu(), ws.cpu()).t()
    assert torch.allclose(got, ref, atol=3e-2, rtol=3e-2), \
        (got - ref).abs().max()


def test_fp8_llama_gpu_generation():
    """Full fp8 decode path on GPU: fused-quant kernels + fp8 skinny GEMM
    through the engine; outputs correlate with the bf16 model."""
    import asyncio

    from clearml_serving_amd.engines.llm.engine import (
        LlmEngine, LlmEngineConfig, SamplingParams)

    def gen(quant):
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                              block_size=16, max_model_len=128,
                              device=DEV, quantization=quant,
                              decode_graphs=False)
        eng = LlmEngine(cfg)
        eng.start()

        async def go():
            toks = []
            async for item in eng.generate(
                    "fp8 smoke", SamplingParams(temperature=0.0,
                                                max_tokens=8,
                                                ignore_eos=True)):
                toks.extend(item["token_ids"])
            return toks

        loop = asyncio.new_event_loop()
        out = loop.run_until_complete(go())
        eng.stop()
        return out

    bf16 = gen(None)
    fp8 = gen("fp8")
    assert len(bf16) == 8 and len(fp8) == 8
    # random-init tiny model: quantization noise may flip late tokens, but
    # the first greedy steps should agree
    assert bf16[0] == fp8[0]


# ------------------------------------------------------------------ #
# hand-written 3x3 conv (csrc/conv3x3.hip)
# ------------------------------------------------------------------ #
@pytest.mark.parametrize("n,c,k,w", [
    (2, 64, 64, 56),    # conv2-class
    (2, 128, 128, 28),  # conv3-class
    (3, 256, 256, 14),  # conv4-class
    (2, 96, 128, 28),   # C not equal K
])
def test_conv3x3_matches_fp32_reference(n, c, k, w):
    torch.manual_seed(n * 100 + w)
    x = (torch.randn(n, c, w, w, device=DEV) / 4).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device=DEV) / 8).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    b = torch.randn(k, device=DEV).to(torch.bfloat16)
    ext = ops._require_ext("conv3x3_nhwc")
    got = ext.conv3x3_nhwc(x, wt, b, False, None).float().cpu()
    ref = torch.nn.functional.conv2d(
        x.float().cpu(), wt.float().cpu(), b.float().cpu(),
        stride=1, padding=1)
    assert torch.allclose(got, ref, atol=6e-2, rtol=6e-2), \
        (got - ref).abs().max()


def test_conv3x3_fused_relu_residual():
    torch.manual_seed(7)
    n, c, k, w = 2, 64, 64, 56
    x = (torch.randn(n, c, w, w, device=DEV) / 4).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device=DEV) / 8).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    b = torch.randn(k, device=DEV).to(torch.bfloat16)
    res = (torch.randn(n, k, w, w, device=DEV) / 4).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    ext = ops._require_ext("conv3x3_nhwc")
    got = ext.conv3x3_nhwc(x, wt, b, True, res).float().cpu()
    ref = torch.relu(torch.nn.functional.conv2d(
        x.float().cpu(), wt.float().cpu(), b.float().cpu(), stride=1,
        padding=1) + res.float().cpu())
    assert torch.allclose(got, ref, atol=6e-2, rtol=6e-2), \
        (got - ref).abs().max()


def test_conv3x3_asymmetric_identity():
    """Delta-filter with asymmetric weights: catches tap/row/col mixups."""
    n, c, k, w = 1, 32, 64, 14
    x = torch.zeros(n, c, w, w, device=DEV, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    x[0, 3, 5, 9] = 1.0  # single impulse
    wt = torch.zeros(k, c, 3, 3, device=DEV, dtype=torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt[7, 3, 0, 2] = 2.0  # tap (dy=0, dx=2)
    ext = ops._require_ext("conv3x3_nhwc")
    got = ext.conv3x3_nhwc(x, wt, None, False, None).float().cpu()
    ref = torch.nn.functional.conv2d(x.float().cpu(), wt.float().cpu(),
                                     stride=1, padding=1)
    assert torch.allclose(got, ref, atol=1e-3), (got - ref).abs().max()


def test_resnet_stem_im2col_matches_conv2d():
    """StemConv (im2col+GEMM; avoids MIOpen's naive C=3 NHWC fallback)
    matches the conv2d reference."""
    from clearml_serving_amd.models.resnet import StemConv

    torch.manual_seed(0)
    stem = StemConv(3, 64, 7, stride=2, padding=3, relu=True)
    stem = stem.to(DEV).to(torch.bfloat16)
    x = (torch.randn(4, 3, 224, 224, device=DEV) / 4).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    with torch.inference_mode():
        got = stem(x).float()
        ref = torch.relu(torch.nn.functional.conv2d(
            x.float(), stem.conv.weight.float(), stem.conv.bias.float(),
            stride=2, padding=3))
    assert got.shape == ref.shape == (4, 64, 112, 112)
    assert got.is_contiguous(memory_format=torch.channels_last)
    assert torch.allclose(got, ref, atol=6e-2, rtol=6e-2), \
        (got - ref).abs().max()


@pytest.mark.parametrize("m", [17, 32, 48, 64])
def test_skinny_gemm_fp8_v2_matches_dequant_matmul(m):
    torch.manual_seed(m)
    k, n = 256, 320
    a = (torch.randn(m, k, device=DEV) / 8).to(torch.bfloat16)
    w = (torch.randn(n, k, device=DEV) / 8).to(torch.bfloat16)
    a8, as_ = ops.quant_fp8(a)
    w8, ws = ops.quant_fp8(w)
    got = ops.skinny_gemm_fp8_v2(a8, as_, w8, ws).float().cpu()
    ref = _dequant(a8.cpu(), as_.cpu()) @ _dequant(w8.cpu(), ws.cpu()).t()
    assert torch.allclose(got, ref, atol=5e-2, rtol=5e-2), \
        (got - ref).abs().max()


def test_skinny_gemm_fp8_v2_asymmetric_identity():
    k = 128
    a = torch.eye(32, k, device=DEV).to(torch.bfloat16)
    w = torch.arange(32 * k, device=DEV, dtype=torch.float32) \
        .reshape(32, k).to(torch.bfloat16) / (32 * k)
    a8, as_ = ops.quant_fp8(a)
    w8, ws = ops.quant_fp8(w)
    got = ops.skinny_gemm_fp8_v2(a8, as_, w8, ws).float().cpu()
    ref = _dequant(a8.cpu(), as_.cpu()) @ _dequant(w8.cpu(), ws.cpu()).t()
    assert torch.allclose(got, ref, atol=3e-2, rtol=3e-2), \
        (got - ref).abs().max()
