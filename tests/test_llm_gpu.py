"""GPU tests: paged decode attention numerics + end-to-end LLM generation."""

import asyncio

import pytest
import torch

import clearml_serving_amd.ops as ops
from clearml_serving_amd.engines.llm.engine import (
    LlmEngine,
    LlmEngineConfig,
    SamplingParams,
)

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        # cancel lingering workers (batcher/engine loops) before closing
        pending = asyncio.all_tasks(loop)
        for t in pending:
            t.cancel()
        if pending:
            loop.run_until_complete(
                asyncio.gather(*pending, return_exceptions=True))
        loop.close()


@pytest.mark.parametrize("b,h,hkv,d,seqs", [
    (4, 32, 8, 128, [1, 17, 200, 1000]),
    (2, 8, 8, 64, [33, 64]),
    (3, 16, 2, 128, [5, 130, 257]),
])
def test_attention_decode_numerics(b, h, hkv, d, seqs):
    torch.manual_seed(0)
    block_size = 16
    max_blocks = (max(seqs) + block_size - 1) // block_size
    nblocks = b * max_blocks + 1
    k_cache = torch.randn(nblocks, hkv, block_size, d, device=DEV,
                          dtype=torch.bfloat16)
    v_cache = torch.randn(nblocks, hkv, block_size, d, device=DEV,
                          dtype=torch.bfloat16)
    # shuffled physical block assignment
    perm = torch.randperm(nblocks - 1) + 1
    block_table = perm[:b * max_blocks].reshape(b, max_blocks).to(
        torch.int32).to(DEV)
    seq_lens = torch.tensor(seqs, dtype=torch.int32, device=DEV)
    q = torch.randn(b, h, d, device=DEV, dtype=torch.bfloat16)

    got = ops.attention_decode(q, k_cache, v_cache, block_table, seq_lens)
    ref = ops.attention_decode(
        q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
        block_table.cpu(), seq_lens.cpu())
    torch.testing.assert_close(got.float().cpu(), ref, atol=3e-2, rtol=3e-2)


def test_kv_cache_write_scatter():
    t, hkv, d, bs = 7, 4, 128, 16
    k_new = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    v_new = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    k_cache = torch.zeros(4, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    v_cache = torch.zeros(4, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    slots = torch.tensor([0, 1, 15, 16, 33, 63, -1], dtype=torch.int32,
                         device=DEV)
    ops.kv_cache_write(k_new, v_new, k_cache, v_cache, slots)
    torch.testing.assert_close(k_cache[0, :, 0], k_new[0])
    torch.testing.assert_close(k_cache[0, :, 1], k_new[1])
    torch.testing.assert_close(k_cache[0, :, 15], k_new[2])
    torch.testing.assert_close(k_cache[1, :, 0], k_new[3])
    torch.testing.assert_close(v_cache[2, :, 1], v_new[4])
    torch.testing.assert_close(v_cache[3, :, 15], v_new[5])
    assert k_cache[3, :, 14].abs().sum() == 0  # slot -1 skipped


def test_llm_engine_gpu_generation():
    cfg = LlmEngineConfig(preset="llama-3-1b", num_kv_blocks=2048,
                          block_size=16, max_model_len=1024, device=DEV)
    eng = LlmEngine(cfg)
    eng.start()

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=16,
                                ignore_eos=True)

        async def one(text):
            toks = []
            async for item in eng.generate(text, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one("prompt %d" % i) for i in range(4)])

    outs = run(main())
    assert all(len(o) == 16 for o in outs)
    # determinism under greedy on GPU
    outs2 = run(main())
    assert outs == outs2
    assert eng.allocator.available == eng.allocator.num_blocks


def test_llm_engine_gpu_decode_matches_prefill():
    """GPU paged decode vs teacher-forced prefill on the same weights."""
    cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=256,
                          block_size=16, max_model_len=256, device=DEV)
    eng = LlmEngine(cfg)
    eng.start()
    prompt = [3, 7, 11, 19, 23, 29, 31, 37]

    async def gen():
        seq = await eng.add_request(
            list(prompt), SamplingParams(temperature=0.0, max_tokens=8,
                                         ignore_eos=True))
        toks = []
        while True:
            item = await seq.stream.get()
            toks.extend(item["token_ids"])
            if item["finished"]:
                return toks

    generated = run(gen())
    full = prompt + generated
    t = len(full)
    tokens = torch.tensor(full, dtype=torch.long, device=DEV)
    positions = torch.arange(t, dtype=torch.int32, device=DEV)
    attn_ctx = {"mode": "prefill", "batch": 1, "seq": t,
                "seq_lens": torch.tensor([t], dtype=torch.int32, device=DEV),
                "slot_mapping": torch.full((t,), -1, dtype=torch.int32,
                                           device=DEV)}
    with torch.inference_mode():
        logits = eng.model(tokens, positions, kv_caches=None,
                           attn_ctx=attn_ctx)
    # bf16 nondeterminism tolerance: compare argmax agreement, allowing ties
    agree = 0
    for step in range(8):
        pos = len(prompt) + step - 1
        top2 = logits[pos].float().topk(2)
        if generated[step] == int(top2.indices[0]) or \
                (top2.values[0] - top2.values[1]) < 0.05:
            agree += 1
    assert agree >= 7, "paged decode diverges from teacher-forced prefill"


def test_attention_prefill_paged_numerics():
    torch.manual_seed(11)
    b, h, hkv, d, bs = 2, 8, 2, 128, 16
    hist = [40, 7]          # cached history lengths
    chunk = [24, 16]        # new chunk lengths (padded batch to 24)
    sq = max(chunk)
    nb = 8
    k_cache = torch.randn(nb, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    v_cache = torch.randn_like(k_cache)
    block_table = torch.tensor([[0, 2, 4, 6], [1, 3, 5, 7]],
                               dtype=torch.int32, device=DEV)
    kv_lens = torch.tensor([hist[0] + chunk[0], hist[1] + chunk[1]],
                           dtype=torch.int32, device=DEV)
    q_lens = torch.tensor(chunk, dtype=torch.int32, device=DEV)
    q = torch.randn(b, sq, h, d, device=DEV, dtype=torch.bfloat16)

    got = ops.attention_prefill_paged(q, k_cache, v_cache, block_table,
                                      kv_lens, q_lens)
    ref = ops.attention_prefill_paged(
        q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
        block_table.cpu(), kv_lens.cpu(), q_lens.cpu())
    for i in range(b):
        torch.testing.assert_close(got[i, :chunk[i]].float().cpu(),
                                   ref[i, :chunk[i]].float(),
                                   atol=3e-2, rtol=3e-2)


def test_chunked_prefill_gpu_matches_unchunked():
    prompt = [(7 * i + 3) % 400 for i in range(200)]

    def gen(chunk):
        torch.manual_seed(77)
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=256,
                              block_size=16, max_model_len=512,
                              prefill_chunk=chunk, device=DEV)
        eng = LlmEngine(cfg)
        eng.start()

        async def go():
            seq = await eng.add_request(list(prompt), SamplingParams(
                temperature=0.0, max_tokens=8, ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go())

    full = gen(1024)
    chunked = gen(48)
    assert full == chunked


def test_fp8_quantized_engine():
    """fp8-weight llama serves correctly (experimental path; token-level
    agreement with bf16 is NOT asserted on random-init weights -- their
    near-uniform logit margins sit below fp8 quantization error)."""
    def gen(quant):
        torch.manual_seed(31)
        cfg = LlmEngineConfig(preset="llama-3-1b", num_kv_blocks=1024,
                              block_size=16, max_model_len=512, device=DEV,
                              quantization=quant)
        eng = LlmEngine(cfg)
        eng.start()

        async def go():
            seq = await eng.add_request(
                [(i * 13 + 7) % 1000 for i in range(64)],
                SamplingParams(temperature=0.0, max_tokens=12,
                               ignore_eos=True))
            toks = []
            while True:
                item = await seq.stream.get()
                toks.extend(item["token_ids"])
                if item["finished"]:
                    return toks

        return run(go())

    fp8 = gen("fp8")
    assert len(fp8) == 12
    assert all(isinstance(t, int) for t in fp8)


def test_embeddings_gpu_matches_cpu():
    """embed_batch on GPU (HIP bshd attention + rmsnorm) agrees with the
    same engine run on CPU (fp32 torch reference ops)."""
    def emb(device):
        torch.manual_seed(17)
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=64,
                              block_size=16, max_model_len=128, device=device)
        eng = LlmEngine(cfg)
        eng.start()
        out = run(eng.embed_batch(["embedding parity probe",
                                   "second, longer text for the batch"]))
        return torch.tensor(out)

    g = emb(DEV)
    c = emb("cpu")
    # bf16 GPU vs fp32 CPU: compare direction (the semantic content of a
    # normalized embedding), not elementwise values
    cos = (g * c).sum(-1)
    assert (cos > 0.99).all(), cos


def test_decode_graphs_match_eager():
    """Greedy generation with per-bucket decode hipGraphs must be token-
    identical to eager decode (same weights, same prompts)."""
    def gen(graphs):
        torch.manual_seed(5)
        cfg = LlmEngineConfig(preset="llama-tiny", num_kv_blocks=128,
                              block_size=16, max_model_len=256, device=DEV,
                              max_num_seqs=8, decode_graphs=graphs)
        eng = LlmEngine(cfg)
        eng.start()

        async def go():
            prompts = [[(i * 7 + j) % 500 for j in range(9 + i)]
                       for i in range(5)]
            outs = [[] for _ in prompts]

            async def one(i):
                seq = await eng.add_request(prompts[i], SamplingParams(
                    temperature=0.0, max_tokens=24, ignore_eos=True))
                while True:
                    item = await seq.stream.get()
                    outs[i].extend(item["token_ids"])
                    if item["finished"]:
                        return

            await asyncio.gather(*[one(i) for i in range(len(prompts))])
            return outs

        return run(go())

    eager = gen(False)
    graphed = gen(True)
    assert eager == graphed


# --------------------------------------------------------------------- #
# fp8 KV cache (attention_decode.hip FP8KV / kv_cache_write_fp8_kernel /
# attention_v2.hip FP8KV paged prefill)
# --------------------------------------------------------------------- #
def test_kv_cache_write_fp8_matches_cpu_quantizer():
    torch.manual_seed(4)
    t, hkv, d, bs, nb = 9, 4, 128, 16, 4
    k_new = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16) * 2
    v_new = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    k8 = torch.zeros(nb, hkv, bs, d, device=DEV, dtype=torch.uint8)
    v8 = torch.zeros(nb, hkv, bs, d, device=DEV, dtype=torch.uint8)
    ks = torch.ones(nb, hkv, bs, device=DEV)
    vs = torch.ones(nb, hkv, bs, device=DEV)
    slots = torch.tensor([0, 1, 15, 16, 33, 63, -1, 40, 5],
                         dtype=torch.int32, device=DEV)
    ops.kv_cache_write(k_new, v_new, k8, v8, slots, ks, vs)
    # CPU reference quantizer on the same inputs
    k8c = torch.zeros(nb, hkv, bs, d, dtype=torch.uint8)
    v8c = torch.zeros(nb, hkv, bs, d, dtype=torch.uint8)
    ksc = torch.ones(nb, hkv, bs)
    vsc = torch.ones(nb, hkv, bs)
    ops.kv_cache_write(k_new.cpu(), v_new.cpu(), k8c, v8c, slots.cpu(),
                       ksc, vsc)
    # GPU and CPU quantizers may differ on round-to-nearest ties (different
    # fp32 intermediate orderings), so compare each against the SOURCE rows
    # within the e4m3 bound, and require near-total byte agreement
    for g8, gsc, c8, csc, new in ((k8, ks, k8c, ksc, k_new),
                                  (v8, vs, v8c, vsc, v_new)):
        gdq = ops._dequant_kv_cpu(g8.cpu(), gsc.cpu())
        for i, s in enumerate(slots.tolist()):
            if s < 0:
                continue
            blk, off = s // bs, s % bs
            row = new[i].float().cpu()
            tol = row.abs().amax(dim=-1, keepdim=True) / 16.0 + 1e-6
            assert ((gdq[blk, :, off] - row).abs() <= tol).all()
        match = (g8.cpu() == c8).float().mean().item()
        assert match > 0.999, "byte agreement {:.4f}".format(match)
    assert (k8[3, :, 14] == 0).all()  # slot -1 skipped


@pytest.mark.parametrize("b,h,hkv,d,seqs", [
    (4, 32, 8, 128, [1, 17, 200, 1000]),
    (2, 8, 8, 64, [33, 64]),
])
def test_attention_decode_fp8_kv_numerics(b, h, hkv, d, seqs):
    torch.manual_seed(5)
    block_size = 16
    max_blocks = (max(seqs) + block_size - 1) // block_size
    nblocks = b * max_blocks + 1
    kf = torch.randn(nblocks, hkv, block_size, d, device=DEV,
                     dtype=torch.bfloat16)
    vf = torch.randn(nblocks, hkv, block_size, d, device=DEV,
                     dtype=torch.bfloat16)
    k8 = torch.zeros(nblocks, hkv, block_size, d, device=DEV,
                     dtype=torch.uint8)
    v8 = torch.zeros_like(k8)
    ks = torch.ones(nblocks, hkv, block_size, device=DEV)
    vs = torch.ones_like(ks)
    slots = torch.arange(nblocks * block_size, dtype=torch.int32, device=DEV)
    ops.kv_cache_write(
        kf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
        vf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
        k8, v8, slots, ks, vs)
    perm = torch.randperm(nblocks - 1) + 1
    block_table = perm[:b * max_blocks].reshape(b, max_blocks).to(
        torch.int32).to(DEV)
    seq_lens = torch.tensor(seqs, dtype=torch.int32, device=DEV)
    q = torch.randn(b, h, d, device=DEV, dtype=torch.bfloat16)
    got = ops.attention_decode(q, k8, v8, block_table, seq_lens,
                               k_scale=ks, v_scale=vs)
    # fp32 CPU reference on the DEQUANTIZED cache isolates the kernel's
    # fp8 read path from the quantization error itself
    ref = ops.attention_decode(
        q.float().cpu(), ops._dequant_kv_cpu(k8.cpu(), ks.cpu()),
        ops._dequant_kv_cpu(v8.cpu(), vs.cpu()),
        block_table.cpu(), seq_lens.cpu())
    torch.testing.assert_close(got.float().cpu(), ref, atol=3e-2, rtol=3e-2)


def test_attention_prefill_paged_fp8_kv_numerics():
    torch.manual_seed(6)
    bsz, sq, h, hkv, d, bs = 2, 32, 8, 2, 128, 16
    kv_lens = torch.tensor([200, 77], dtype=torch.int32)
    q_lens = torch.tensor([32, 20], dtype=torch.int32)
    max_blocks = (200 + bs - 1) // bs
    nblocks = bsz * max_blocks + 1
    kf = torch.randn(nblocks, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    vf = torch.randn(nblocks, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    k8 = torch.zeros(nblocks, hkv, bs, d, device=DEV, dtype=torch.uint8)
    v8 = torch.zeros_like(k8)
    ks = torch.ones(nblocks, hkv, bs, device=DEV)
    vs = torch.ones_like(ks)
    slots = torch.arange(nblocks * bs, dtype=torch.int32, device=DEV)
    ops.kv_cache_write(
        kf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
        vf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
        k8, v8, slots, ks, vs)
    perm = torch.randperm(nblocks - 1) + 1
    block_table = perm[:bsz * max_blocks].reshape(bsz, max_blocks).to(
        torch.int32).to(DEV)
    q = torch.randn(bsz, sq, h, d, device=DEV, dtype=torch.bfloat16)
    got = ops.attention_prefill_paged(
        q, k8, v8, block_table, kv_lens.to(DEV), q_lens.to(DEV),
        k_scale=ks, v_scale=vs)
    ref = ops.attention_prefill_paged(
        q.float().cpu(), ops._dequant_kv_cpu(k8.cpu(), ks.cpu()),
        ops._dequant_kv_cpu(v8.cpu(), vs.cpu()), block_table.cpu(),
        kv_lens, q_lens)
    for i in range(bsz):
        n = int(q_lens[i])
        torch.testing.assert_close(got[i, :n].float().cpu(), ref[i, :n],
                                   atol=3e-2, rtol=3e-2)


def test_llm_engine_gpu_fp8_kv_generation():
    """End-to-end: fp8 weights + fp8 KV cache decode on the GPU."""
    cfg = LlmEngineConfig(preset="llama-3-1b", num_kv_blocks=2048,
                          block_size=16, max_model_len=1024, device=DEV,
                          kv_dtype="fp8")
    eng = LlmEngine(cfg)
    eng.start()
    assert eng.kv_caches[0][0].dtype == torch.uint8

    async def main():
        params = SamplingParams(temperature=0.0, max_tokens=16,
                                ignore_eos=True)

        async def one(text):
            toks = []
            async for item in eng.generate(text, params):
                toks.extend(item["token_ids"])
            return toks

        return await asyncio.gather(*[one("prompt %d" % i) for i in range(4)])

    outs = run(main())
    eng.stop()
    assert all(len(o) == 16 for o in outs)
