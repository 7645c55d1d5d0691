"""Decode-attention A/B: bf16 KV cache vs fp8 (e4m3 + per-token scales).

Long-context decode is KV-read bound; fp8 halves the bytes. Llama-3-8B
shapes (H=32, Hkv=8, D=128), B x S grid, CUDA-event timing.

Run on the GPU box:  python scripts/fp8kv_ab.py
"""

import sys

import torch

sys.path.insert(0, ".")
import clearml_serving_amd.ops as ops  # noqa: E402

DEV = "cuda:0"


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3  # us


def main():
    h, hkv, d, bs = 32, 8, 128, 16
    torch.manual_seed(0)
    print("%6s %6s %12s %12s %8s" % ("B", "S", "bf16_us", "fp8_us", "speedup"))
    for b, s in [(16, 1024), (16, 4096), (64, 1024), (64, 4096), (128, 2048)]:
        max_blocks = (s + bs - 1) // bs
        nb = b * max_blocks + 1
        kf = torch.randn(nb, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
        vf = torch.randn(nb, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
        k8 = torch.zeros(nb, hkv, bs, d, device=DEV, dtype=torch.uint8)
        v8 = torch.zeros_like(k8)
        ks = torch.ones(nb, hkv, bs, device=DEV)
        vs = torch.ones_like(ks)
        slots = torch.arange(nb * bs, dtype=torch.int32, device=DEV)
        ops.kv_cache_write(
            kf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
            vf.permute(0, 2, 1, 3).reshape(-1, hkv, d).contiguous(),
            k8, v8, slots, ks, vs)
        bt = (torch.randperm(nb - 1, device=DEV)[:b * max_blocks] + 1) \
            .reshape(b, max_blocks).to(torch.int32)
        sl = torch.full((b,), s, dtype=torch.int32, device=DEV)
        q = torch.randn(b, h, d, device=DEV, dtype=torch.bfloat16)

        t16 = bench(lambda: ops.attention_decode(q, kf, vf, bt, sl))
        t8 = bench(lambda: ops.attention_decode(q, k8, v8, bt, sl,
                                                k_scale=ks, v_scale=vs))
        # numerics spot check
        got = ops.attention_decode(q, k8, v8, bt, sl, k_scale=ks, v_scale=vs)
        ref = ops.attention_decode(q, kf, vf, bt, sl)
        err = (got - ref).abs().max().item()
        print("%6d %6d %12.1f %12.1f %7.2fx  maxerr=%.3f"
              % (b, s, t16, t8, t16 / t8, err))
        del kf, vf, k8, v8, ks, vs
        torch.cuda.empty_cache()

    # quantizer cost: kv_cache_write bf16 vs fp8 at decode token counts
    t, nb2 = 64, 512
    kn = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    vn = torch.randn(t, hkv, d, device=DEV, dtype=torch.bfloat16)
    kc16 = torch.zeros(nb2, hkv, bs, d, device=DEV, dtype=torch.bfloat16)
    vc16 = torch.zeros_like(kc16)
    kc8 = torch.zeros(nb2, hkv, bs, d, device=DEV, dtype=torch.uint8)
    vc8 = torch.zeros_like(kc8)
    ks2 = torch.ones(nb2, hkv, bs, device=DEV)
    vs2 = torch.ones_like(ks2)
    sl2 = torch.arange(t, dtype=torch.int32, device=DEV) * 7 % (nb2 * bs)
    w16 = bench(lambda: ops.kv_cache_write(kn, vn, kc16, vc16, sl2))
    w8 = bench(lambda: ops.kv_cache_write(kn, vn, kc8, vc8, sl2, ks2, vs2))
    print("kv_cache_write T=64: bf16 %.1fus  fp8-quant %.1fus" % (w16, w8))


if __name__ == "__main__":
    main()
