#!/usr/bin/env python3
"""Per-kernel microbenchmarks on MI355X (run under gpurun).

Reports achieved TFLOP/s (attention) and GB/s (memory-bound ops) so kernel
iterations can be compared run to run. Usage:
    python benchmarks/kernel_bench.py [--csv out.csv]
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import clearml_serving_amd.ops as ops  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_attention(results):
    for (b, h, s, d) in [(16, 12, 128, 64), (8, 12, 384, 64),
                         (16, 32, 1024, 128), (8, 32, 2048, 128)]:
        q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
        k = torch.randn_like(q)
        v = torch.randn_like(q)
        dt = timeit(lambda: ops.attention(q, k, v))
        flops = 4.0 * b * h * s * s * d  # QK^T + PV
        results.append(dict(op="attention_prefill",
                            shape="b{}h{}s{}d{}".format(b, h, s, d),
                            us=dt * 1e6, tflops=flops / dt / 1e12))
        dtc = timeit(lambda: ops.attention(q, k, v, causal=True))
        results.append(dict(op="attention_prefill_causal",
                            shape="b{}h{}s{}d{}".format(b, h, s, d),
                            us=dtc * 1e6, tflops=0.5 * flops / dtc / 1e12))


def bench_decode(results):
    bs = 16
    for (b, h, hkv, d, seq) in [(32, 32, 8, 128, 1024), (64, 32, 8, 128, 2048),
                                (128, 32, 8, 128, 4096)]:
        nb = b * ((seq + bs - 1) // bs)
        k_cache = torch.randn(nb, hkv, bs, d, device="cuda",
                              dtype=torch.bfloat16)
        v_cache = torch.randn_like(k_cache)
        bt = torch.arange(nb, dtype=torch.int32, device="cuda").reshape(b, -1)
        sl = torch.full((b,), seq, dtype=torch.int32, device="cuda")
        q = torch.randn(b, h, d, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: ops.attention_decode(q, k_cache, v_cache, bt, sl))
        bytes_moved = 2.0 * b * hkv * seq * d * 2  # K+V read, bf16
        results.append(dict(op="attention_decode",
                            shape="b{}h{}hkv{}d{}s{}".format(b, h, hkv, d, seq),
                            us=dt * 1e6, gbps=bytes_moved / dt / 1e9))


def bench_memops(results):
    for (rows, h) in [(8192, 768), (16384, 4096), (4096, 8192)]:
        x = torch.randn(rows, h, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(h, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(h, device="cuda", dtype=torch.bfloat16)
        r = torch.randn_like(x)
        dt = timeit(lambda: ops.layernorm(x, w, b, residual=r))
        bytes_moved = x.numel() * 2 * 3  # read x + r, write out
        results.append(dict(op="layernorm_residual",
                            shape="{}x{}".format(rows, h),
                            us=dt * 1e6, gbps=bytes_moved / dt / 1e9))
        dt = timeit(lambda: ops.rmsnorm(x, w))
        results.append(dict(op="rmsnorm", shape="{}x{}".format(rows, h),
                            us=dt * 1e6,
                            gbps=x.numel() * 2 * 2 / dt / 1e9))

    x = torch.randn(16384, 3072, device="cuda", dtype=torch.bfloat16)
    bb = torch.randn(3072, device="cuda", dtype=torch.bfloat16)
    dt = timeit(lambda: ops.bias_gelu(x, bb))
    results.append(dict(op="bias_gelu", shape="16384x3072", us=dt * 1e6,
                        gbps=x.numel() * 2 * 2 / dt / 1e9))

    g = torch.randn(8192, 14336, device="cuda", dtype=torch.bfloat16)
    u = torch.randn_like(g)
    dt = timeit(lambda: ops.silu_mul(g, u))
    results.append(dict(op="silu_mul", shape="8192x14336", us=dt * 1e6,
                        gbps=g.numel() * 2 * 3 / dt / 1e9))

    x = torch.randn(4, 64, 56, 56 * 64, device="cuda", dtype=torch.bfloat16)
    r = torch.randn_like(x)
    dt = timeit(lambda: ops.bias_relu_add(x, residual=r))
    results.append(dict(op="relu_add", shape="4x64x56x3584", us=dt * 1e6,
                        gbps=x.numel() * 2 * 3 / dt / 1e9))


def bench_gemm(results):
    for (m, n, k) in [(4096, 4096, 4096), (8192, 8192, 8192),
                      (8192, 3072, 768)]:
        a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
        dt = timeit(lambda: ops.gemm_bf16(a, b), iters=20)
        results.append(dict(op="gemm_bf16(ours)",
                            shape="{}x{}x{}".format(m, n, k),
                            us=dt * 1e6, tflops=2.0 * m * n * k / dt / 1e12))
        dtl = timeit(lambda: a @ b.T, iters=20)
        results.append(dict(op="gemm_hipblaslt",
                            shape="{}x{}x{}".format(m, n, k),
                            us=dtl * 1e6, tflops=2.0 * m * n * k / dtl / 1e12))


def bench_sampling(results):
    logits = torch.randn(64, 128256, device="cuda", dtype=torch.bfloat16)
    dt = timeit(lambda: ops.sample_top_k_top_p(logits, temperature=0.8))
    results.append(dict(op="sample_gumbel", shape="64x128256", us=dt * 1e6,
                        gbps=logits.numel() * 2 / dt / 1e9))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", type=str, default=None)
    args = ap.parse_args()
    assert torch.cuda.is_available() and ops.has_extension()
    results = []
    bench_attention(results)
    bench_decode(results)
    bench_memops(results)
    bench_gemm(results)
    bench_sampling(results)
    for r in results:
        perf = ("{:8.1f} TF".format(r["tflops"]) if "tflops" in r
                else "{:8.1f} GB/s".format(r["gbps"]))
        print("{:28s} {:22s} {:9.1f} us {}".format(
            r["op"], r["shape"], r["us"], perf))
    if args.out:
        with open(args.out, "wt") as f:
            json.dump(results, f, indent=1)


if __name__ == "__main__":
    main()
