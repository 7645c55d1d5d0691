#!/usr/bin/env python3
"""Diagnose attention-prefill scaling: causal tile-skip effectiveness."""

import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import clearml_serving_amd.ops as ops  # noqa: E402


def t(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    b, h, d = 8, 32, 128
    for s in (512, 1024, 2048):
        q = torch.randn(b, h, s, d, device="cuda", dtype=torch.bfloat16)
        k, v = torch.randn_like(q), torch.randn_like(q)
        full = t(lambda: ops.attention(q, k, v))
        causal = t(lambda: ops.attention(q, k, v, causal=True))
        # half-key control: same q length, kv length s/2 (separate tensors)
        k2 = k[:, :, :s // 2].contiguous()
        v2 = v[:, :, :s // 2].contiguous()
        half = t(lambda: ops.attention(q, k2, v2))
        # seq_lens control: mask to half length
        sl = torch.full((b,), s // 2, dtype=torch.int32, device="cuda")
        masked = t(lambda: ops.attention(q, k, v, seq_lens=sl))
        # per-block fixed-cost control: 64-key kv (1 tile per block)
        k1 = k[:, :, :64].contiguous()
        v1 = v[:, :, :64].contiguous()
        kv64 = t(lambda: ops.attention(q, k1, v1))
        print("s={:5d}: full {:8.1f}us  causal {:8.1f}us ({:.2f}x)  "
              "half-kv {:8.1f}us ({:.2f}x)  seqlen-half {:8.1f}us ({:.2f}x)  "
              "kv64 {:8.1f}us ({:.3f}x)"
              .format(s, full, causal, causal / full, half, half / full,
                      masked, masked / full, kv64, kv64 / full))


if __name__ == "__main__":
    main()
