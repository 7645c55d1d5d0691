#!/usr/bin/env python3
"""Localize the rope kernel mismatch: which positions/dims disagree."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import clearml_serving_amd.ops as ops  # noqa: E402


def manual(qf, pos, theta=10000.0):
    d = qf.shape[-1]
    half = d // 2
    inv = 1.0 / (theta ** (torch.arange(half, dtype=torch.float64) / half))
    ang = float(pos) * inv
    c, s = ang.cos().float(), ang.sin().float()
    lo, hi = qf[..., :half].clone(), qf[..., half:].clone()
    out = qf.clone()
    out[..., :half] = lo * c - hi * s
    out[..., half:] = hi * c + lo * s
    return out


def main():
    d = 128
    torch.manual_seed(0)
    base = torch.randn(1, 1, d)
    for pos in (0, 1, 7, 33, 100, 121, 163, 1000):
        q = base.clone().to("cuda", torch.bfloat16).contiguous()
        k = q.clone()
        p = torch.tensor([pos], dtype=torch.int32, device="cuda")
        ops.rope_inplace(q, k, p)
        ref = manual(base[0, 0].double(), pos).float()
        diff = (q[0, 0].float().cpu() - ref).abs()
        print("pos {:5d}: max diff {:9.5f} at dim {:3d}".format(
            pos, diff.max().item(), int(diff.argmax())))
        if diff.max() > 0.05:
            idx = int(diff.argmax())
            print("   kernel:", q[0, 0, max(0, idx - 2):idx + 3].float().cpu().tolist())
            print("   ref   :", ref[max(0, idx - 2):idx + 3].tolist())

    # full-tensor shaped like the failing test
    torch.manual_seed(5)
    t, h, hkv = 64, 8, 2
    q = torch.randn(t, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(t, hkv, d, device="cuda", dtype=torch.bfloat16)
    pos = torch.arange(100, 100 + t, device="cuda", dtype=torch.int32)
    qc = q.float().cpu().clone()
    ops.rope_inplace(q, k, pos)
    bad_tokens = []
    for tt in range(t):
        ref = manual(qc[tt].double(), 100 + tt).float()
        dm = (q[tt].float().cpu() - ref).abs().max().item()
        if dm > 0.05:
            bad_tokens.append((tt, round(dm, 3)))
    print("bad tokens ({} of {}):".format(len(bad_tokens), t), bad_tokens[:20])
    # per-head breakdown of first bad token
    if bad_tokens:
        tt = bad_tokens[0][0]
        ref = manual(qc[tt].double(), 100 + tt).float()
        per_head = (q[tt].float().cpu() - ref).abs().amax(dim=-1)
        print("token {} per-head max diff: {}".format(tt, per_head.tolist()))


if __name__ == "__main__":
    main()
