#!/bin/bash
set -x
mkdir -p gpurun_out/profnew
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
cat > /tmp/newk.py <<'PY'
import sys, torch
sys.path.insert(0, ".")
import clearml_serving_amd.ops as ops
dev = "cuda:0"
ext = ops._require_ext("skinny_gemm")
x16 = (torch.randn(16, 4096, device=dev) / 8).to(torch.bfloat16)
wq = (torch.randn(6144, 4096, device=dev) / 8).to(torch.bfloat16)
wo = (torch.randn(4096, 4096, device=dev) / 8).to(torch.bfloat16)
q = (torch.randn(8, 2048, 32, 128, device=dev) / 8).to(torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
for _ in range(50):
    ext.skinny_gemm(x16, wq, 1); ext.skinny_gemm(x16, wo, 1)
    ops.attention(q, k, v, causal=True, layout="bshd")
torch.cuda.synchronize()
PY
timeout 200 rocprofv3 --kernel-trace --stats -d gpurun_out/profnew -- python /tmp/newk.py > gpurun_out/profnew/stats.txt 2>&1
grep -E "skinny|attn|Kernel Name|--" gpurun_out/profnew/stats.txt | head -20
echo PROFDONE
