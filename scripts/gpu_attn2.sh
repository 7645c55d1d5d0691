#!/bin/bash
# Verify attention fixes: numerics, scaling, PMC conflicts, kernel bench.
set -x
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py -q 2>&1 | tail -3 > gpurun_out/t4.log
timeout 300 python benchmarks/attn_diag.py > gpurun_out/attn_diag3.txt 2>&1
timeout 300 python benchmarks/kernel_bench.py > gpurun_out/kbench3.txt 2>&1

cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
cat > /tmp/attn_once.py <<'EOF'
import torch, sys
sys.path.insert(0, ".")
import clearml_serving_amd.ops as ops
q = torch.randn(8, 32, 2048, 128, device="cuda", dtype=torch.bfloat16)
k = torch.randn_like(q); v = torch.randn_like(q)
for _ in range(3):
    ops.attention(q, k, v)
torch.cuda.synchronize()
EOF
timeout 300 rocprofv3 --pmc SQ_INSTS_MFMA SQ_LDS_BANK_CONFLICT SQ_WAVE_CYCLES SQ_BUSY_CYCLES \
    -d gpurun_out/attnpmc2 -o attn2 --output-format csv -- python /tmp/attn_once.py \
    > gpurun_out/pmc_run2.log 2>&1 || true
cat gpurun_out/t4.log gpurun_out/attn_diag3.txt
echo ALLDONE
