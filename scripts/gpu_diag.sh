#!/bin/bash
# GPU diagnostics: rope localization + attention scaling + PMC counters.
set -x
mkdir -p gpurun_out
timeout 200 python benchmarks/rope_debug.py > gpurun_out/rope_dbg.txt 2>&1
timeout 300 python benchmarks/attn_diag.py > gpurun_out/attn_diag2.txt 2>&1

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
    -d gpurun_out/attnpmc -o attn --output-format csv -- python /tmp/attn_once.py \
    > gpurun_out/pmc_run.log 2>&1 || true
ls gpurun_out/attnpmc >> gpurun_out/pmc_run.log 2>&1
echo ALLDONE
