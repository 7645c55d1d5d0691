#!/bin/bash
# PMC counters for the round-2 kernels (conv3x3, attention v2, skinny fp8):
# MFMA/VALU/LDS instruction mix + bank conflicts. Counters-only run
# (rocprofv3 --pmc must not be combined with trace domains).
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"

cat > /tmp/pmc_kernels.py <<'PY'
import sys, torch
sys.path.insert(0, ".")
from clearml_serving_amd import ops
ext = ops.extension_or_none()

# conv3 shape
x = (torch.randn(64, 128, 28, 28, device="cuda") / 4).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
wt = (torch.randn(128, 128, 3, 3, device="cuda") / 8).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
for _ in range(20):
    ext.conv3x3_nhwc(x, wt, None, True, None)

# attention v2, llama-8b s2048 causal
q = (torch.randn(4, 32, 2048, 128, device="cuda") / 4).to(torch.bfloat16)
k = (torch.randn(4, 8, 2048, 128, device="cuda") / 4).to(torch.bfloat16)
v = (torch.randn(4, 8, 2048, 128, device="cuda") / 4).to(torch.bfloat16)
for _ in range(10):
    ext.attention_prefill_v2(q, k, v, True, 128 ** -0.5, None, False)

# skinny fp8, qkv M=1
a = (torch.randn(1, 4096, device="cuda") / 8).to(torch.bfloat16)
w = (torch.randn(6144, 4096, device="cuda") / 8).to(torch.bfloat16)
a8, as_ = ops.quant_fp8(a); w8, ws = ops.quant_fp8(w)
wsw = ops.swizzle_fp8_weight(w8)
for _ in range(50):
    ext.skinny_gemm_fp8(a8, as_, wsw, ws)
torch.cuda.synchronize()
PY
timeout 420 rocprofv3 --pmc SQ_INSTS_MFMA SQ_INSTS_VALU SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT \
    -d gpurun_out/pmc2 -o pmc2 --output-format csv -- python /tmp/pmc_kernels.py \
    > gpurun_out/pmc2.log 2>&1

python - <<'PY' 2>&1 | tee gpurun_out/pmc2_summary.txt
import csv, glob, collections
files = glob.glob("gpurun_out/pmc2/**/*counter_collection.csv", recursive=True) + \
        glob.glob("gpurun_out/pmc2/*counter_collection.csv")
print("files:", files)
agg = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.Counter()
for f in files:
    for r in csv.DictReader(open(f)):
        name = r["Kernel_Name"]
        key = None
        if "conv3x3" in name: key = "conv3x3"
        elif "attn_prefill_v2" in name: key = "attn_v2"
        elif "skinny_gemm_fp8" in name: key = "skinny_fp8"
        if key:
            agg[key][r["Counter_Name"]] += float(r["Counter_Value"])
            cnt[(key, r["Counter_Name"])] += 1
for k, d in agg.items():
    mfma = d.get("SQ_INSTS_MFMA", 0) or 1
    print("%s: MFMA %.0f  VALU %.2fx  LDS %.2fx  bank-conflict/LDS %.3f" % (
        k, mfma, d.get("SQ_INSTS_VALU", 0) / mfma,
        d.get("SQ_INSTS_LDS", 0) / mfma,
        d.get("SQ_LDS_BANK_CONFLICT", 0) / max(d.get("SQ_INSTS_LDS", 1), 1)))
PY
rm -rf gpurun_out/pmc2  # raw CSVs are large; the summary is the artifact
echo PMC2DONE
