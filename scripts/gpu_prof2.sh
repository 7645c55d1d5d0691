#!/bin/bash
# Round-2 rocprof evidence: serving bench kernel trace (conv3x3 + fused
# kernels visible) + GPU busy fraction; fp8 llama decode trace.
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"

timeout 420 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof_bench2 -o bench2 \
    -- python bench.py --steps 20 --warmup 10 > gpurun_out/prof_bench2.log 2>&1

# GPU busy fraction + top kernels from the trace
python - <<'PY' 2>&1 | tee gpurun_out/busy2.txt
import csv, glob, collections
f = sorted(glob.glob("gpurun_out/prof_bench2/**/*kernel_trace.csv",
                     recursive=True) +
           glob.glob("gpurun_out/prof_bench2/*kernel_trace.csv"))
print("trace files:", f)
if f:
    rows = list(csv.DictReader(open(f[-1])))
    beg = min(int(r["Start_Timestamp"]) for r in rows)
    end = max(int(r["End_Timestamp"]) for r in rows)
    # merge intervals for true busy time
    iv = sorted((int(r["Start_Timestamp"]), int(r["End_Timestamp"]))
                for r in rows)
    busy, ce = 0, 0
    for s, e in iv:
        if s > ce:
            busy += e - s
            ce = e
        elif e > ce:
            busy += e - ce
            ce = e
    print("kernels:", len(rows))
    print("wall %.3fs busy %.3fs -> GPU busy %.1f%%" % (
        (end-beg)/1e9, busy/1e9, 100.0*busy/(end-beg)))
    agg = collections.Counter()
    for r in rows:
        agg[r["Kernel_Name"][:70]] += int(r["End_Timestamp"]) - int(r["Start_Timestamp"])
    tot = sum(agg.values())
    for name, ns in agg.most_common(15):
        print("%6.2f%% %9.2fms  %s" % (100.0*ns/tot, ns/1e6, name))
PY
ls gpurun_out/prof_bench2 2>/dev/null | head -5
echo PROF2DONE
