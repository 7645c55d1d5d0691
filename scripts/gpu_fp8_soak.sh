#!/bin/bash
# fp8 serving soak (>=60 s steady state) + kernel-trace proof that the fused
# fp8 kernels carry the decode hot path.
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"

cat > /tmp/fp8_soak.py <<'PY'
import asyncio, sys, time
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                      max_num_seqs=128, gpu_memory_fraction=0.7,
                      quantization="fp8")
eng = LlmEngine(cfg); eng.start()

async def one(i, n_out):
    seq = await eng.add_request([(i*13+j) % 100000 for j in range(32)],
                                SamplingParams(temperature=0.8, max_tokens=n_out, ignore_eos=True))
    while True:
        item = await seq.stream.get()
        if item["finished"]: return

async def soak(seconds):
    loop_t0 = time.time()
    total = 0
    rounds = 0
    while time.time() - loop_t0 < seconds:
        await asyncio.gather(*[one(i + rounds * 64, 256) for i in range(64)])
        total += 64 * 256
        rounds += 1
    dt = time.time() - loop_t0
    print("fp8 soak: %.1fs, %d rounds, %d tokens, out_tok/s=%.0f" % (
        dt, rounds, total, total / dt), flush=True)

loop = asyncio.new_event_loop()
loop.run_until_complete(soak(10))   # warmup + graph capture
loop.run_until_complete(soak(65))   # the soak
print("stats:", {k: v for k, v in eng.stats.items()}, flush=True)
eng.stop()
PY

# kernel share of a short fp8 decode run (stats only; small CSV kept)
cat > /tmp/fp8_short.py <<'PY'
import asyncio, sys
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams
cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096, max_num_seqs=64,
                      gpu_memory_fraction=0.6, quantization="fp8",
                      decode_graphs=False)  # trace shows per-kernel names
eng = LlmEngine(cfg); eng.start()
async def one(i):
    seq = await eng.add_request([(i*13+j) % 100000 for j in range(32)],
                                SamplingParams(temperature=0.8, max_tokens=64, ignore_eos=True))
    while True:
        item = await seq.stream.get()
        if item["finished"]: return
async def main():
    await asyncio.gather(*[one(i) for i in range(32)])
loop = asyncio.new_event_loop()
loop.run_until_complete(main())
eng.stop()
PY
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/fp8prof -o fp8dec -- python /tmp/fp8_short.py \
    > gpurun_out/fp8prof.log 2>&1
python - <<'PY' 2>&1 | tee gpurun_out/fp8_kernel_share.txt
import csv, glob
f = glob.glob("gpurun_out/fp8prof/**/*kernel_stats.csv", recursive=True) + \
    glob.glob("gpurun_out/fp8prof/*kernel_stats.csv")
rows = list(csv.DictReader(open(f[-1])))
rows.sort(key=lambda r: -float(r["TotalDurationNs"]))
tot = sum(float(r["TotalDurationNs"]) for r in rows)
for r in rows[:14]:
    print("%6.2f%%  %8.1fms  %6d calls  %s" % (
        100*float(r["TotalDurationNs"])/tot, float(r["TotalDurationNs"])/1e6,
        int(r["Calls"]), r["Name"][:80]))
PY
rm -rf gpurun_out/fp8prof
echo FP8SOAKDONE
