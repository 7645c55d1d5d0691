#!/bin/bash
# Collect rocprof evidence for profiles/ + re-validate the full GPU suite.
set -x
mkdir -p gpurun_out
timeout 400 python -m pytest tests/ -q -m gpu 2>&1 | tail -3 > gpurun_out/t6.log
python __graft_entry__.py smoke >> gpurun_out/t6.log 2>&1

cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"

# 1) serving bench under kernel trace -> per-kernel stats
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof_bench -o bench \
    -- python bench.py --steps 3 --warmup 2 > gpurun_out/prof_bench.log 2>&1

# 2) llama decode loop under kernel trace
cat > /tmp/llm_loop.py <<'EOF'
import asyncio, sys
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams
cfg = LlmEngineConfig(preset="llama-3-1b", max_model_len=2048, max_num_seqs=32,
                      gpu_memory_fraction=0.4)
eng = LlmEngine(cfg); eng.start()
async def one(i):
    params = SamplingParams(temperature=0.8, max_tokens=32, ignore_eos=True)
    seq = await eng.add_request([(i*31+j) % 100000 for j in range(256)], params)
    while True:
        item = await seq.stream.get()
        if item["finished"]: return
async def main():
    await asyncio.gather(*[one(i) for i in range(16)])
asyncio.new_event_loop().run_until_complete(main())
EOF
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv \
    -d gpurun_out/prof_llm -o llm \
    -- python /tmp/llm_loop.py > gpurun_out/prof_llm.log 2>&1

# 3) fresh serving bench number (no profiler)
timeout 300 python bench.py --steps 8 --warmup 3 > gpurun_out/bench4.json 2>/dev/null

cat gpurun_out/t6.log
cat gpurun_out/bench4.json
ls gpurun_out/prof_bench gpurun_out/prof_llm
echo ALLDONE
