#!/bin/bash
# Verify decode-split + sampling-split: numerics + perf, plus LLM e2e bench.
set -x
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_ops_gpu.py tests/test_llm_gpu.py -q 2>&1 | tail -3 > gpurun_out/t5.log
timeout 300 python benchmarks/kernel_bench.py > gpurun_out/kbench4.txt 2>&1

# llama-3-8B single-GPU serving throughput probe (random weights)
cat > /tmp/llm_bench.py <<'EOF'
import asyncio, time, sys, torch
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                      max_num_seqs=64, gpu_memory_fraction=0.7)
eng = LlmEngine(cfg)
t0 = time.time(); eng.start(); print("engine start %.1fs, kv blocks=%d" % (time.time()-t0, eng.allocator.num_blocks))

async def one(i, n_in=512, n_out=64):
    params = SamplingParams(temperature=0.8, max_tokens=n_out, ignore_eos=True)
    ids = [(i*13+j) % 100000 for j in range(n_in)]
    seq = await eng.add_request(ids, params)
    while True:
        item = await seq.stream.get()
        if item["finished"]:
            return

async def bench(n_seqs):
    t0 = time.time()
    await asyncio.gather(*[one(i) for i in range(n_seqs)])
    dt = time.time() - t0
    total_out = n_seqs * 64
    total_in = n_seqs * 512
    print("seqs=%d: %.2fs  decode+prefill tok/s=%.0f  out tok/s=%.0f" % (
        n_seqs, dt, (total_in+total_out)/dt, total_out/dt))
    print("engine stats:", eng.stats)

loop = asyncio.new_event_loop()
loop.run_until_complete(bench(4))   # warm
loop.run_until_complete(bench(32))
EOF
timeout 600 python /tmp/llm_bench.py > gpurun_out/llm_bench.txt 2>&1
cat gpurun_out/t5.log
grep -E "decode|sample" gpurun_out/kbench4.txt
cat gpurun_out/llm_bench.txt | tail -6
echo ALLDONE
