#!/bin/bash
# Round-final validation: full GPU suite, smoke, serving + kernel + llm
# benchmarks, memory-stability soak. Outputs land in gpurun_out/ for
# profiles/ refresh.
set -x
mkdir -p gpurun_out

timeout 500 python -m pytest tests/ -q -m gpu 2>&1 | tail -2 > gpurun_out/final_tests.log
python __graft_entry__.py smoke >> gpurun_out/final_tests.log 2>&1

for i in 1 2 3; do
  timeout 400 python bench.py --steps 8 --warmup 4 >> gpurun_out/final_bench.json 2>/dev/null
done
timeout 400 python benchmarks/kernel_bench.py > gpurun_out/final_kbench.txt 2>&1
timeout 300 python benchmarks/ensemble_bench.py --steps 4 --warmup 2 > gpurun_out/final_ens.json 2>/dev/null
timeout 500 python benchmarks/http_load.py -n 8000 -c 128 > gpurun_out/final_http.json 2>/dev/null

# llama-8B throughput + soak: repeated generation rounds, HBM must plateau
cat > /tmp/llm_soak.py <<'EOF'
import asyncio, time, sys, torch
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams
cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                      max_num_seqs=128, gpu_memory_fraction=0.7)
eng = LlmEngine(cfg); eng.start()
async def one(i, n_in, n_out):
    seq = await eng.add_request([(i*13+j) % 100000 for j in range(n_in)],
                                SamplingParams(temperature=0.8, max_tokens=n_out, ignore_eos=True))
    while True:
        item = await seq.stream.get()
        if item["finished"]: return
async def round_(n, n_in, n_out):
    t0 = time.time(); await asyncio.gather(*[one(i, n_in, n_out) for i in range(n)])
    return time.time() - t0
loop = asyncio.new_event_loop()
mem = []
for r in range(6):
    dt = loop.run_until_complete(round_(64, 256, 64))
    mem.append(torch.cuda.memory_allocated() // (1 << 20))
    print("round %d: %.2fs out_tok/s=%.0f mem=%dMiB free_blocks=%d" % (
        r, dt, 64*64/dt, mem[-1], eng.allocator.available))
assert mem[-1] <= mem[1] + 64, "HBM growth across rounds: %s" % mem
assert eng.allocator.available == eng.allocator.num_blocks
print("SOAK-OK")
EOF
timeout 600 python /tmp/llm_soak.py > gpurun_out/final_soak.txt 2>&1

cat gpurun_out/final_tests.log
grep value gpurun_out/final_bench.json
tail -3 gpurun_out/final_soak.txt
cat gpurun_out/final_ens.json gpurun_out/final_http.json
echo ALLDONE
