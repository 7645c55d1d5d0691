#!/bin/bash
# e2e long-context serving A/B: bf16 KV vs fp8 KV (both fp8 weights).
# 64 seqs x (1024 prompt + 192 out) -- decode reads ~1.1-1.2k keys/step.
set -x
mkdir -p gpurun_out
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
cat > /tmp/fp8kv_e2e.py <<'PY'
import asyncio, sys, time
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

kv = sys.argv[1]
cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=2048,
                      max_num_seqs=64, gpu_memory_fraction=0.7,
                      quantization="fp8", kv_dtype=kv)
eng = LlmEngine(cfg); eng.start()

async def one(i, rounds):
    prompt = [(i * 13 + j) % 100000 for j in range(1024)]
    seq = await eng.add_request(prompt, SamplingParams(
        temperature=0.8, max_tokens=192, ignore_eos=True))
    while True:
        item = await seq.stream.get()
        if item["finished"]:
            return

async def batch(n, tag):
    t0 = time.time()
    await asyncio.gather(*[one(i, 0) for i in range(n)])
    dt = time.time() - t0
    toks = n * 192
    print("%s kv=%s: %d seqs (1024 in / 192 out) in %.1fs -> %.0f out-tok/s"
          % (tag, kv, n, dt, toks / dt), flush=True)

loop = asyncio.new_event_loop()
loop.run_until_complete(batch(64, "warmup"))
loop.run_until_complete(batch(64, "run1"))
loop.run_until_complete(batch(64, "run2"))
eng.stop()
PY
timeout 260 python /tmp/fp8kv_e2e.py bfloat16 2>&1 | grep -E "run|warmup|tok/s" | tee gpurun_out/fp8kv_e2e_bf16.log
timeout 260 python /tmp/fp8kv_e2e.py fp8 2>&1 | grep -E "run|warmup|tok/s" | tee gpurun_out/fp8kv_e2e_fp8.log
echo E2EDONE
