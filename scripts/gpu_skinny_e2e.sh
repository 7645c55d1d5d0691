#!/bin/bash
set -x
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_llm_gpu.py -q 2>&1 | tail -2 | tee gpurun_out/skinny_llm_tests.txt
cat > /tmp/llm_skinny_ab.py <<'PY'
import asyncio, os, time, sys
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

def bench(n_seqs, n_in, n_out):
    cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                          max_num_seqs=128, gpu_memory_fraction=0.7)
    eng = LlmEngine(cfg); eng.start()
    async def one(i):
        seq = await eng.add_request([(i*13+j) % 100000 for j in range(n_in)],
                                    SamplingParams(temperature=0.8, max_tokens=n_out, ignore_eos=True))
        while True:
            item = await seq.stream.get()
            if item["finished"]: return
    async def round_():
        t0 = time.time(); await asyncio.gather(*[one(i) for i in range(n_seqs)])
        return time.time() - t0
    loop = asyncio.new_event_loop()
    loop.run_until_complete(round_())
    ts = [loop.run_until_complete(round_()) for _ in range(3)]
    dt = min(ts)
    print("skinny=%s n_seqs=%-3d: %.3fs out_tok/s=%.0f" % (
        os.environ.get("CMLS_SKINNY", "1"), n_seqs, dt, n_seqs*n_out/dt), flush=True)
    eng.stop()
    import torch, gc; gc.collect(); torch.cuda.empty_cache()

for n in (1, 8, 16):
    bench(n, 32, 192)
PY
CMLS_SKINNY=0 timeout 500 python /tmp/llm_skinny_ab.py 2>&1 | grep -v "Task was" | tee gpurun_out/skinny_e2e_off.txt
CMLS_SKINNY=1 timeout 500 python /tmp/llm_skinny_ab.py 2>&1 | grep -v "Task was" | tee gpurun_out/skinny_e2e_on.txt
echo E2EDONE
