#!/bin/bash
# Decode-dominated A/B: short prompts, long generations.
set -x
mkdir -p gpurun_out
cat > /tmp/llm_graph_ab2.py <<'PY'
import asyncio, time, sys
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

def bench(graphs, n_seqs, n_in, n_out):
    cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                          max_num_seqs=128, gpu_memory_fraction=0.7,
                          decode_graphs=graphs)
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
    loop.run_until_complete(round_())  # warm + capture
    ts = [loop.run_until_complete(round_()) for _ in range(3)]
    dt = min(ts)
    d = eng.stats["decode_batches"]
    print("graphs=%s n_seqs=%-3d n_out=%d: %.2fs out_tok/s=%.0f decode_batches=%d" % (
        graphs, n_seqs, n_out, dt, n_seqs*n_out/dt, d), flush=True)
    eng.stop()
    import torch, gc; gc.collect(); torch.cuda.empty_cache()

for g in (False, True):
    bench(g, 64, 32, 256)
for g in (False, True):
    bench(g, 8, 32, 256)
for g in (False, True):
    bench(g, 1, 32, 128)
PY
timeout 1100 python /tmp/llm_graph_ab2.py 2>&1 | grep -v "Task was destroyed\|Task pending" | tee gpurun_out/graph_ab2.txt
echo GRAPHDONE2
