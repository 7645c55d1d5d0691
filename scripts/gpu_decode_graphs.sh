#!/bin/bash
# Decode hipGraph A/B: llama-3-8B throughput with and without per-bucket
# decode graph capture, plus the numerics gate.
set -x
mkdir -p gpurun_out
timeout 300 python -m pytest tests/test_llm_gpu.py::test_decode_graphs_match_eager -x -q 2>&1 | tail -2 | tee gpurun_out/graphs_test.txt
cat > /tmp/llm_graph_ab.py <<'PY'
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
    loop.run_until_complete(round_())  # warm (captures graphs)
    ts = [loop.run_until_complete(round_()) for _ in range(3)]
    dt = min(ts)
    print("graphs=%s n_seqs=%d: %.2fs out_tok/s=%.0f" % (graphs, n_seqs, dt, n_seqs*n_out/dt), flush=True)
    eng.stop()
    import torch, gc; gc.collect(); torch.cuda.empty_cache()

for g in (False, True):
    bench(g, 64, 256, 64)
for g in (False, True):
    bench(g, 16, 256, 64)
PY
timeout 900 python /tmp/llm_graph_ab.py 2>&1 | tee gpurun_out/graph_ab.txt
echo GRAPHDONE
