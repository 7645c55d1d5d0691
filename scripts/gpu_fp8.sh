#!/bin/bash
# fp8 decode path A/B: skinny fp8 kernel at llama-8B projection shapes,
# then end-to-end llama-8B out-tok/s bf16 vs fp8 (fused-quant path).
set -x
mkdir -p gpurun_out

cat > /tmp/fp8_kernels.py <<'PY'
import sys, time, torch
sys.path.insert(0, ".")
from clearml_serving_amd import ops

def t(fn, iters=200):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

SHAPES = [("qkv", 4096, 6144), ("o_proj", 4096, 4096),
          ("gate_up", 4096, 28672), ("down", 14336, 4096)]
for M in (1, 16, 32, 64):
    tot8 = totb = totlt = 0.0
    for name, K, N in SHAPES:
        a = (torch.randn(M, K, device="cuda") / 8).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") / 8).to(torch.bfloat16)
        a8, as_ = ops.quant_fp8(a); w8, ws = ops.quant_fp8(w)
        wsw = ops.swizzle_fp8_weight(w8)
        ext = ops._require_ext("skinny_gemm_fp8")
        us8 = t(lambda: ext.skinny_gemm_fp8(a8, as_, wsw, ws))
        usv2 = t(lambda: ext.skinny_gemm_fp8_v2(a8, as_, w8, ws))
        usb = t(lambda: ops.skinny_linear(a, w))
        uslt = t(lambda: torch.nn.functional.linear(a, w))
        # fp8 scaled_mm (hipBLASLt) comparison at the same precision
        pad = (-M) % 16
        a8p = torch.nn.functional.pad(a8, (0, 0, 0, pad)) if pad else a8
        asp = torch.nn.functional.pad(as_.reshape(-1,1), (0, 0, 0, pad), value=1.0) if pad else as_.reshape(-1,1)
        f8 = torch.float8_e4m3fn
        uslt8 = t(lambda: torch._scaled_mm(a8p.view(f8), w8.view(f8).t(),
                  scale_a=asp.contiguous(), scale_b=ws.reshape(1,-1),
                  out_dtype=torch.bfloat16))
        gb8 = (N*K + M*K) / us8 / 1e3   # fp8 bytes
        gbv2 = (N*K + M*K) / usv2 / 1e3
        tot8 += min(us8, usv2); totb += usb; totlt += uslt
        print(f"M={M:<3} {name:8} fp8v1 {us8:6.1f}us {gb8:5.0f}GB/s | "
              f"fp8v2 {usv2:6.1f}us {gbv2:5.0f}GB/s | lt8 {uslt8:6.1f}us | "
              f"bf16-auto {usb:6.1f}us | lt {uslt:6.1f}us", flush=True)
    print(f"M={M:<3} ALL: fp8-best {tot8:6.1f}us | bf16-auto {totb:6.1f}us | "
          f"lt {totlt:6.1f}us  speedup vs lt {totlt/tot8:.2f}x", flush=True)
PY
timeout 300 python /tmp/fp8_kernels.py 2>&1 | grep -v Warn | tee gpurun_out/fp8_kernels.txt

cat > /tmp/fp8_e2e.py <<'PY'
import asyncio, sys, time, gc, torch
sys.path.insert(0, ".")
from clearml_serving_amd.engines.llm.engine import LlmEngine, LlmEngineConfig, SamplingParams

def bench(quant, n_seqs, n_in, n_out, graphs=True):
    cfg = LlmEngineConfig(preset="llama-3-8b", max_model_len=4096,
                          max_num_seqs=128, gpu_memory_fraction=0.7,
                          quantization=quant, decode_graphs=graphs)
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
    loop.run_until_complete(round_())  # warm + graph capture
    ts = [loop.run_until_complete(round_()) for _ in range(3)]
    dt = min(ts)
    print("quant=%-5s graphs=%s n_seqs=%-3d n_out=%d: %.2fs out_tok/s=%.0f" % (
        quant, graphs, n_seqs, n_out, dt, n_seqs*n_out/dt), flush=True)
    eng.stop(); gc.collect(); torch.cuda.empty_cache()

bench(None, 64, 32, 256)
bench("fp8", 64, 32, 256)
bench(None, 8, 32, 256)
bench("fp8", 8, 32, 256)
bench(None, 1, 32, 128)
bench("fp8", 1, 32, 128)
PY
timeout 900 python /tmp/fp8_e2e.py 2>&1 | grep -v "Task was destroyed\|Task pending\|Warn" | tee gpurun_out/fp8_e2e.txt
echo FP8DONE
