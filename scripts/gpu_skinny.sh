#!/bin/bash
set -x
mkdir -p gpurun_out
timeout 200 python -m pytest tests/test_ops_gpu.py -q -k skinny 2>&1 | tail -2 | tee gpurun_out/skinny_test.txt
cat > /tmp/skinny_bench.py <<'PY'
import sys, time, torch
sys.path.insert(0, ".")
import clearml_serving_amd.ops as ops
dev = "cuda:0"
shapes = [("qkv", 4096, 6144), ("o_proj", 4096, 4096),
          ("gate_up", 4096, 28672), ("down", 14336, 4096)]
for M in (1, 16, 64, 128):
    tot = {"lt": 0.0, "v1": 0.0, "v2": 0.0, "sk": 0.0}; tb = 0
    for name, K, N in shapes:
        x = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
        w = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
        ext = ops._require_ext("skinny_gemm")
        fns = {"lt": lambda: torch.nn.functional.linear(x, w),
               "v1": lambda: ext.skinny_gemm(x, w, 1),
               "v2": lambda: ext.skinny_gemm(x, w, 2),
               "sk": lambda: ops.skinny_linear(x, w)}
        gb = (K * N * 2 + M * (K + N) * 2) / 1e9
        tb += gb
        res = {}
        for key, f in fns.items():
            for _ in range(10): f()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(200): f()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 200
            res[key] = dt; tot[key] += dt
        print(f"M={M:<4} {name:<8} lt {res['lt']*1e6:6.1f}us {gb/res['lt']:5.0f}GB/s | v1 {res['v1']*1e6:6.1f}us {gb/res['v1']:5.0f}GB/s | v2 {res['v2']*1e6:6.1f}us {gb/res['v2']:5.0f}GB/s | auto {res['sk']*1e6:6.1f}us  best={min(res, key=res.get)}")
    print(f"M={M:<4} ALL: lt {tot['lt']*1e6:6.1f}us | v1 {tot['v1']*1e6:6.1f}us | v2 {tot['v2']*1e6:6.1f}us | auto {tot['sk']*1e6:6.1f}us ({tb/tot['sk']:4.0f} GB/s) {tot['lt']/tot['sk']:5.2f}x")
PY
timeout 300 python /tmp/skinny_bench.py 2>&1 | tee gpurun_out/skinny_bench.txt
echo SKDONE
