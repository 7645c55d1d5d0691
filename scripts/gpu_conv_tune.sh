#!/bin/bash
# Does MIOpen find-mode / torch benchmark mode speed up ResNet-50 bf16 NHWC?
set -x
mkdir -p gpurun_out
cat > /tmp/convbench.py <<'EOF'
import sys, time, torch
sys.path.insert(0, ".")
from clearml_serving_amd.models import build_model

bench_flag = sys.argv[1] == "1"
torch.backends.cudnn.benchmark = bench_flag
m = build_model({"arch": "resnet50", "num_classes": 1000,
                 "dtype": "bfloat16"}, device="cuda")
x = torch.randn(64, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
with torch.inference_mode():
    for _ in range(6):
        m(x)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(30):
        m(x)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 30
print("benchmark=%s: resnet50 b64 fwd %.2f ms  (%.0f img/s)" % (
    bench_flag, dt * 1000, 64 / dt))
EOF
timeout 300 python /tmp/convbench.py 0 > gpurun_out/conv_default.txt 2>&1
timeout 600 python /tmp/convbench.py 1 > gpurun_out/conv_benchmark.txt 2>&1
MIOPEN_FIND_MODE=1 timeout 900 python /tmp/convbench.py 0 > gpurun_out/conv_find1.txt 2>&1
tail -1 gpurun_out/conv_default.txt gpurun_out/conv_benchmark.txt gpurun_out/conv_find1.txt
echo ALLDONE
