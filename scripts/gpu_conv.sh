#!/bin/bash
# conv3x3 A/B vs MIOpen at the ResNet-50 bottleneck shapes (batch 64) and
# end-to-end ResNet-50 forward with/without the in-tree conv routing.
set -x
mkdir -p gpurun_out
cat > /tmp/conv_ab.py <<'PY'
import os, sys, time, torch
sys.path.insert(0, ".")
from clearml_serving_amd import ops

torch.backends.cudnn.benchmark = True  # MIOpen find mode (production config)

def t(fn, iters=100):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

SHAPES = [("conv2 56x56x64",  64, 64, 64, 56),
          ("conv3 28x28x128", 64, 128, 128, 28),
          ("conv4 14x14x256", 64, 256, 256, 14)]
ext = ops._require_ext("conv3x3_nhwc")
for name, n, c, k, w in SHAPES:
    x = (torch.randn(n, c, w, w, device="cuda") / 4).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device="cuda") / 8).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    b = torch.randn(k, device="cuda").to(torch.bfloat16)
    us_ours = t(lambda: ext.conv3x3_nhwc(x, wt, b, True, None))
    us_mi = t(lambda: torch.relu(torch.nn.functional.conv2d(
        x, wt, b, stride=1, padding=1)))
    flops = 2.0 * n * w * w * k * 9 * c
    print(f"{name}: ours {us_ours:7.1f}us {flops/us_ours/1e6:6.0f}TF | "
          f"MIOpen+relu {us_mi:7.1f}us {flops/us_mi/1e6:6.0f}TF | "
          f"{us_mi/us_ours:.2f}x", flush=True)

# stem A/B: im2col+GEMM vs MIOpen (naive fallback at C=3 NHWC)
from clearml_serving_amd.models.resnet import StemConv
stem = StemConv(3, 64, 7, stride=2, padding=3).cuda().to(torch.bfloat16)
xs = torch.randn(64, 3, 224, 224, device="cuda", dtype=torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
with torch.inference_mode():
    us_stem = t(lambda: stem(xs), iters=50)
    us_mi_stem = t(lambda: torch.relu(torch.nn.functional.conv2d(
        xs, stem.conv.weight, stem.conv.bias, stride=2, padding=3)), iters=50)
print(f"stem 7x7s2 C=3 b64: im2col+GEMM {us_stem:7.1f}us | MIOpen {us_mi_stem:7.1f}us | {us_mi_stem/us_stem:.2f}x", flush=True)

# end-to-end ResNet-50 b64
from clearml_serving_amd.models import build_model
m = build_model({"arch": "resnet50", "num_classes": 1000,
                 "dtype": "bfloat16"}, device="cuda")
img = torch.randn(64, 3, 224, 224, device="cuda", dtype=torch.bfloat16)
with torch.inference_mode():
    for routed in ("1", "0"):
        os.environ["CMLS_CONV3"] = routed
        us = t(lambda: m(img), iters=50)
        print(f"resnet50 b64 e2e (conv3 routed={routed}): {us/1000:.3f}ms",
              flush=True)
PY
timeout 600 python /tmp/conv_ab.py 2>&1 | grep -v Warn | tee gpurun_out/conv_ab.txt
echo CONVDONE
