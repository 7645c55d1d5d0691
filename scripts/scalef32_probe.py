"""Ground truth for v_cvt_scalef32_pk_f32_fp8 scale semantics on gfx950:
full f32 multiply, or exponent-only (MX)? Compares against cvt_pk * scale."""
import sys

import torch
from torch.utils.cpp_extension import load_inline

SRC = r"""
#include <hip/hip_runtime.h>
#include <torch/extension.h>
typedef __attribute__((ext_vector_type(2))) float f2;
__global__ void probe(const unsigned char* b, float sc, float* out) {
  int w = b[0] | (b[1] << 8);
  f2 scaled = __builtin_amdgcn_cvt_scalef32_pk_f32_fp8(w, sc, false);
  f2 plain = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  out[0] = scaled[0]; out[1] = scaled[1];
  out[2] = plain[0] * sc; out[3] = plain[1] * sc;
}
torch::Tensor run(torch::Tensor bytes, double sc) {
  auto out = torch::zeros({4}, bytes.options().dtype(at::kFloat));
  hipLaunchKernelGGL(probe, dim3(1), dim3(1), 0, 0,
                     bytes.data_ptr<unsigned char>(), (float)sc,
                     out.data_ptr<float>());
  return out;
}
"""
CPP = "torch::Tensor run(torch::Tensor bytes, double sc);"
mod = load_inline(name="scalef32_probe", cpp_sources=CPP, cuda_sources=SRC,
                  functions=["run"], with_cuda=True, verbose=False)
b = torch.tensor([0x3C, 0x44], dtype=torch.uint8, device="cuda")  # e4m3 vals
for sc in (1.0, 2.0, 3.0, 0.7, 448.0):
    r = mod.run(b, sc).cpu()
    print("scale=%6.2f  scalef32=(%g, %g)  plain*sc=(%g, %g)  %s"
          % (sc, r[0], r[1], r[2], r[3],
             "MATCH" if torch.allclose(r[:2], r[2:]) else "DIFFER"))
