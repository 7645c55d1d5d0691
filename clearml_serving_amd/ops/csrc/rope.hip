// Rotary position embedding, in place, NeoX pairing (i, i + D/2).
// LLM-engine hot op (the reference delegates rope to vLLM internals,
// preprocess_service.py:619-1095).
// q: [T, H, D], k: [T, Hkv, D], positions: int32 [T].
#include "common.h"

namespace {

template <typename T>
__global__ void rope_kernel(T* __restrict__ q, T* __restrict__ k,
                            const int* __restrict__ positions, int tokens,
                            int hq, int hkv, int d, long q_tstride,
                            long k_tstride, float theta) {
  const int half = d / 2;
  const int t = blockIdx.x;
  if (t >= tokens) return;
  const float pos = (float)positions[t];
  const int total_heads = hq + hkv;
  for (int hi = blockIdx.y; hi < total_heads; hi += gridDim.y) {
    T* base = hi < hq ? q + (long)t * q_tstride + (long)hi * d
                      : k + (long)t * k_tstride + (long)(hi - hq) * d;
    for (int i = threadIdx.x; i < half; i += blockDim.x) {
      const float freq = __powf(theta, -(float)i / (float)half);
      const float angle = pos * freq;
      // precise sincos: fast-math __sincosf loses range reduction past a few
      // pi and breaks large-position rotations (measured: 30% mismatch)
      float c, s;
      sincosf(angle, &s, &c);
      const float lo = to_f32(base[i]);
      const float hi_v = to_f32(base[i + half]);
      base[i] = from_f32<T>(lo * c - hi_v * s);
      base[i + half] = from_f32<T>(hi_v * c + lo * s);
    }
  }
}

}  // namespace

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  double theta) {
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [T, H, D]");
  // token-strided views of a merged QKV projection are accepted (head and
  // element dims must be dense: stride(1) == D, stride(2) == 1)
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q heads must be dense");
  TORCH_CHECK(k.stride(2) == 1 && k.stride(1) == k.size(2),
              "k heads must be dense");
  const int tokens = q.size(0), hq = q.size(1), d = q.size(2);
  const int hkv = k.size(1);
  TORCH_CHECK(k.size(0) == tokens && k.size(2) == d);
  const long qs = q.stride(0), ks = k.stride(0);
  auto pos = positions.to(q.device(), at::kInt).contiguous();
  TORCH_CHECK(pos.numel() == tokens);
  dim3 grid(tokens, std::min(hq + hkv, 16));
  dim3 block(std::min(d / 2, 256));
  hipStream_t stream_ = cmls::current_stream();
  const auto st = q.scalar_type();
  if (st == at::kBFloat16) {
    hipLaunchKernelGGL(rope_kernel<__hip_bfloat16>, grid, block, 0,
                       stream_, (__hip_bfloat16*)q.data_ptr(),
                       (__hip_bfloat16*)k.data_ptr(), pos.data_ptr<int>(),
                       tokens, hq, hkv, d, qs, ks, (float)theta);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(rope_kernel<__half>, grid, block, 0, stream_,
                       (__half*)q.data_ptr(), (__half*)k.data_ptr(),
                       pos.data_ptr<int>(), tokens, hq, hkv, d, qs, ks, (float)theta);
  } else if (st == at::kFloat) {
    hipLaunchKernelGGL(rope_kernel<float>, grid, block, 0, stream_,
                       (float*)q.data_ptr(), (float*)k.data_ptr(),
                       pos.data_ptr<int>(), tokens, hq, hkv, d, qs, ks, (float)theta);
  } else {
    TORCH_CHECK(false, "rope: unsupported dtype ", st);
  }
}
