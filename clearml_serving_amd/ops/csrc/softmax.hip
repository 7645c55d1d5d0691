// Row softmax over the last dim (numerically stable, one workgroup per row).
// SURVEY.md §2.6 kernel-library row ('LayerNorm/Softmax').
// Used standalone (parity op); attention uses its own fused online softmax.
#include "common.h"

namespace {

template <typename T>
__global__ void softmax_kernel(const T* __restrict__ x, T* __restrict__ out,
                               int rows, int h) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* xrow = x + (long)row * h;
  T* orow = out + (long)row * h;

  constexpr int VALS = 32;
  float v[VALS];
  const int per_lane = (h + blockDim.x - 1) / blockDim.x;
  const bool in_regs = per_lane <= VALS;

  float m = -INFINITY;
  for (int i = 0; i < per_lane; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    float f = idx < h ? to_f32(xrow[idx]) : -INFINITY;
    if (in_regs && i < VALS) v[i] = f;
    m = fmaxf(m, f);
  }
  m = block_reduce(m, tmp, MaxOp{}, -INFINITY);
  __syncthreads();

  float sum = 0.f;
  for (int i = 0; i < per_lane; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= h) break;
    float f = (in_regs && i < VALS) ? v[i] : to_f32(xrow[idx]);
    f = __expf(f - m);
    if (in_regs && i < VALS) v[i] = f;
    sum += f;
  }
  sum = block_reduce(sum, tmp, SumOp{}, 0.f);
  const float inv = 1.0f / sum;

  for (int i = 0; i < per_lane; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= h) break;
    float f = (in_regs && i < VALS) ? v[i] : __expf(to_f32(xrow[idx]) - m);
    orow[idx] = from_f32<T>(f * inv);
  }
}

}  // namespace

torch::Tensor softmax_lastdim(torch::Tensor x) {
  CHECK_LASTDIM_CONTIG(x);
  const int h = x.size(-1);
  const int rows = x.numel() / h;
  auto out = torch::empty_like(x);
  const int block = std::min(1024, ((h + 63) / 64) * 64);
  hipStream_t stream_ = cmls::current_stream();
  const auto st = x.scalar_type();
  if (st == at::kBFloat16) {
    hipLaunchKernelGGL(softmax_kernel<__hip_bfloat16>, dim3(rows), dim3(block),
                       16 * sizeof(float), stream_,
                       (const __hip_bfloat16*)x.data_ptr(),
                       (__hip_bfloat16*)out.data_ptr(), rows, h);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(softmax_kernel<__half>, dim3(rows), dim3(block),
                       16 * sizeof(float), stream_,
                       (const __half*)x.data_ptr(), (__half*)out.data_ptr(),
                       rows, h);
  } else if (st == at::kFloat) {
    hipLaunchKernelGGL(softmax_kernel<float>, dim3(rows), dim3(block),
                       16 * sizeof(float), stream_,
                       (const float*)x.data_ptr(), (float*)out.data_ptr(),
                       rows, h);
  } else {
    TORCH_CHECK(false, "softmax: unsupported dtype ", st);
  }
  return out;
}
