// Fused normalization kernels: LayerNorm(+residual) and RMSNorm(+residual).
//
// One workgroup per row, single pass (sum + sumsq), block reduction, then a
// normalize+scale pass -- the row stays in registers between the two phases
// for hidden sizes up to 8 floats/lane * 256 lanes * REG_ROWS. Larger rows
// fall back to a re-read pass (still one extra L2-resident read, not HBM,
// for serving-sized hiddens).
#include "common.h"

namespace {

template <typename T, bool HAS_RESIDUAL, bool RMS, bool WRITE_RESIDUAL>
__global__ void norm_kernel(const T* __restrict__ x,
                            const T* __restrict__ weight,
                            const T* __restrict__ bias,
                            T* __restrict__ residual,  // in/out when WRITE
                            T* __restrict__ out, int rows, int h, float eps) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* xrow = x + (long)row * h;
  const T* rrow = HAS_RESIDUAL ? residual + (long)row * h : nullptr;
  T* orow = out + (long)row * h;

  // registers hold up to VALS values per lane (h <= VALS*blockDim.x fast path)
  constexpr int VALS = 16;
  float v[VALS];
  const int per_lane = (h + blockDim.x - 1) / blockDim.x;
  const bool in_regs = per_lane <= VALS;

  float sum = 0.f, sumsq = 0.f;
  for (int i = 0; i < per_lane; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    float f = 0.f;
    if (idx < h) {
      f = to_f32(xrow[idx]);
      if (HAS_RESIDUAL) f += to_f32(rrow[idx]);
      if (WRITE_RESIDUAL && idx < h)
        (residual + (long)row * h)[idx] = from_f32<T>(f);
    }
    if (in_regs && i < VALS) v[i] = f;
    sum += f;
    sumsq += f * f;
  }
  const float mean = RMS ? 0.f
                         : block_reduce(sum, tmp, SumOp{}, 0.f) / (float)h;
  __syncthreads();  // tmp reuse
  float ssq = block_reduce(sumsq, tmp, SumOp{}, 0.f);
  float var = RMS ? ssq / (float)h : ssq / (float)h - mean * mean;
  const float inv_std = rsqrtf(var + eps);

  for (int i = 0; i < per_lane; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= h) break;
    float f;
    if (in_regs && i < VALS) {
      f = v[i];
    } else {
      f = to_f32(xrow[idx]);
      if (HAS_RESIDUAL) f += to_f32(rrow[idx]);
    }
    float y = (f - mean) * inv_std * to_f32(weight[idx]);
    if (!RMS) y += to_f32(bias[idx]);
    orow[idx] = from_f32<T>(y);
  }
}

#define DISPATCH_FLOAT_TYPES_N(TENSOR, NAME, ...)                            \
  [&] {                                                                      \
    const auto _st = (TENSOR).scalar_type();                                 \
    if (_st == at::kBFloat16) {                                              \
      using scalar_t = __hip_bfloat16;                                       \
      return __VA_ARGS__();                                                  \
    } else if (_st == at::kHalf) {                                           \
      using scalar_t = __half;                                               \
      return __VA_ARGS__();                                                  \
    } else if (_st == at::kFloat) {                                          \
      using scalar_t = float;                                                \
      return __VA_ARGS__();                                                  \
    }                                                                        \
    TORCH_CHECK(false, NAME, ": unsupported dtype ", (TENSOR).scalar_type()); \
  }()

template <typename T>
void launch_norm(const torch::Tensor& x, const torch::Tensor& weight,
                 const torch::Tensor* bias, torch::Tensor* residual,
                 torch::Tensor& out, float eps, bool rms,
                 bool write_residual) {
  const int h = x.size(-1);
  const int rows = x.numel() / h;
  const int block = std::min(1024, ((h + 63) / 64) * 64);
  const int smem = 16 * sizeof(float);
  hipStream_t stream_ = cmls::current_stream();

  const T* xp = (const T*)x.data_ptr();
  const T* wp = (const T*)weight.data_ptr();
  const T* bp = bias ? (const T*)bias->data_ptr() : nullptr;
  T* rp = residual ? (T*)residual->data_ptr() : nullptr;
  T* op = (T*)out.data_ptr();

#define LAUNCH(HAS_R, IS_RMS, WR)                                            \
  hipLaunchKernelGGL((norm_kernel<T, HAS_R, IS_RMS, WR>), dim3(rows),        \
                     dim3(block), smem, stream_, xp, wp, bp, rp, op, \
                     rows, h, eps)
  if (rms) {
    if (rp && write_residual) LAUNCH(true, true, true);
    else if (rp) LAUNCH(true, true, false);
    else LAUNCH(false, true, false);
  } else {
    if (rp) LAUNCH(true, false, false);
    else LAUNCH(false, false, false);
  }
#undef LAUNCH
}

}  // namespace

torch::Tensor layernorm(torch::Tensor x, torch::Tensor weight,
                        torch::Tensor bias, double eps,
                        c10::optional<torch::Tensor> residual) {
  CHECK_LASTDIM_CONTIG(x);
  TORCH_CHECK(weight.is_contiguous() && bias.is_contiguous());
  TORCH_CHECK(weight.numel() == x.size(-1), "weight/hidden mismatch");
  torch::Tensor res;
  if (residual.has_value() && residual->defined()) {
    res = residual->contiguous();
    TORCH_CHECK(res.sizes() == x.sizes(), "residual shape mismatch");
  }
  auto out = torch::empty_like(x);
  DISPATCH_FLOAT_TYPES_N(x, "layernorm", [&] {
    launch_norm<scalar_t>(x, weight, &bias, res.defined() ? &res : nullptr,
                          out, (float)eps, /*rms=*/false, false);
  });
  return out;
}

torch::Tensor rmsnorm(torch::Tensor x, torch::Tensor weight, double eps,
                      c10::optional<torch::Tensor> residual) {
  CHECK_LASTDIM_CONTIG(x);
  TORCH_CHECK(weight.is_contiguous());
  torch::Tensor res;
  bool write_res = false;
  if (residual.has_value() && residual->defined()) {
    res = *residual;
    TORCH_CHECK(res.is_contiguous() && res.sizes() == x.sizes());
    write_res = true;  // llama running-residual update (ops/__init__.py)
  }
  auto out = torch::empty_like(x);
  DISPATCH_FLOAT_TYPES_N(x, "rmsnorm", [&] {
    launch_norm<scalar_t>(x, weight, nullptr, res.defined() ? &res : nullptr,
                          out, (float)eps, /*rms=*/true, write_res);
  });
  return out;
}
