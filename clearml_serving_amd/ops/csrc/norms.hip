// Fused normalization kernels: LayerNorm(+residual) and RMSNorm(+residual).
// Part of the in-process kernel tier that replaces the reference's Triton
// delegation (SURVEY.md §2.6 'LayerNorm/Softmax' row; the reference has no
// kernels of its own -- Triton/vLLM supply them).
//
// One workgroup per row; 16-byte vector loads (8 bf16 / 4 fp32 per lane per
// access -- scalar bf16 loads measured ~870 GB/s at h=768, vectorized path
// targets the HBM roofline). Row data stays in registers between the
// statistics pass and the normalize pass (up to VECS * 16 B per lane).
#include "common.h"

namespace {

template <typename T, int VECS, bool HAS_RESIDUAL, bool RMS, bool WRITE_RESIDUAL>
__global__ void norm_vec_kernel(const T* __restrict__ x,
                                const T* __restrict__ weight,
                                const T* __restrict__ bias,
                                T* __restrict__ residual,  // in/out when WRITE
                                T* __restrict__ out, int rows, int h,
                                float eps) {
  constexpr int VE = 16 / sizeof(T);  // elements per 16-B vector
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int nvec = h / VE;  // guaranteed divisible by caller
  using V = uint32x4;
  const V* xrow = reinterpret_cast<const V*>(x + (long)row * h);
  V* rrow = HAS_RESIDUAL
                ? reinterpret_cast<V*>(residual + (long)row * h)
                : nullptr;
  V* orow = reinterpret_cast<V*>(out + (long)row * h);

  union U { V v; T e[VE]; };
  float f[VECS][VE];
  float sum = 0.f, sumsq = 0.f;
#pragma unroll
  for (int i = 0; i < VECS; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx < nvec) {
      U a;
      a.v = xrow[idx];
      U r;
      if (HAS_RESIDUAL) r.v = rrow[idx];
#pragma unroll
      for (int j = 0; j < VE; ++j) {
        float val = to_f32(a.e[j]);
        if (HAS_RESIDUAL) val += to_f32(r.e[j]);
        f[i][j] = val;
        sum += val;
        sumsq += val * val;
      }
      if (WRITE_RESIDUAL) {
        U w;
#pragma unroll
        for (int j = 0; j < VE; ++j) w.e[j] = from_f32<T>(f[i][j]);
        rrow[idx] = w.v;
      }
    } else {
#pragma unroll
      for (int j = 0; j < VE; ++j) f[i][j] = 0.f;
    }
  }
  float mean = 0.f;
  if (!RMS) {
    mean = block_reduce(sum, tmp, SumOp{}, 0.f) / (float)h;
    __syncthreads();
  }
  const float ssq = block_reduce(sumsq, tmp, SumOp{}, 0.f);
  const float var = RMS ? ssq / (float)h : ssq / (float)h - mean * mean;
  const float inv_std = rsqrtf(var + eps);

  const V* wrow = reinterpret_cast<const V*>(weight);
  const V* brow = RMS ? nullptr : reinterpret_cast<const V*>(bias);
#pragma unroll
  for (int i = 0; i < VECS; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= nvec) break;
    U w, b, o;
    w.v = wrow[idx];
    if (!RMS) b.v = brow[idx];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float y = (f[i][j] - mean) * inv_std * to_f32(w.e[j]);
      if (!RMS) y += to_f32(b.e[j]);
      o.e[j] = from_f32<T>(y);
    }
    orow[idx] = o.v;
  }
}

// generic fallback for odd hidden sizes (scalar path)
template <typename T, bool HAS_RESIDUAL, bool RMS, bool WRITE_RESIDUAL>
__global__ void norm_scalar_kernel(const T* __restrict__ x,
                                   const T* __restrict__ weight,
                                   const T* __restrict__ bias,
                                   T* __restrict__ residual,
                                   T* __restrict__ out, int rows, int h,
                                   float eps) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* xrow = x + (long)row * h;
  T* rrow = HAS_RESIDUAL ? residual + (long)row * h : nullptr;
  T* orow = out + (long)row * h;

  float sum = 0.f, sumsq = 0.f;
  for (int idx = threadIdx.x; idx < h; idx += blockDim.x) {
    float val = to_f32(xrow[idx]);
    if (HAS_RESIDUAL) val += to_f32(rrow[idx]);
    if (WRITE_RESIDUAL) rrow[idx] = from_f32<T>(val);
    sum += val;
    sumsq += val * val;
  }
  float mean = 0.f;
  if (!RMS) {
    mean = block_reduce(sum, tmp, SumOp{}, 0.f) / (float)h;
    __syncthreads();
  }
  const float ssq = block_reduce(sumsq, tmp, SumOp{}, 0.f);
  const float var = RMS ? ssq / (float)h : ssq / (float)h - mean * mean;
  const float inv_std = rsqrtf(var + eps);
  for (int idx = threadIdx.x; idx < h; idx += blockDim.x) {
    float val = to_f32(xrow[idx]);
    if (HAS_RESIDUAL) val += to_f32(rrow[idx]);
    float y = (val - mean) * inv_std * to_f32(weight[idx]);
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
  constexpr int VE = 16 / sizeof(T);
  const int h = x.size(-1);
  const int rows = x.numel() / h;
  const int smem = 16 * sizeof(float);
  hipStream_t stream_ = cmls::current_stream();

  const T* xp = (const T*)x.data_ptr();
  const T* wp = (const T*)weight.data_ptr();
  const T* bp = bias ? (const T*)bias->data_ptr() : nullptr;
  T* rp = residual ? (T*)residual->data_ptr() : nullptr;
  T* op = (T*)out.data_ptr();

  const bool has_r = rp != nullptr;
  if (h % VE == 0) {
    const int nvec = h / VE;
    // one vector per lane per step; block sized so VECS stays small
    int block = std::min(1024, ((nvec + 63) / 64) * 64);
    const int vecs = (nvec + block - 1) / block;
#define LV(VECS_, HR, RM, WR)                                                 \
  hipLaunchKernelGGL((norm_vec_kernel<T, VECS_, HR, RM, WR>), dim3(rows),     \
                     dim3(block), smem, stream_, xp, wp, bp, rp, op, rows, h, \
                     eps)
#define LV_ALL(VECS_)                                                         \
  do {                                                                        \
    if (rms) {                                                                \
      if (has_r && write_residual) LV(VECS_, true, true, true);               \
      else if (has_r) LV(VECS_, true, true, false);                           \
      else LV(VECS_, false, true, false);                                     \
    } else {                                                                  \
      if (has_r) LV(VECS_, true, false, false);                               \
      else LV(VECS_, false, false, false);                                    \
    }                                                                         \
  } while (0)
    if (vecs <= 1) LV_ALL(1);
    else if (vecs == 2) LV_ALL(2);
    else if (vecs <= 4) LV_ALL(4);
    else if (vecs <= 8) LV_ALL(8);
    else {
      goto scalar;  // very large hidden: re-read path
    }
#undef LV_ALL
#undef LV
    return;
  }
scalar : {
  const int block = std::min(1024, ((h + 63) / 64) * 64);
#define LS(HR, RM, WR)                                                       \
  hipLaunchKernelGGL((norm_scalar_kernel<T, HR, RM, WR>), dim3(rows),        \
                     dim3(block), smem, stream_, xp, wp, bp, rp, op, rows,   \
                     h, eps)
  if (rms) {
    if (has_r && write_residual) LS(true, true, true);
    else if (has_r) LS(true, true, false);
    else LS(false, true, false);
  } else {
    if (has_r) LS(true, false, false);
    else LS(false, false, false);
  }
#undef LS
}
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
