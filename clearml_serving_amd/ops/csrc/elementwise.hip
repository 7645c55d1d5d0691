// Fused elementwise epilogues (memory-bound: one HBM pass, 16 B/lane vectors).
//
// bias_relu_add : out = relu(x [+ bias_c] [+ residual])   (ResNet join)
// bias_gelu     : out = gelu_erf(x + bias_col)            (BERT MLP)
// silu_mul      : out = silu(gate) * up                   (llama MLP)
//
// Replaces work the reference delegates to Triton's engine internals
// (SURVEY.md §2.6 delegation table).
#include "common.h"

namespace {

// ------------------------- bias_relu_add ------------------------------ //
// vectorized path: no per-channel bias (the common ResNet case after BN
// folding -- conv carries the bias), 8 elements per lane.
template <typename T>
__global__ void relu_add_vec8_kernel(const T* __restrict__ x,
                                     const T* __restrict__ residual,
                                     T* __restrict__ out, long n8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  using V = uint32x4;  // 16 B
  const V* xv = reinterpret_cast<const V*>(x);
  const V* rv = reinterpret_cast<const V*>(residual);
  V* ov = reinterpret_cast<V*>(out);
  for (; i < n8; i += stride) {
    union { V v; T e[16 / sizeof(T)]; } a, b;
    a.v = xv[i];
    if (residual) b.v = rv[i];
#pragma unroll
    for (int j = 0; j < (int)(16 / sizeof(T)); ++j) {
      float f = to_f32(a.e[j]) + (residual ? to_f32(b.e[j]) : 0.f);
      a.e[j] = from_f32<T>(fmaxf(f, 0.f));
    }
    ov[i] = a.v;
  }
}

// scalar path with per-channel bias (NCHW): c = (i / hw) % C
template <typename T>
__global__ void bias_relu_add_kernel(const T* __restrict__ x,
                                     const T* __restrict__ bias,
                                     const T* __restrict__ residual,
                                     T* __restrict__ out, long n, long hw,
                                     long c) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float f = to_f32(x[i]);
    if (bias) f += to_f32(bias[(i / hw) % c]);
    if (residual) f += to_f32(residual[i]);
    out[i] = from_f32<T>(fmaxf(f, 0.f));
  }
}

// --------------------------- bias_gelu -------------------------------- //
__device__ __forceinline__ float gelu_erf(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

template <typename T>
__global__ void bias_gelu_vec8_kernel(const T* __restrict__ x,
                                      const T* __restrict__ bias,
                                      T* __restrict__ out, long n8, long c8) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  constexpr int VE = 16 / sizeof(T);
  using V = uint32x4;
  const V* xv = reinterpret_cast<const V*>(x);
  const V* bv = reinterpret_cast<const V*>(bias);
  V* ov = reinterpret_cast<V*>(out);
  for (; i < n8; i += stride) {
    union { V v; T e[VE]; } a, b;
    a.v = xv[i];
    if (bias) b.v = bv[i % c8];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float f = to_f32(a.e[j]) + (bias ? to_f32(b.e[j]) : 0.f);
      a.e[j] = from_f32<T>(gelu_erf(f));
    }
    ov[i] = a.v;
  }
}

// --------------------------- silu_mul --------------------------------- //
template <typename T>
__global__ void silu_mul_vec8_kernel(const T* __restrict__ gate,
                                     const T* __restrict__ up,
                                     T* __restrict__ out, long n8, long c8,
                                     long gs8, long us8) {
  // rows of length c8 vectors; gate/up row strides gs8/us8 (in vectors) --
  // accepts views into a merged gate_up projection without a copy
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  constexpr int VE = 16 / sizeof(T);
  using V = uint32x4;
  const V* gv = reinterpret_cast<const V*>(gate);
  const V* uv = reinterpret_cast<const V*>(up);
  V* ov = reinterpret_cast<V*>(out);
  for (; i < n8; i += stride) {
    const long r = i / c8, c = i % c8;
    union { V v; T e[VE]; } g, u;
    g.v = gv[r * gs8 + c];
    u.v = uv[r * us8 + c];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float x = to_f32(g.e[j]);
      float s = x / (1.0f + __expf(-x));
      g.e[j] = from_f32<T>(s * to_f32(u.e[j]));
    }
    ov[i] = g.v;
  }
}

inline int grid_for(long work_items, int block = 256) {
  // >= 2048 workgroups fills 256 CUs across 8 XCDs with headroom
  long g = (work_items + block - 1) / block;
  return (int)std::min<long>(g, 65535L * 8);
}

}  // namespace

#define DISPATCH_FLOAT_TYPES(TENSOR, NAME, ...)                              \
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

torch::Tensor bias_relu_add(torch::Tensor x,
                            c10::optional<torch::Tensor> bias,
                            c10::optional<torch::Tensor> residual) {
  const bool has_bias = bias.has_value() && bias->defined();
  const bool has_res = residual.has_value() && residual->defined();
  // the no-bias path is pure elementwise over dense memory -- any dense
  // layout (incl. channels_last NHWC) is fine as long as x and residual
  // share it; the channel-bias path needs standard NCHW strides
  if (has_bias) {
    TORCH_CHECK(x.is_contiguous(), "bias path needs standard contiguous x");
  } else {
    TORCH_CHECK(x.is_non_overlapping_and_dense(), "x must be dense");
  }
  auto out = torch::empty_like(x);
  const long n = x.numel();
  if (n == 0) return out;
  hipStream_t stream_ = cmls::current_stream();
  if (has_res) {
    TORCH_CHECK(residual->sizes() == x.sizes() &&
                    residual->strides() == x.strides(),
                "residual must match x layout");
  }

  DISPATCH_FLOAT_TYPES(x, "bias_relu_add", [&] {
    constexpr int VE = 16 / sizeof(scalar_t);
    if (!has_bias && n % VE == 0) {
      const long n8 = n / VE;
      hipLaunchKernelGGL(relu_add_vec8_kernel<scalar_t>, dim3(grid_for(n8)),
                         dim3(256), 0, stream_,
                         (const scalar_t*)x.data_ptr(),
                         has_res ? (const scalar_t*)residual->data_ptr() : nullptr,
                         (scalar_t*)out.data_ptr(), n8);
    } else {
      long c = 1, hw = 1;
      if (has_bias) {
        TORCH_CHECK(x.dim() >= 2, "bias needs channel dim");
        c = x.size(1);
        hw = n / (x.size(0) * c);
        TORCH_CHECK(bias->numel() == c, "bias numel != channels");
      }
      hipLaunchKernelGGL(bias_relu_add_kernel<scalar_t>, dim3(grid_for(n)),
                         dim3(256), 0, stream_,
                         (const scalar_t*)x.data_ptr(),
                         has_bias ? (const scalar_t*)bias->data_ptr() : nullptr,
                         has_res ? (const scalar_t*)residual->data_ptr() : nullptr,
                         (scalar_t*)out.data_ptr(), n, hw, c);
    }
  });
  return out;
}

torch::Tensor bias_gelu(torch::Tensor x, c10::optional<torch::Tensor> bias) {
  CHECK_LASTDIM_CONTIG(x);
  auto out = torch::empty_like(x);
  const long n = x.numel();
  if (n == 0) return out;
  const long c = x.size(-1);
  const bool has_bias = bias.has_value() && bias->defined();
  if (has_bias)
    TORCH_CHECK(bias->numel() == c && bias->is_contiguous(),
                "bias must be contiguous [last_dim]");
  hipStream_t stream_ = cmls::current_stream();
  DISPATCH_FLOAT_TYPES(x, "bias_gelu", [&] {
    constexpr int VE = 16 / sizeof(scalar_t);
    TORCH_CHECK(c % VE == 0, "last dim must be a multiple of ", VE);
    const long n8 = n / VE;
    hipLaunchKernelGGL(bias_gelu_vec8_kernel<scalar_t>, dim3(grid_for(n8)),
                       dim3(256), 0, stream_,
                       (const scalar_t*)x.data_ptr(),
                       has_bias ? (const scalar_t*)bias->data_ptr() : nullptr,
                       (scalar_t*)out.data_ptr(), n8, c / VE);
  });
  return out;
}

torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up) {
  CHECK_LASTDIM_CONTIG(gate);
  CHECK_LASTDIM_CONTIG(up);
  TORCH_CHECK(gate.sizes() == up.sizes());
  TORCH_CHECK(gate.dim() <= 2 || gate.is_contiguous(),
              "silu_mul: >2D inputs must be contiguous");
  const long rows = gate.dim() == 2 ? gate.size(0) : 1;
  const long cols = gate.numel() / rows;
  auto out = torch::empty({rows, cols}, gate.options()).view(gate.sizes());
  const long n = gate.numel();
  if (n == 0) return out;
  hipStream_t stream_ = cmls::current_stream();
  const long gs = gate.dim() == 2 ? gate.stride(0) : cols;
  const long us = up.dim() == 2 ? up.stride(0) : cols;
  DISPATCH_FLOAT_TYPES(gate, "silu_mul", [&] {
    constexpr int VE = 16 / sizeof(scalar_t);
    TORCH_CHECK(cols % VE == 0 && gs % VE == 0 && us % VE == 0,
                "row length/strides must be multiples of ", VE);
    hipLaunchKernelGGL(silu_mul_vec8_kernel<scalar_t>, dim3(grid_for(n / VE)),
                       dim3(256), 0, stream_,
                       (const scalar_t*)gate.data_ptr(),
                       (const scalar_t*)up.data_ptr(),
                       (scalar_t*)out.data_ptr(), n / VE, cols / VE,
                       gs / VE, us / VE);
  });
  return out;
}
