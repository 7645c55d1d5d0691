// Common helpers for the gfx950 (CDNA4) kernel library.
// Wave size is 64 on CDNA4 -- every cross-lane idiom below is 64-wide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

// CMLS_KERNEL_ONLY: torch-free build for standalone kernel harnesses
// (scripts/*.hip) that #include a kernel source directly.
#ifndef CMLS_KERNEL_ONLY
#include <c10/hip/HIPStream.h>
#include <torch/extension.h>

#include <algorithm>

// current-stream accessor for the ROCm build (ATen/cuda/CUDAContext.h pulls
// cuda_runtime_api.h, which does not exist here)
namespace cmls {
inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}
}  // namespace cmls

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));    \
  } while (0)

#define CHECK_LASTDIM_CONTIG(x)                                             \
  TORCH_CHECK((x).stride(-1) == 1, #x " must be contiguous in last dim")
#endif  // CMLS_KERNEL_ONLY

#define WAVE_SIZE 64

// ---------------------------------------------------------------------- //
// dtype conversion
// ---------------------------------------------------------------------- //
template <typename T> struct VecTraits;

__device__ __forceinline__ float to_f32(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ float to_f32(__half v) { return __half2float(v); }
__device__ __forceinline__ float to_f32(float v) { return v; }

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ __hip_bfloat16 from_f32(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ __half from_f32(float v) {
  return __float2half(v);
}
template <> __device__ __forceinline__ float from_f32(float v) { return v; }

// ---------------------------------------------------------------------- //
// wave + workgroup reductions (wave64)
// ---------------------------------------------------------------------- //
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// block reduction across up to 16 waves; `tmp` must hold >= nwaves floats.
template <typename Op>
__device__ __forceinline__ float block_reduce(float v, float* tmp, Op op,
                                              float identity) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wave = threadIdx.x / WAVE_SIZE;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) / WAVE_SIZE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE_SIZE));
  if (lane == 0) tmp[wave] = v;
  __syncthreads();
  v = (lane < nwaves) ? tmp[lane] : identity;
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE_SIZE));
  // every lane of wave 0 has the result; broadcast through tmp
  if (threadIdx.x == 0) tmp[0] = v;
  __syncthreads();
  return tmp[0];
}

struct SumOp {
  __device__ float operator()(float a, float b) const { return a + b; }
};
struct MaxOp {
  __device__ float operator()(float a, float b) const { return fmaxf(a, b); }
};

// ---------------------------------------------------------------------- //
// bf16x8 (16-byte) vector load/store helpers
// ---------------------------------------------------------------------- //
typedef __attribute__((ext_vector_type(4))) unsigned int uint32x4;
typedef __attribute__((ext_vector_type(2))) unsigned int uint32x2;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(8))) float floatx8;
typedef __attribute__((ext_vector_type(16))) float floatx16;
// 8 bf16 packed in 4 dwords
union bf16x8 {
  uint32x4 u;
  __hip_bfloat16 h[8];
};
