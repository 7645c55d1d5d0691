// fp8 (OCP e4m3) decode path: fused activation quantization + skinny GEMM.
//
// Decode is weight-streaming bound; fp8 weights halve the bytes per step
// (non-scaled fp8 MFMA runs at the bf16 rate on gfx950 -- the win is HBM
// bytes, not matrix-core rate). Round 1 shipped fp8 via hipBLASLt with FOUR
// un-fused elementwise passes of dynamic activation quantization per
// projection and measured SLOWER than bf16 end to end (models/quant.py,
// docs/ROADMAP.md). This file removes that overhead:
//
//   rmsnorm_fp8  : residual += x; y = rmsnorm(residual) -> fp8 + row scale
//   silu_mul_fp8 : y = silu(gate) * up                  -> fp8 + row scale
//   quant_fp8    : y = x                                -> fp8 + row scale
//   skinny_gemm_fp8 : C[M,N] = (Aq * s_a) @ (Wq * s_w)^T, M <= 64,
//                     one workgroup per 16 columns (fills 256 CUs at the
//                     llama projection N), 8 waves split K, MFMA
//                     16x16x32_fp8_fp8, deterministic LDS reduction.
//
// Scales: per-token (row) for activations, per-output-channel for weights
// (quantized once at load, models/quant.py). e4m3 max = 448.
//
// Replaces work the reference delegates to vLLM's quantization stack
// (reference preprocess_service.py:619-1095).
#include "common.h"

namespace {

constexpr float F8_MAX = 448.0f;

// pack 8 fp32 -> 8 e4m3 bytes (saturating hardware cvt)
__device__ __forceinline__ void pack_fp8x8(const float* v, unsigned char* out) {
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int packed = __builtin_amdgcn_cvt_pk_fp8_f32(v[2 * p], v[2 * p + 1], 0,
                                                 false);
    out[2 * p] = packed & 0xff;
    out[2 * p + 1] = (packed >> 8) & 0xff;
  }
}

// ------------------------------------------------------------------- //
// rmsnorm_fp8: one workgroup per row; residual updated in place (llama
// running-residual contract, same as norms.hip rmsnorm), normalized row
// quantized to fp8 with scale = absmax/448.
// ------------------------------------------------------------------- //
template <typename T, int VECS, bool HAS_RESIDUAL>
__global__ void rmsnorm_fp8_kernel(const T* __restrict__ x,
                                   const T* __restrict__ weight,
                                   T* __restrict__ residual,  // in/out
                                   unsigned char* __restrict__ out8,
                                   float* __restrict__ scales, int rows,
                                   int h, float eps) {
  constexpr int VE = 16 / sizeof(T);
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const int nvec = h / VE;
  using V = uint32x4;
  const V* xrow = reinterpret_cast<const V*>(x + (long)row * h);
  V* rrow = HAS_RESIDUAL ? reinterpret_cast<V*>(residual + (long)row * h)
                         : nullptr;

  union U { V v; T e[VE]; };
  float f[VECS][VE];
  float sumsq = 0.f;
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
        sumsq += val * val;
      }
      if (HAS_RESIDUAL) {
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
  const float ssq = block_reduce(sumsq, tmp, SumOp{}, 0.f);
  const float inv_std = rsqrtf(ssq / (float)h + eps);
  __syncthreads();

  // normalized values + row absmax
  const V* wrow = reinterpret_cast<const V*>(weight);
  float amax = 0.f;
#pragma unroll
  for (int i = 0; i < VECS; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= nvec) break;
    U w;
    w.v = wrow[idx];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      f[i][j] = f[i][j] * inv_std * to_f32(w.e[j]);
      amax = fmaxf(amax, fabsf(f[i][j]));
    }
  }
  const float row_amax = block_reduce(amax, tmp, MaxOp{}, 0.f);
  const float scale = fmaxf(row_amax, 1e-8f) / F8_MAX;
  const float inv_scale = 1.0f / scale;
  if (threadIdx.x == 0) scales[row] = scale;

  static_assert(VE % 8 == 0 || VE == 4, "vector width");
  unsigned char* orow = out8 + (long)row * h;
#pragma unroll
  for (int i = 0; i < VECS; ++i) {
    const int idx = i * blockDim.x + threadIdx.x;
    if (idx >= nvec) break;
    float q[VE];
#pragma unroll
    for (int j = 0; j < VE; ++j) q[j] = f[i][j] * inv_scale;
    if constexpr (VE == 8) {
      unsigned char bytes[8];
      pack_fp8x8(q, bytes);
      *reinterpret_cast<uint2*>(orow + (long)idx * 8) =
          *reinterpret_cast<const uint2*>(bytes);
    } else {  // fp32 input: VE == 4
      unsigned char bytes[4];
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        int packed = __builtin_amdgcn_cvt_pk_fp8_f32(q[2 * p], q[2 * p + 1],
                                                     0, false);
        bytes[2 * p] = packed & 0xff;
        bytes[2 * p + 1] = (packed >> 8) & 0xff;
      }
      *reinterpret_cast<unsigned int*>(orow + (long)idx * 4) =
          *reinterpret_cast<const unsigned int*>(bytes);
    }
  }
}

// ------------------------------------------------------------------- //
// silu_mul_fp8 / quant_fp8: one workgroup per row, values re-read on the
// quantize pass when the row exceeds the register budget (decode rows are
// tiny; one extra L2-hot pass is cheaper than spilling).
// ------------------------------------------------------------------- //
template <typename T>
__global__ void silu_mul_fp8_kernel(const T* __restrict__ gate,
                                    const T* __restrict__ up,
                                    unsigned char* __restrict__ out8,
                                    float* __restrict__ scales, int rows,
                                    int h, long gstride, long ustride) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  constexpr int VE = 16 / sizeof(T);
  using V = uint32x4;
  const V* grow = reinterpret_cast<const V*>(gate + (long)row * gstride);
  const V* urow = reinterpret_cast<const V*>(up + (long)row * ustride);
  const int nvec = h / VE;

  union U { V v; T e[VE]; };
  float amax = 0.f;
  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    U g, u;
    g.v = grow[idx];
    u.v = urow[idx];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float xg = to_f32(g.e[j]);
      float s = xg / (1.0f + __expf(-xg));
      amax = fmaxf(amax, fabsf(s * to_f32(u.e[j])));
    }
  }
  const float row_amax = block_reduce(amax, tmp, MaxOp{}, 0.f);
  const float scale = fmaxf(row_amax, 1e-8f) / F8_MAX;
  const float inv_scale = 1.0f / scale;
  if (threadIdx.x == 0) scales[row] = scale;

  unsigned char* orow = out8 + (long)row * h;
  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    U g, u;
    g.v = grow[idx];
    u.v = urow[idx];
    float q[VE];
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float xg = to_f32(g.e[j]);
      float s = xg / (1.0f + __expf(-xg));
      q[j] = s * to_f32(u.e[j]) * inv_scale;
    }
    if constexpr (VE == 8) {
      unsigned char bytes[8];
      pack_fp8x8(q, bytes);
      *reinterpret_cast<uint2*>(orow + (long)idx * 8) =
          *reinterpret_cast<const uint2*>(bytes);
    } else {
      unsigned char bytes[4];
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        int packed = __builtin_amdgcn_cvt_pk_fp8_f32(q[2 * p], q[2 * p + 1],
                                                     0, false);
        bytes[2 * p] = packed & 0xff;
        bytes[2 * p + 1] = (packed >> 8) & 0xff;
      }
      *reinterpret_cast<unsigned int*>(orow + (long)idx * 4) =
          *reinterpret_cast<const unsigned int*>(bytes);
    }
  }
}

template <typename T>
__global__ void quant_fp8_kernel(const T* __restrict__ x,
                                 unsigned char* __restrict__ out8,
                                 float* __restrict__ scales, int rows,
                                 int h) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  constexpr int VE = 16 / sizeof(T);
  using V = uint32x4;
  const V* xrow = reinterpret_cast<const V*>(x + (long)row * h);
  const int nvec = h / VE;
  union U { V v; T e[VE]; };
  float amax = 0.f;
  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    U a;
    a.v = xrow[idx];
#pragma unroll
    for (int j = 0; j < VE; ++j) amax = fmaxf(amax, fabsf(to_f32(a.e[j])));
  }
  const float row_amax = block_reduce(amax, tmp, MaxOp{}, 0.f);
  const float scale = fmaxf(row_amax, 1e-8f) / F8_MAX;
  const float inv_scale = 1.0f / scale;
  if (threadIdx.x == 0) scales[row] = scale;
  unsigned char* orow = out8 + (long)row * h;
  for (int idx = threadIdx.x; idx < nvec; idx += blockDim.x) {
    U a;
    a.v = xrow[idx];
    float q[VE];
#pragma unroll
    for (int j = 0; j < VE; ++j) q[j] = to_f32(a.e[j]) * inv_scale;
    if constexpr (VE == 8) {
      unsigned char bytes[8];
      pack_fp8x8(q, bytes);
      *reinterpret_cast<uint2*>(orow + (long)idx * 8) =
          *reinterpret_cast<const uint2*>(bytes);
    } else {
      unsigned char bytes[4];
#pragma unroll
      for (int p = 0; p < 2; ++p) {
        int packed = __builtin_amdgcn_cvt_pk_fp8_f32(q[2 * p], q[2 * p + 1],
                                                     0, false);
        bytes[2 * p] = packed & 0xff;
        bytes[2 * p + 1] = (packed >> 8) & 0xff;
      }
      *reinterpret_cast<unsigned int*>(orow + (long)idx * 4) =
          *reinterpret_cast<const unsigned int*>(bytes);
    }
  }
}

// ------------------------------------------------------------------- //
// skinny_gemm_fp8: structure of skinny_gemm.hip v1 (one WG per 16 columns
// of W, 8 waves split K, deterministic LDS reduce), operands fp8 so every
// 16-byte lane load covers TWO MFMA k-steps (K=64 per load).
//   mfma_f32_16x16x32_fp8_fp8: a/b lane l holds row (l%16),
//   k = (l>>4)*8 + e (8 e4m3 bytes = one i64 operand); C map as bf16.
// ------------------------------------------------------------------- //
typedef __attribute__((ext_vector_type(2))) long longx2_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA16F8(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8((A), (B), (C), 0, 0, 0)

constexpr int F8_NT = 16;    // N columns per workgroup
constexpr int F8_NW = 8;     // waves (split K)
constexpr int F8_KSTEP = 64; // k per 16-byte load (2 MFMAs)

template <int MT>  // 16-row m tiles: M <= 16*MT (MT 1..4 -> M <= 64)
__global__ __launch_bounds__(F8_NW * 64, 2) void skinny_gemm_fp8_kernel(
    const unsigned char* __restrict__ a,  // [M, K] e4m3 row-major
    const float* __restrict__ a_scale,    // [M]
    const unsigned char* __restrict__ w,  // [N, K] e4m3 SWIZZLED (see note)
    const float* __restrict__ w_scale,    // [N]
    __hip_bfloat16* __restrict__ c,       // [M, N]
    int M, int N, int K) {
  const int n0 = blockIdx.x * F8_NT;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row = lane & 15;
  const int g = lane >> 4;  // lane group (k fragment)

  const int kchunk = ((K / F8_KSTEP + F8_NW - 1) / F8_NW) * F8_KSTEP;
  const int k_lo = wave * kchunk;
  const int k_hi = min(K, k_lo + kchunk);

  f32x4_t acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  // W layout is PRE-SWIZZLED at quantization time (models/quant.py
  // swizzle_fp8_weight): within each 64-byte k window-pair, the bytes are
  // reordered [group][window][8] so ONE 16-byte lane load yields both MFMA
  // operands of the pair. An un-swizzled row would force two 8-byte loads
  // per pair (one per 32-wide k window: this lane group's bytes live at
  // +8g and +32+8g), doubling the HBM request rate -- measured to erase
  // the fp8 byte advantage entirely (67.7us vs bf16's 69.6us at M=1).
  const unsigned char* wrow = w + (long)(n0 + row) * K + 16 * g;
  // A stays row-major (L2-resident; every WG reads the same M*K tile):
  // its two operands per pair load as 2x8 bytes at +8g and +32+8g.
  const unsigned char* arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    const int m = t * 16 + row;
    arow[t] = a + (long)(m < M ? m : 0) * K + 8 * g;
  }

  int k = k_lo;
  for (; k + 2 * F8_KSTEP <= k_hi; k += 2 * F8_KSTEP) {
    longx2_t bw0 = *reinterpret_cast<const longx2_t*>(wrow + k);
    longx2_t bw1 = *reinterpret_cast<const longx2_t*>(wrow + k + 64);
    long ba0a[MT], ba0b[MT], ba1a[MT], ba1b[MT];
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      ba0a[t] = *reinterpret_cast<const long*>(arow[t] + k);
      ba0b[t] = *reinterpret_cast<const long*>(arow[t] + k + 32);
      ba1a[t] = *reinterpret_cast<const long*>(arow[t] + k + 64);
      ba1b[t] = *reinterpret_cast<const long*>(arow[t] + k + 96);
    }
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      acc[t] = MFMA16F8(ba0a[t], bw0[0], acc[t]);
      acc[t] = MFMA16F8(ba0b[t], bw0[1], acc[t]);
    }
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      acc[t] = MFMA16F8(ba1a[t], bw1[0], acc[t]);
      acc[t] = MFMA16F8(ba1b[t], bw1[1], acc[t]);
    }
  }
  for (; k < k_hi; k += F8_KSTEP) {
    longx2_t bw = *reinterpret_cast<const longx2_t*>(wrow + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      long baa = *reinterpret_cast<const long*>(arow[t] + k);
      long bab = *reinterpret_cast<const long*>(arow[t] + k + 32);
      acc[t] = MFMA16F8(baa, bw[0], acc[t]);
      acc[t] = MFMA16F8(bab, bw[1], acc[t]);
    }
  }

  __shared__ float lds[F8_NW][MT * 16][F8_NT];
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int t = 0; t < MT; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r)
      lds[wave][t * 16 + crow + r][lane & 15] = acc[t][r];
  }
  __syncthreads();
  if (wave == 0) {
    const int n = lane & 15;
    const int m_base = (lane >> 4) * 4;
    const float ws = (n0 + n < N) ? w_scale[n0 + n] : 0.f;
#pragma unroll
    for (int t = 0; t < MT; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = t * 16 + m_base + r;
        float s = 0.f;
#pragma unroll
        for (int wv = 0; wv < F8_NW; ++wv) s += lds[wv][m][n];
        if (m < M && n0 + n < N)
          c[(long)m * N + n0 + n] =
              __float2bfloat16(s * a_scale[m] * ws);
      }
    }
  }
}

}  // namespace

// ------------------------------------------------------------------- //
// skinny_gemm_fp8 v2: LDS-staged variant for M 17..64 (v1's direct
// fragment loads replay A rows per m-tile and collapse there, like the
// bf16 v1 -> v2 story in skinny_gemm.hip). All 4 waves share one k
// progression; a cooperative full-line stage brings W[16, 128B] and
// A[M16, 128B] k-windows into LDS (pitch 144 B: 36-dword row stride lands
// the 16 fragment rows in 16 distinct banks, rows stay 16-B aligned),
// register-pipelined (issue window w+1's loads before computing w).
// Weights here use the PLAIN fp8 layout (staging reads full lines, so the
// v1 byte swizzle is unnecessary).
// ------------------------------------------------------------------- //
namespace {

constexpr int F8V2_KW = 128;     // k per staged window (128 B/row)
constexpr int F8V2_PITCH = 144;  // LDS row pitch (bytes): 36-dword row
                                 // stride lands the 16 fragment rows in 16
                                 // distinct banks AND keeps rows 16-byte
                                 // aligned for the b128 staging stores
                                 // (136 B would misalign odd rows)
constexpr int F8V2_NW = 4;       // waves

template <int MT>  // m tiles (2..4 -> M 17..64)
__global__ __launch_bounds__(F8V2_NW * 64, 2) void skinny_gemm_fp8_v2_kernel(
    const unsigned char* __restrict__ a,  // [M, K] e4m3 row-major (PLAIN)
    const float* __restrict__ a_scale,    // [M]
    const unsigned char* __restrict__ w,  // [N, K] e4m3 row-major (PLAIN)
    const float* __restrict__ w_scale,    // [N]
    __hip_bfloat16* __restrict__ c,       // [M, N]
    int M, int N, int K) {
  constexpr int M16 = MT * 16;
  constexpr int ROWS = 16 + M16;  // W rows then A rows
  const int n0 = blockIdx.x * F8_NT;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int frow = lane & 15;
  const int fko = (lane >> 4) * 8;  // byte offset of this group's operand

  __shared__ unsigned char lds[2][ROWS * F8V2_PITCH];

  constexpr int TPW = (MT + F8V2_NW - 1) / F8V2_NW;
  f32x4_t acc[TPW];
#pragma unroll
  for (int t = 0; t < TPW; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int n_win = K / F8V2_KW;

  // register-staged pipeline: each thread owns SEGS 16-byte segments of
  // the (ROWS x 128 B) window; consecutive threads cover full lines
  constexpr int SEGS = (ROWS * 8 + F8V2_NW * 64 - 1) / (F8V2_NW * 64);
  longx2_t regs[SEGS];

  auto load_win = [&](int win) {
    const long kbase = (long)win * F8V2_KW;
#pragma unroll
    for (int sgi = 0; sgi < SEGS; ++sgi) {
      const int i = tid + sgi * F8V2_NW * 64;
      if (i >= ROWS * 8) break;
      const int row = i >> 3;
      const int seg = (i & 7) * 16;
      const unsigned char* src =
          row < 16 ? w + (long)(n0 + row) * K + kbase + seg
                   : a + (long)min(row - 16, M - 1) * K + kbase + seg;
      regs[sgi] = *reinterpret_cast<const longx2_t*>(src);
    }
  };
  auto store_win = [&](int buf) {
#pragma unroll
    for (int sgi = 0; sgi < SEGS; ++sgi) {
      const int i = tid + sgi * F8V2_NW * 64;
      if (i >= ROWS * 8) break;
      *reinterpret_cast<longx2_t*>(
          &lds[buf][(i >> 3) * F8V2_PITCH + (i & 7) * 16]) = regs[sgi];
    }
  };

  load_win(0);
  store_win(0);
  if (n_win > 1) load_win(1);
  __syncthreads();
  for (int win = 0; win < n_win; ++win) {
    const int cur = win & 1;
    // four 32-wide MFMA k-steps per window
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const long bw = *reinterpret_cast<const long*>(
          &lds[cur][frow * F8V2_PITCH + s * 32 + fko]);
#pragma unroll
      for (int t = 0; t < TPW; ++t) {
        const int tile = wave + F8V2_NW * t;
        if (tile < MT) {
          const long ba = *reinterpret_cast<const long*>(
              &lds[cur][(16 + tile * 16 + frow) * F8V2_PITCH + s * 32 + fko]);
          acc[t] = MFMA16F8(ba, bw, acc[t]);
        }
      }
    }
    if (win + 1 < n_win) {
      __syncthreads();
      store_win(cur ^ 1);
      if (win + 2 < n_win) load_win(win + 2);
      __syncthreads();
    }
  }

  // store: each wave owns its tiles (waves shared k, no reduction)
  const int crow = (lane >> 4) * 4;
  const int n = n0 + (lane & 15);
  const float ws = (n < N) ? w_scale[n] : 0.f;
#pragma unroll
  for (int t = 0; t < TPW; ++t) {
    const int tile = wave + F8V2_NW * t;
    if (tile >= MT) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = tile * 16 + crow + r;
      if (m < M && n < N)
        c[(long)m * N + n] = __float2bfloat16(acc[t][r] * a_scale[m] * ws);
    }
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
#define DISPATCH_FLOAT_TYPES(TENSOR, NAME, ...)                              \
  [&] {                                                                     \
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

template <typename scalar_t>
void launch_rmsnorm_fp8(const torch::Tensor& x, const torch::Tensor& weight,
                        torch::Tensor& res, torch::Tensor& out8,
                        torch::Tensor& scales, int rows, int h, float eps) {
  constexpr int VE = 16 / sizeof(scalar_t);
  TORCH_CHECK(h % VE == 0, "rmsnorm_fp8: hidden % ", VE, " != 0");
  const int nvec = h / VE;
  int block = std::min(1024, ((nvec + 63) / 64) * 64);
  const int vecs = (nvec + block - 1) / block;
  const bool has_r = res.defined();
  hipStream_t stream_ = cmls::current_stream();
  const int smem = 16 * sizeof(float);
#define LRF(V_, HR)                                                          \
  hipLaunchKernelGGL((rmsnorm_fp8_kernel<scalar_t, V_, HR>), dim3(rows),     \
                     dim3(block), smem, stream_,                             \
                     (const scalar_t*)x.data_ptr(),                          \
                     (const scalar_t*)weight.data_ptr(),                     \
                     has_r ? (scalar_t*)res.data_ptr() : nullptr,            \
                     (unsigned char*)out8.data_ptr(),                        \
                     (float*)scales.data_ptr(), rows, h, eps)
#define LRF_ALL(V_) do { if (has_r) LRF(V_, true); else LRF(V_, false); } while (0)
  TORCH_CHECK(vecs <= 8, "rmsnorm_fp8: hidden too large");
  if (vecs <= 1) LRF_ALL(1);
  else if (vecs == 2) LRF_ALL(2);
  else if (vecs <= 4) LRF_ALL(4);
  else LRF_ALL(8);
#undef LRF_ALL
#undef LRF
}

std::vector<torch::Tensor> rmsnorm_fp8(torch::Tensor x, torch::Tensor weight,
                                       double eps,
                                       c10::optional<torch::Tensor> residual) {
  CHECK_LASTDIM_CONTIG(x);
  TORCH_CHECK(weight.is_contiguous());
  const int h = x.size(-1);
  const int rows = x.numel() / h;
  torch::Tensor res;
  if (residual.has_value() && residual->defined()) {
    res = *residual;
    TORCH_CHECK(res.is_contiguous() && res.sizes() == x.sizes());
  }
  auto out8 = torch::empty({rows, h},
                           x.options().dtype(torch::kUInt8));
  auto scales = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  DISPATCH_FLOAT_TYPES(x, "rmsnorm_fp8", [&] {
    launch_rmsnorm_fp8<scalar_t>(x, weight, res, out8, scales, rows, h,
                                 (float)eps);
  });
  return {out8, scales};
}

std::vector<torch::Tensor> silu_mul_fp8(torch::Tensor gate, torch::Tensor up) {
  TORCH_CHECK(gate.sizes() == up.sizes(), "silu_mul_fp8: shape mismatch");
  TORCH_CHECK(gate.stride(-1) == 1 && up.stride(-1) == 1);
  const int h = gate.size(-1);
  const int rows = gate.numel() / h;
  auto out8 = torch::empty({rows, h}, gate.options().dtype(torch::kUInt8));
  auto scales = torch::empty({rows}, gate.options().dtype(torch::kFloat32));
  hipStream_t stream_ = cmls::current_stream();
  const int smem = 16 * sizeof(float);
  DISPATCH_FLOAT_TYPES(gate, "silu_mul_fp8", [&] {
    constexpr int VE = 16 / sizeof(scalar_t);
    TORCH_CHECK(h % VE == 0, "silu_mul_fp8: width % ", VE, " != 0");
    const long gs = gate.dim() >= 2 ? gate.stride(-2) : h;
    const long us = up.dim() >= 2 ? up.stride(-2) : h;
    hipLaunchKernelGGL((silu_mul_fp8_kernel<scalar_t>), dim3(rows),
                       dim3(256), smem, stream_,
                       (const scalar_t*)gate.data_ptr(),
                       (const scalar_t*)up.data_ptr(),
                       (unsigned char*)out8.data_ptr(),
                       (float*)scales.data_ptr(), rows, h, gs, us);
  });
  return {out8, scales};
}

std::vector<torch::Tensor> quant_fp8(torch::Tensor x) {
  CHECK_LASTDIM_CONTIG(x);
  const int h = x.size(-1);
  const int rows = x.numel() / h;
  auto out8 = torch::empty({rows, h}, x.options().dtype(torch::kUInt8));
  auto scales = torch::empty({rows}, x.options().dtype(torch::kFloat32));
  hipStream_t stream_ = cmls::current_stream();
  const int smem = 16 * sizeof(float);
  DISPATCH_FLOAT_TYPES(x, "quant_fp8", [&] {
    constexpr int VE = 16 / sizeof(scalar_t);
    TORCH_CHECK(h % VE == 0, "quant_fp8: width % ", VE, " != 0");
    hipLaunchKernelGGL((quant_fp8_kernel<scalar_t>), dim3(rows), dim3(256),
                       smem, stream_, (const scalar_t*)x.data_ptr(),
                       (unsigned char*)out8.data_ptr(),
                       (float*)scales.data_ptr(), rows, h);
  });
  return {out8, scales};
}

torch::Tensor skinny_gemm_fp8(torch::Tensor a8, torch::Tensor a_scale,
                              torch::Tensor w8, torch::Tensor w_scale) {
  TORCH_CHECK(a8.dim() == 2 && w8.dim() == 2, "skinny_gemm_fp8: 2-D only");
  TORCH_CHECK(a8.scalar_type() == at::kByte ||
              a8.scalar_type() == at::kFloat8_e4m3fn);
  TORCH_CHECK(w8.scalar_type() == at::kByte ||
              w8.scalar_type() == at::kFloat8_e4m3fn);
  TORCH_CHECK(a8.is_contiguous() && w8.is_contiguous());
  TORCH_CHECK(a_scale.scalar_type() == at::kFloat &&
              w_scale.scalar_type() == at::kFloat);
  const int M = a8.size(0), K = a8.size(1), N = w8.size(0);
  TORCH_CHECK(w8.size(1) == K, "skinny_gemm_fp8: K mismatch");
  TORCH_CHECK(M >= 1 && M <= 64, "skinny_gemm_fp8: M must be 1..64");
  TORCH_CHECK(K % F8_KSTEP == 0, "skinny_gemm_fp8: K % 64 != 0");
  TORCH_CHECK(N % F8_NT == 0, "skinny_gemm_fp8: N % 16 != 0");
  TORCH_CHECK(a_scale.numel() == M && w_scale.numel() == N);
  auto c = torch::empty({M, N},
                        a8.options().dtype(torch::kBFloat16));
  hipStream_t stream_ = cmls::current_stream();
  dim3 grid(N / F8_NT);
  const int mt = (M + 15) / 16;
#define LAUNCH_SKF8(T)                                                       \
  hipLaunchKernelGGL((skinny_gemm_fp8_kernel<T>), grid, dim3(F8_NW * 64), 0, \
                     stream_, (const unsigned char*)a8.data_ptr(),           \
                     (const float*)a_scale.data_ptr(),                       \
                     (const unsigned char*)w8.data_ptr(),                    \
                     (const float*)w_scale.data_ptr(),                       \
                     (__hip_bfloat16*)c.data_ptr(), M, N, K)
  switch (mt) {
    case 1: LAUNCH_SKF8(1); break;
    case 2: LAUNCH_SKF8(2); break;
    case 3: LAUNCH_SKF8(3); break;
    default: LAUNCH_SKF8(4); break;
  }
#undef LAUNCH_SKF8
  return c;
}

torch::Tensor skinny_gemm_fp8_v2(torch::Tensor a8, torch::Tensor a_scale,
                                 torch::Tensor w8, torch::Tensor w_scale) {
  // LDS-staged variant for M 17..64; w8 is the PLAIN fp8 layout
  TORCH_CHECK(a8.dim() == 2 && w8.dim() == 2);
  TORCH_CHECK(a8.is_contiguous() && w8.is_contiguous());
  TORCH_CHECK(a_scale.scalar_type() == at::kFloat &&
              w_scale.scalar_type() == at::kFloat);
  const int M = a8.size(0), K = a8.size(1), N = w8.size(0);
  TORCH_CHECK(w8.size(1) == K);
  TORCH_CHECK(M >= 1 && M <= 64, "skinny_gemm_fp8_v2: M must be 1..64");
  TORCH_CHECK(K % F8V2_KW == 0, "skinny_gemm_fp8_v2: K % 128 != 0");
  TORCH_CHECK(N % F8_NT == 0);
  TORCH_CHECK(a_scale.numel() == M && w_scale.numel() == N);
  auto c = torch::empty({M, N}, a8.options().dtype(torch::kBFloat16));
  hipStream_t stream_ = cmls::current_stream();
  dim3 grid(N / F8_NT);
  const int mt = (M + 15) / 16;
#define LAUNCH_SKF8V2(T)                                                     \
  hipLaunchKernelGGL((skinny_gemm_fp8_v2_kernel<T>), grid,                   \
                     dim3(F8V2_NW * 64), 0, stream_,                         \
                     (const unsigned char*)a8.data_ptr(),                    \
                     (const float*)a_scale.data_ptr(),                       \
                     (const unsigned char*)w8.data_ptr(),                    \
                     (const float*)w_scale.data_ptr(),                       \
                     (__hip_bfloat16*)c.data_ptr(), M, N, K)
  switch (mt) {
    case 1: LAUNCH_SKF8V2(1); break;
    case 2: LAUNCH_SKF8V2(2); break;
    case 3: LAUNCH_SKF8V2(3); break;
    default: LAUNCH_SKF8V2(4); break;
  }
#undef LAUNCH_SKF8V2
  return c;
}
#endif  // CMLS_KERNEL_ONLY
