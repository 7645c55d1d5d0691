// Skinny (decode-shaped) GEMM for gfx950: C[M, N] = A[M, K] @ W[N, K]^T,
// bf16 in / bf16 out, fp32 MFMA accumulation, M <= 128.
//
// Why it exists (measured, profiles/skinny_probe_hipblaslt.txt): at llama
// decode shapes hipBLASLt runs the small-N projections at 1.8-2.8 TB/s
// (o_proj N=4096: 28% of HBM peak) because its tile heuristics launch only
// N/256 ~ 16-24 workgroups on a 256-CU chip. The reference outsources this
// to vLLM/Triton; MI355X-native means owning the shape.
//
// Design: one workgroup per 16-column slice of W, so N=4096 fills 256 CUs.
// NW waves split K internally (deterministic LDS reduce at the end, no
// atomics -- greedy decoding must be reproducible). Both MFMA operands are
// 16-byte contiguous per lane straight from global: A rows ride in L2
// (every workgroup reads the same M*K tile), W streams from HBM exactly
// once -- no LDS staging on the hot path.
//   mfma_f32_16x16x32_bf16 fragment map (guide §4): a/b lane l holds
//   row (l%16), k = (l>>4)*8 + e; C lane l holds col (l&15),
//   row (l>>4)*4 + r.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA16(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int NT = 16;   // N columns per workgroup
constexpr int NW = 8;    // waves (split K)
constexpr int KSTEP = 32;

// MT = number of 16-row M tiles (1..8 -> M <= 128)
template <int MT>
__global__ __launch_bounds__(NW * 64, 2) void skinny_gemm_kernel(
    const __hip_bfloat16* __restrict__ a,  // [M, K] row-major (lda = K)
    const __hip_bfloat16* __restrict__ w,  // [N, K] row-major (Linear weight)
    __hip_bfloat16* __restrict__ c,        // [M, N] row-major
    int M, int N, int K) {
  const int n0 = blockIdx.x * NT;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row = lane & 15;          // fragment row (m or n within tile)
  const int koff = (lane >> 4) * 8;   // fragment k offset within KSTEP

  // wave's K range (K % (NW*KSTEP) handled by the last wave's bound)
  const int kchunk = ((K / KSTEP + NW - 1) / NW) * KSTEP;
  const int k_lo = wave * kchunk;
  const int k_hi = min(K, k_lo + kchunk);

  f32x4_t acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const __hip_bfloat16* wrow = w + (long)(n0 + row) * K + koff;
  // A row for each m tile this lane contributes to
  const __hip_bfloat16* arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
    const int m = t * 16 + row;
    // clamp: rows >= M replay row 0 (their C never stores)
    arow[t] = a + (long)(m < M ? m : 0) * K + koff;
  }

  // 2-deep unrolled K loop: up to 2*(MT+1) 16-byte loads in flight
  int k = k_lo;
  for (; k + 2 * KSTEP <= k_hi; k += 2 * KSTEP) {
    bf16x8_t bw0 = *reinterpret_cast<const bf16x8_t*>(wrow + k);
    bf16x8_t bw1 = *reinterpret_cast<const bf16x8_t*>(wrow + k + KSTEP);
    bf16x8_t ba0[MT], ba1[MT];
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      ba0[t] = *reinterpret_cast<const bf16x8_t*>(arow[t] + k);
      ba1[t] = *reinterpret_cast<const bf16x8_t*>(arow[t] + k + KSTEP);
    }
#pragma unroll
    for (int t = 0; t < MT; ++t) acc[t] = MFMA16(ba0[t], bw0, acc[t]);
#pragma unroll
    for (int t = 0; t < MT; ++t) acc[t] = MFMA16(ba1[t], bw1, acc[t]);
  }
  for (; k < k_hi; k += KSTEP) {
    bf16x8_t bw = *reinterpret_cast<const bf16x8_t*>(wrow + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      bf16x8_t ba = *reinterpret_cast<const bf16x8_t*>(arow[t] + k);
      acc[t] = MFMA16(ba, bw, acc[t]);
    }
  }

  // deterministic cross-wave K reduction through LDS:
  // lds[wave][m(16*MT)][n(16)]; then wave 0 sums in a fixed order
  __shared__ float lds[NW][MT * 16][NT];
  const int crow = (lane >> 4) * 4;  // C fragment rows
#pragma unroll
  for (int t = 0; t < MT; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r)
      lds[wave][t * 16 + crow + r][lane & 15] = acc[t][r];
  }
  __syncthreads();
  if (wave == 0) {
    // 64 lanes cover the 16x16 x MT output: lane -> (m_sub, n)
    const int n = lane & 15;
    const int m_base = (lane >> 4) * 4;
#pragma unroll
    for (int t = 0; t < MT; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mrow = t * 16 + m_base + r;
        float s = 0.f;
#pragma unroll
        for (int wv = 0; wv < NW; ++wv) s += lds[wv][mrow][n];
        const int m = mrow;
        if (m < M && n0 + n < N)
          c[(long)m * N + n0 + n] = __float2bfloat16(s);
      }
    }
  }
}

}  // namespace

// ------------------------------------------------------------------- //
// v2: LDS-staged variant for M >= 32. v1's direct fragment loads touch
// one 128-byte line per FOUR lanes (rows are K*2 bytes apart), so its
// line-request rate scales with m-tiles and collapses at M >= 64
// (measured: o_proj M=64 1.6 TB/s). Here all waves share one k
// progression; a cooperative flat loop stages the W[16, 64] and
// A[M16, 64] k-windows into LDS with consecutive lanes covering FULL
// lines, and the MFMA fragments come from LDS (row pitch padded 64->72
// elements so the 16 fragment rows land in 16 distinct banks).
// ------------------------------------------------------------------- //
namespace {

constexpr int KW = 64;        // k elements per staged window (128 B/row)
constexpr int PITCH = 72;     // LDS row pitch (elements): 64 + 8 pad
constexpr int V2_NW = 4;      // waves

template <int MT>
__global__ __launch_bounds__(V2_NW * 64, 2) void skinny_gemm_v2_kernel(
    const __hip_bfloat16* __restrict__ a,  // [M, K]
    const __hip_bfloat16* __restrict__ w,  // [N, K]
    __hip_bfloat16* __restrict__ c,        // [M, N]
    int M, int N, int K) {
  constexpr int M16 = MT * 16;
  constexpr int ROWS = 16 + M16;            // W rows then A rows
  const int n0 = blockIdx.x * NT;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  __shared__ __hip_bfloat16 lds[2][ROWS * PITCH];

  // accumulators: up to 2 m-tiles per wave (MT=8 with 4 waves)
  constexpr int TPW = (MT + V2_NW - 1) / V2_NW;
  f32x4_t acc[TPW];
#pragma unroll
  for (int t = 0; t < TPW; ++t) acc[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int n_win = K / KW;

  // register-staged software pipeline: issue the NEXT window's global
  // loads (into registers) before computing the current one, store them
  // to LDS only after the compute -- the loads' latency overlaps the
  // MFMA work instead of blocking an immediate LDS store.
  // Each thread owns SEGS 16-byte segments of the (ROWS x 128 B) window;
  // consecutive threads read consecutive 16 B -> full-line coalescing.
  constexpr int SEGS = (ROWS * 8 + V2_NW * 64 - 1) / (V2_NW * 64);
  bf16x8_t regs[SEGS];

  auto load_win = [&](int win) {
    const long kbase = (long)win * KW;
#pragma unroll
    for (int sgi = 0; sgi < SEGS; ++sgi) {
      const int i = tid + sgi * V2_NW * 64;
      if (i >= ROWS * 8) break;
      const int row = i >> 3;
      const int seg = (i & 7) * 8;
      const __hip_bfloat16* src =
          row < 16 ? w + (long)(n0 + row) * K + kbase + seg
                   : a + (long)min(row - 16, M - 1) * K + kbase + seg;
      regs[sgi] = *reinterpret_cast<const bf16x8_t*>(src);
    }
  };
  auto store_win = [&](int buf) {
#pragma unroll
    for (int sgi = 0; sgi < SEGS; ++sgi) {
      const int i = tid + sgi * V2_NW * 64;
      if (i >= ROWS * 8) break;
      *reinterpret_cast<bf16x8_t*>(
          &lds[buf][(i >> 3) * PITCH + (i & 7) * 8]) = regs[sgi];
    }
  };

  load_win(0);
  store_win(0);
  if (n_win > 1) load_win(1);
  __syncthreads();
  for (int win = 0; win < n_win; ++win) {
    const int cur = win & 1;
    // two mfma k-steps per window
    const int frow = lane & 15;
    const int fkoff = (lane >> 4) * 8;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      const bf16x8_t bw = *reinterpret_cast<const bf16x8_t*>(
          &lds[cur][frow * PITCH + s * 32 + fkoff]);
#pragma unroll
      for (int t = 0; t < TPW; ++t) {
        const int tile = wave + V2_NW * t;
        if (tile < MT) {
          const bf16x8_t ba = *reinterpret_cast<const bf16x8_t*>(
              &lds[cur][(16 + tile * 16 + frow) * PITCH + s * 32 + fkoff]);
          acc[t] = MFMA16(ba, bw, acc[t]);
        }
      }
    }
    if (win + 1 < n_win) {
      __syncthreads();        // everyone done reading buf[cur^1]
      store_win(cur ^ 1);     // regs of window win+1 -> LDS
      if (win + 2 < n_win) load_win(win + 2);
      __syncthreads();        // buf[cur^1] ready for next iteration
    }
  }

  // store: each wave owns its tiles outright (no cross-wave reduction --
  // waves shared k, they did not split it)
  const int crow = (lane >> 4) * 4;
#pragma unroll
  for (int t = 0; t < TPW; ++t) {
    const int tile = wave + V2_NW * t;
    if (tile >= MT) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = tile * 16 + crow + r;
      if (m < M && n0 + (lane & 15) < N)
        c[(long)m * N + n0 + (lane & 15)] = __float2bfloat16(acc[t][r]);
    }
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w,
                          int64_t variant) {
  TORCH_CHECK(a.dim() == 2 && w.dim() == 2, "skinny_gemm: 2-D only");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "skinny_gemm: bf16 only");
  TORCH_CHECK(a.stride(1) == 1 && a.stride(0) == a.size(1),
              "skinny_gemm: A must be row-major contiguous");
  TORCH_CHECK(w.is_contiguous(), "skinny_gemm: W must be contiguous");
  const int M = a.size(0), K = a.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "skinny_gemm: K mismatch");
  TORCH_CHECK(M >= 1 && M <= 128, "skinny_gemm: M must be 1..128");
  TORCH_CHECK(K % KSTEP == 0, "skinny_gemm: K % 32 != 0");
  TORCH_CHECK(N % NT == 0, "skinny_gemm: N % 16 != 0");
  auto c = torch::empty({M, N}, a.options());
  hipStream_t stream_ = cmls::current_stream();
  dim3 grid(N / NT);
  const int mt = (M + 15) / 16;
  // variant: 0 = auto (v1 direct loads for small M, v2 LDS-staged above),
  // 1 = force v1, 2 = force v2 (A/B harnesses)
  const bool v2 = (variant == 2) ||
                  (variant == 0 && mt >= 2 && (K % KW == 0));
  if (v2) {
    TORCH_CHECK(K % KW == 0, "skinny_gemm v2: K % 64 != 0");
#define LAUNCH_SK2(T)                                                       \
  hipLaunchKernelGGL((skinny_gemm_v2_kernel<T>), grid, dim3(V2_NW * 64), 0, \
                     stream_, (const __hip_bfloat16*)a.data_ptr(),          \
                     (const __hip_bfloat16*)w.data_ptr(),                   \
                     (__hip_bfloat16*)c.data_ptr(), M, N, K)
    switch (mt) {
      case 1: LAUNCH_SK2(1); break;
      case 2: LAUNCH_SK2(2); break;
      case 3: LAUNCH_SK2(3); break;
      case 4: LAUNCH_SK2(4); break;
      case 5: LAUNCH_SK2(5); break;
      case 6: LAUNCH_SK2(6); break;
      case 7: LAUNCH_SK2(7); break;
      default: LAUNCH_SK2(8); break;
    }
#undef LAUNCH_SK2
    return c;
  }
#define LAUNCH_SK(T)                                                        \
  hipLaunchKernelGGL((skinny_gemm_kernel<T>), grid, dim3(NW * 64), 0,       \
                     stream_, (const __hip_bfloat16*)a.data_ptr(),          \
                     (const __hip_bfloat16*)w.data_ptr(),                   \
                     (__hip_bfloat16*)c.data_ptr(), M, N, K)
  switch (mt) {
    case 1: LAUNCH_SK(1); break;
    case 2: LAUNCH_SK(2); break;
    case 3: LAUNCH_SK(3); break;
    case 4: LAUNCH_SK(4); break;
    case 5: LAUNCH_SK(5); break;
    case 6: LAUNCH_SK(6); break;
    case 7: LAUNCH_SK(7); break;
    default: LAUNCH_SK(8); break;
  }
#undef LAUNCH_SK
  return c;
}
#endif  // CMLS_KERNEL_ONLY
