// bf16 MFMA GEMM for gfx950: C[M,N] = A[M,K] @ B[N,K]^T, fp32 accumulate.
//
// Linear-layer orientation (torch nn.Linear weights are [out,in] = [N,K]):
// both operands are K-major, so every MFMA fragment read is 8 consecutive
// bf16 (one ds_read_b128).
//
// Structure (cdna guide §5 "step-3" anatomy):
//   tile 128x128, BK=32, 4 waves as 2x2 (each computes 64x64 via 4x4
//   mfma_f32_16x16x32_bf16 fragments), double-buffered LDS staged with
//   global_load_lds (16-B direct-to-LDS DMA).
//   glds writes lane-linearly, so bank-conflict avoidance XOR-swizzles the
//   per-lane global SOURCE address and the LDS read with the same key
//   ((row>>2)&3 on byte bits 4-5): fragment reads become conflict-free and
//   the source permutation stays inside each row's 64-B segment (coalescing
//   unaffected).
//
// This is the in-tree GEMM of the kernel library (SURVEY.md §2.6 inventory);
// plain library GEMMs in the serving models go through hipBLASLt
// (torch.matmul), which the dispatcher uses whenever it wins.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA_16x16x32(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int TILE_BYTES = BM * BK * 2;  // 8 KiB per operand per buffer

__device__ __forceinline__ int swz_group(int row, int g) {
  return g ^ ((row >> 2) & 3);  // byte-group permutation key
}

__global__ __launch_bounds__(256, 2) void gemm_bf16_kernel(
    const __hip_bfloat16* __restrict__ A,  // [M, K]
    const __hip_bfloat16* __restrict__ B,  // [N, K]
    __hip_bfloat16* __restrict__ C,        // [M, N]
    int M, int N, int K) {
  __shared__ short lds_a[2][BM * BK];
  __shared__ short lds_b[2][BN * BK];

  const int tiles_n = N / BN;
  // bijective XCD remap (guide T1): the dispatcher places block b on XCD
  // b%8, so give each XCD a CONTIGUOUS chunk of the N-first walk -- then
  // neighbor tiles sharing A/B panels hit the same XCD's L2
  const int nwg = gridDim.x;
  const int xcd = blockIdx.x % 8;
  const int idx = blockIdx.x / 8;
  const int q8 = nwg / 8, r8 = nwg % 8;
  const int bid = (xcd < r8 ? xcd * (q8 + 1)
                            : r8 * (q8 + 1) + (xcd - r8) * q8) + idx;
  const int bm = bid / tiles_n;
  const int bn = bid % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = (wave >> 1) * 64;  // wave row offset in tile
  const int wn = (wave & 1) * 64;

  const int frag_i = lane & 15;
  const int frag_g = lane >> 4;           // k byte-group (8 elems = 16 B)

  // staging geometry: each thread DMAs one 16-B piece per 64 rows
  const int st_row = tid >> 2;            // 0..63
  const int st_g = tid & 3;               // byte-group 0..3

  const long lda = K, ldb = K;

  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * BK;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int row = st_row + half * 64;
      {  // A rows bm*128+row
        const int gsrc = swz_group(row, st_g);
        const __hip_bfloat16* src =
            A + ((long)bm * BM + row) * lda + k0 + gsrc * 8;
        short* dst = &lds_a[buf][row * BK + st_g * 8];
        __builtin_amdgcn_global_load_lds(
            reinterpret_cast<const unsigned int*>(src),
            reinterpret_cast<unsigned int*>(dst), 16, 0, 0);
      }
      {  // B rows bn*128+row
        const int gsrc = swz_group(row, st_g);
        const __hip_bfloat16* src =
            B + ((long)bn * BN + row) * ldb + k0 + gsrc * 8;
        short* dst = &lds_b[buf][row * BK + st_g * 8];
        __builtin_amdgcn_global_load_lds(
            reinterpret_cast<const unsigned int*>(src),
            reinterpret_cast<unsigned int*>(dst), 16, 0, 0);
      }
    }
  };

  f32x4_t acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int ntiles = K / BK;
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = 0; kt < ntiles; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < ntiles) stage(cur ^ 1, kt + 1);

    // fragment reads (swizzled): lane reads row, k-group frag_g
    bf16x8_t a_frag[4], b_frag[4];
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      const int row = wm + m * 16 + frag_i;
      a_frag[m] = *reinterpret_cast<const bf16x8_t*>(
          &lds_a[cur][row * BK + swz_group(row, frag_g) * 8]);
    }
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int row = wn + n * 16 + frag_i;
      b_frag[n] = *reinterpret_cast<const bf16x8_t*>(
          &lds_b[cur][row * BK + swz_group(row, frag_g) * 8]);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        acc[m][n] = MFMA_16x16x32(a_frag[m], b_frag[n], acc[m][n]);
    __builtin_amdgcn_s_setprio(0);

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // epilogue: C layout row = (lane>>4)*4 + reg, col = lane&15
  const int crow0 = bm * BM + wm + (lane >> 4) * 4;
  const int ccol0 = bn * BN + wn + (lane & 15);
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = crow0 + m * 16 + r;
      __hip_bfloat16* dst = C + row * (long)N + ccol0;
#pragma unroll
      for (int n = 0; n < 4; ++n)
        dst[n * 16] = __float2bfloat16(acc[m][n][r]);
    }
  }
}

}  // namespace

torch::Tensor gemm_bf16(torch::Tensor a, torch::Tensor b) {
  // C = a @ b^T: a [M, K], b [N, K] (nn.Linear orientation)
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "a [M,K], b [N,K]");
  TORCH_CHECK(a.scalar_type() == at::kBFloat16 &&
              b.scalar_type() == at::kBFloat16, "bf16 only");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "inner dims differ");
  TORCH_CHECK(M % BM == 0 && N % BN == 0 && K % BK == 0,
              "gemm_bf16 needs M%", BM, "==0, N%", BN, "==0, K%", BK,
              "==0 (pad in the caller)");
  auto c = torch::empty({M, N}, a.options());
  dim3 grid((M / BM) * (N / BN));
  hipStream_t stream_ = cmls::current_stream();
  hipLaunchKernelGGL(gemm_bf16_kernel, grid, dim3(256), 0, stream_,
                     (const __hip_bfloat16*)a.data_ptr(),
                     (const __hip_bfloat16*)b.data_ptr(),
                     (__hip_bfloat16*)c.data_ptr(), M, N, K);
  return c;
}
