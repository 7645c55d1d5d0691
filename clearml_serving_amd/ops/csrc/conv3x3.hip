// Hand-written 3x3 stride-1 pad-1 NHWC bf16 convolution (implicit GEMM)
// for the ResNet-50 bottleneck shapes -- the one conv class that carries
// the bulk of ResNet FLOPs. Closes the SURVEY §2.6 "GEMM/conv hand-written"
// row: round 1 delegated conv to MIOpen (models/resnet.py); this kernel is
// routed wherever it MEASURES faster (same discipline as skinny_linear).
//
// Implicit GEMM view: C[M=N*H*W, K] = sum over 9 taps of
//   A_tap[M, C] @ W[K, tap, C]^T
// Tiling (per workgroup, 4 waves):
//   output tile = TY rows x WIDTH cols x NT output channels
//     (TY*WIDTH = 112 = 7 m-tiles of 16 for WIDTH in {56, 28, 14})
//   input window (TY+2) x (WIDTH+2) x 32c staged in LDS per c-chunk,
//     pitch-padded to 40 elements so the 16 fragment rows hit 16 distinct
//     banks (stride 80 B = 20 dwords, coprime with 64 banks)
//   weights [K][3][3][C] (channels_last conv weight layout) read directly
//     from global per tap -- every workgroup reads the same 16-byte lanes,
//     so they ride in L2
// Grid = (y_tiles, batch, K/NT): e.g. conv3 (28x28x128->128, batch 64)
// launches 7 * 64 = 448 workgroups on 256 CUs.
//   mfma_f32_16x16x32_bf16 maps as in skinny_gemm.hip: a/b lane l holds
//   row l%16, k = (l>>4)*8 + e; C lane l holds col lane&15 (k-channel),
//   row (l>>4)*4 + r (m).
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA16(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int CSTEP = 32;   // c-chunk (one MFMA k step)
constexpr int PITCH = 40;   // LDS pitch in elements (80 B: bank-conflict-free
                            // fragment rows, 16-B aligned rows)
constexpr int NWAVES = 4;

// WIDTH: spatial width (=height); TY: output rows per workgroup
// (TY*WIDTH == 112); NTPW: 16-col n-tiles per wave (NT = NTPW*NWAVES*16)
template <int WIDTH, int TY, int NTPW, bool RELU>
__global__ __launch_bounds__(NWAVES * 64, 2) void conv3x3_kernel(
    const __hip_bfloat16* __restrict__ x,   // [N, H, W, C]
    const __hip_bfloat16* __restrict__ w,   // [K, 3, 3, C]
    const __hip_bfloat16* __restrict__ bias,  // [K] or nullptr
    const __hip_bfloat16* __restrict__ residual,  // [N, H, W, K] or null
    __hip_bfloat16* __restrict__ out,       // [N, H, W, K]
    int C, int K) {
  constexpr int MT = 7;              // 112 outputs = 7 m-tiles
  constexpr int IROWS = TY + 2;
  constexpr int ICOLS = WIDTH + 2;
  const int y0 = blockIdx.x * TY;    // first output row
  const int n_img = blockIdx.y;
  const int k0 = blockIdx.z * (NTPW * NWAVES * 16);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // double-buffered input window, register-staged (T14 issue-early /
  // write-late): chunk c+1's global loads are issued before computing on
  // chunk c, so HBM latency hides under the 9-tap MFMA work
  __shared__ __hip_bfloat16 lds[2][IROWS * ICOLS * PITCH];

  f32x4_t acc[MT][NTPW];
#pragma unroll
  for (int t = 0; t < MT; ++t)
#pragma unroll
    for (int nt = 0; nt < NTPW; ++nt) acc[t][nt] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const long in_base = (long)n_img * WIDTH * WIDTH * C;
  const int frow = lane & 15;          // fragment row (m or n within tile)
  const int fk = (lane >> 4) * 8;      // fragment k offset within CSTEP

  constexpr int SITES = IROWS * ICOLS;
  constexpr int SEGS_PER_SITE = CSTEP / 8;  // 4
  constexpr int TOTAL = SITES * SEGS_PER_SITE;
  constexpr int SREGS = (TOTAL + NWAVES * 64 - 1) / (NWAVES * 64);
  bf16x8_t sreg[SREGS];

  // each thread covers 16-byte segments (8 channels); consecutive
  // threads -> consecutive segments of one (iy, ix) site
  auto load_chunk = [&](int c0) {
#pragma unroll
    for (int r = 0; r < SREGS; ++r) {
      const int i = tid + r * NWAVES * 64;
      if (i >= TOTAL) break;
      const int site = i / SEGS_PER_SITE;
      const int seg = (i % SEGS_PER_SITE) * 8;
      const int iy = site / ICOLS;         // 0..IROWS-1
      const int ix = site % ICOLS;         // 0..ICOLS-1
      const int gy = y0 + iy - 1;
      const int gx = ix - 1;
      if (gy >= 0 && gy < WIDTH && gx >= 0 && gx < WIDTH) {
        sreg[r] = *reinterpret_cast<const bf16x8_t*>(
            x + in_base + ((long)gy * WIDTH + gx) * C + c0 + seg);
      } else {
        sreg[r] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };
  auto write_chunk = [&](int buf) {
#pragma unroll
    for (int r = 0; r < SREGS; ++r) {
      const int i = tid + r * NWAVES * 64;
      if (i >= TOTAL) break;
      const int site = i / SEGS_PER_SITE;
      const int seg = (i % SEGS_PER_SITE) * 8;
      *reinterpret_cast<bf16x8_t*>(
          &lds[buf][(long)site * PITCH + seg]) = sreg[r];
    }
  };

  load_chunk(0);
  write_chunk(0);
  if (CSTEP < C) load_chunk(CSTEP);
  __syncthreads();
  int cur = 0;
  for (int c0 = 0; c0 < C; c0 += CSTEP) {
    // ---- 9 taps x 7 m-tiles x NTPW n-tiles ---------------------------- //
#pragma unroll
    for (int dy = 0; dy < 3; ++dy) {
#pragma unroll
      for (int dx = 0; dx < 3; ++dx) {
        // B fragments for this tap (global; L2-resident across WGs)
        bf16x8_t bw[NTPW];
#pragma unroll
        for (int nt = 0; nt < NTPW; ++nt) {
          const int k = k0 + (wave * NTPW + nt) * 16 + frow;
          bw[nt] = *reinterpret_cast<const bf16x8_t*>(
              w + ((long)k * 9 + dy * 3 + dx) * C + c0 + fk);
        }
#pragma unroll
        for (int t = 0; t < MT; ++t) {
          const int m = t * 16 + frow;       // 0..111
          const int oy = m / WIDTH;          // 0..TY-1
          const int ox = m % WIDTH;
          const int site = (oy + dy) * ICOLS + (ox + dx);
          const bf16x8_t ba = *reinterpret_cast<const bf16x8_t*>(
              &lds[cur][(long)site * PITCH + fk]);
#pragma unroll
          for (int nt = 0; nt < NTPW; ++nt)
            acc[t][nt] = MFMA16(ba, bw[nt], acc[t][nt]);
        }
      }
    }
    if (c0 + CSTEP < C) {
      __syncthreads();        // all waves done reading lds[cur ^ 1]
      write_chunk(cur ^ 1);   // regs of chunk c0+CSTEP -> LDS
      if (c0 + 2 * CSTEP < C) load_chunk(c0 + 2 * CSTEP);
      __syncthreads();        // buf ready for the next iteration
      cur ^= 1;
    }
  }

  // ---- epilogue: bias (+residual) (+relu), NHWC store ----------------- //
  const int ccol = lane & 15;            // k within n-tile
  const int crow4 = (lane >> 4) * 4;     // m fragment rows
#pragma unroll
  for (int nt = 0; nt < NTPW; ++nt) {
    const int k = k0 + (wave * NTPW + nt) * 16 + ccol;
    const float b = bias ? to_f32(bias[k]) : 0.f;
#pragma unroll
    for (int t = 0; t < MT; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = t * 16 + crow4 + r;
        const int oy = m / WIDTH;
        if (y0 + oy >= WIDTH) continue;  // grid ceil: partial last y-tile
        const int ox = m % WIDTH;
        const long off =
            (((long)n_img * WIDTH + (y0 + oy)) * WIDTH + ox) * K + k;
        float v = acc[t][nt][r] + b;
        if (residual) v += to_f32(residual[off]);
        if (RELU) v = fmaxf(v, 0.f);
        out[off] = __float2bfloat16(v);
      }
    }
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
torch::Tensor conv3x3_nhwc(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> bias, bool relu,
                           c10::optional<torch::Tensor> residual) {
  // x: channels_last [N, C, H, W] (memory NHWC); w: channels_last conv
  // weight [K, C, 3, 3] (memory [K, 3, 3, C])
  TORCH_CHECK(x.dim() == 4 && w.dim() == 4, "conv3x3: 4-D tensors");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "conv3x3: bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3: x must be channels_last");
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "conv3x3: w must be channels_last");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(0);
  TORCH_CHECK(w.size(1) == C && w.size(2) == 3 && w.size(3) == 3,
              "conv3x3: weight must be [K, C, 3, 3]");
  TORCH_CHECK(H == W, "conv3x3: square inputs only");
  TORCH_CHECK(C % CSTEP == 0, "conv3x3: C % 32 != 0");
  TORCH_CHECK(W == 56 || W == 28 || W == 14,
              "conv3x3: supported widths 56/28/14 (ResNet-50 shapes)");

  auto out = torch::empty_strided(
      {N, K, H, W},
      {(long)H * W * K, 1, (long)W * K, K}, x.options());
  const __hip_bfloat16* bp = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->is_contiguous() && bias->numel() == K);
    bp = (const __hip_bfloat16*)bias->data_ptr();
  }
  const __hip_bfloat16* rp = nullptr;
  if (residual.has_value() && residual->defined()) {
    TORCH_CHECK(residual->is_contiguous(at::MemoryFormat::ChannelsLast));
    TORCH_CHECK(residual->sizes() == out.sizes());
    rp = (const __hip_bfloat16*)residual->data_ptr();
  }
  hipStream_t stream_ = cmls::current_stream();

  const int TY = 112 / W;
  const int NT = (K % 128 == 0) ? 128 : 64;
  TORCH_CHECK(K % NT == 0, "conv3x3: K % 64 != 0");
  dim3 grid((H + TY - 1) / TY, N, K / NT);
  dim3 block(NWAVES * 64);

#define LAUNCH_CONV(W_, TY_, NTPW_, RELU_)                                  \
  hipLaunchKernelGGL((conv3x3_kernel<W_, TY_, NTPW_, RELU_>), grid, block,  \
                     0, stream_, (const __hip_bfloat16*)x.data_ptr(),       \
                     (const __hip_bfloat16*)w.data_ptr(), bp, rp,           \
                     (__hip_bfloat16*)out.data_ptr(), C, K)
#define LAUNCH_CONV_W(W_, TY_)                                              \
  do {                                                                      \
    if (NT == 128) {                                                        \
      if (relu) LAUNCH_CONV(W_, TY_, 2, true);                              \
      else LAUNCH_CONV(W_, TY_, 2, false);                                  \
    } else {                                                                \
      if (relu) LAUNCH_CONV(W_, TY_, 1, true);                              \
      else LAUNCH_CONV(W_, TY_, 1, false);                                  \
    }                                                                       \
  } while (0)
  if (W == 56) LAUNCH_CONV_W(56, 2);
  else if (W == 28) LAUNCH_CONV_W(28, 4);
  else LAUNCH_CONV_W(14, 8);
#undef LAUNCH_CONV_W
#undef LAUNCH_CONV
  return out;
}
#endif  // CMLS_KERNEL_ONLY
