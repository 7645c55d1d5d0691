// Fused attention prefill v2: swapped-QK^T in-register softmax (guide T12).
//
// v1 (attention.hip) computes S = mfma(Q, K) whose C layout spreads each
// softmax ROW over 16 lanes, so P must round-trip through LDS (scatter
// write + lgkmcnt(0) drain + b128 re-read) to become the PV A-fragment --
// PMC showed ~20% of wall in that traffic + 10.5x VALU:MFMA.
//
// v2 swaps the MFMA operands: S^T = mfma(K, Q). The per-lane data of the Q
// and K fragments is IDENTICAL (A[i][k] and B[k][j] have the same lane
// map), but the C layout becomes [key][q]: each lane's 16 S values share
// ONE query column, so the softmax max/sum are 2 shfl_xor ops (groups) and
// the running (m, l) state is a lane-local scalar. P^T re-shapes into the
// PV B-fragment with 4 ds_bpermute lane moves per 16-key tile (no LDS, no
// drain); O accumulates TRANSPOSED (O^T = mfma(V^T, P^T)) and the epilogue
// stores 4-element d-contiguous pieces per lane.
//
// Everything else matches v1: grid/causal dispatch order, T14 issue-early /
// write-late K/V staging, exp2-domain softmax, tile-full mask skip, GQA,
// seq_lens, chunked-prefill q_lens, paged KV.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA_16x16x32(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int BLOCK_M = 128;
constexpr int BLOCK_N = 64;
constexpr int NWAVES = 8;
constexpr int PAD = 8;

struct AttnStrides2 {
  long qb, qh, qs;
  long kb, kh, ks;
  long ob, oh, os;
};

// pack 2 f32 -> u32 of 2 bf16 (RNE via compiler)
__device__ __forceinline__ unsigned int pack_bf16x2(float lo, float hi) {
  __hip_bfloat16 a = __float2bfloat16(lo), b = __float2bfloat16(hi);
  return (unsigned int)*reinterpret_cast<unsigned short*>(&a) |
         ((unsigned int)*reinterpret_cast<unsigned short*>(&b) << 16);
}

// KDIRECT: skip K LDS staging entirely -- QK^T A-fragments are 16-byte
// contiguous per lane in the GLOBAL K layout already, and K tiles are
// re-read by every m-tile workgroup of the same (b,h), so after the first
// touch they come from L2. Halves the LDS footprint and the ds traffic.
// Dense path only (the paged path needs per-row block-table lookups).
typedef __attribute__((ext_vector_type(2))) float av2_f32x2_t;

typedef __attribute__((ext_vector_type(2))) __bf16 av2_bf16x2_t;

// 4 e4m3 bytes -> 2 packed-bf16 pairs with the scale folded into the
// conversion (CDNA4 v_cvt_scalef32_pk_bf16_fp8; word-select is literal)
__device__ __forceinline__ void v2_fp8x4_to_bf16(int w, float sc, short* o) {
  av2_bf16x2_t lo = __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(w, sc, false);
  av2_bf16x2_t hi = __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(w, sc, true);
  *reinterpret_cast<av2_bf16x2_t*>(o) = lo;
  *reinterpret_cast<av2_bf16x2_t*>(o + 2) = hi;
}

template <int HEAD_DIM, bool CAUSAL, bool HAS_SEQLENS, bool PAGED = false,
          bool KDIRECT = false, bool PIPE = false, bool FP8KV = false>
__global__ __launch_bounds__(NWAVES * 64, 2) void attn_prefill_v2_kernel(
    const __hip_bfloat16* __restrict__ q,
    const void* __restrict__ kvoid,   // bf16, or e4m3 bytes when FP8KV
    const void* __restrict__ vvoid,
    const float* __restrict__ k_scale,  // [NB, Hkv, BS] (FP8KV paged)
    const float* __restrict__ v_scale,
    __hip_bfloat16* __restrict__ out,
    const int* __restrict__ seq_lens,
    const int* __restrict__ q_lens,
    const int* __restrict__ block_table,
    int block_size, int max_blocks,
    AttnStrides2 st,
    int B, int H, int Hkv, int Sq, int Sk, float scale) {
  const float scale2 = scale * 1.4426950408889634f;
  const __hip_bfloat16* k = (const __hip_bfloat16*)kvoid;
  const __hip_bfloat16* v = (const __hip_bfloat16*)vvoid;
  constexpr int D = HEAD_DIM;
  constexpr int KSTRIDE = D + PAD;
  constexpr int VSTRIDE = BLOCK_N + PAD;

  // KDIRECT uses a minimal K buffer (compiler elides unused LDS? no --
  // size the array by the flag)
  __shared__ short lds_k[KDIRECT ? 1 : 2][KDIRECT ? 1 : BLOCK_N * KSTRIDE];
  __shared__ short lds_vt[2][D * VSTRIDE];

  const int bh = blockIdx.x;
  const int m_tile = (int)gridDim.y - 1 - (int)blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int hkv = h / (H / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  const int m0 = m_tile * BLOCK_M;
  const int wm0 = m0 + wave * 16;

  const long q_base = (long)b * st.qb + (long)h * st.qh;
  const long kv_base = PAGED ? (long)hkv * block_size * D
                             : (long)b * st.kb + (long)hkv * st.kh;
  const long o_base = (long)b * st.ob + (long)h * st.oh;
  const int kv_len = HAS_SEQLENS ? min(seq_lens[b], Sk) : Sk;
  const int q_len = q_lens ? q_lens[b] : (HAS_SEQLENS ? kv_len : Sq);
  const int causal_off = kv_len - q_len;
  const int kv_hi = CAUSAL ? min(kv_len, m0 + BLOCK_M + causal_off) : kv_len;
  const int* btab = PAGED ? block_table + (long)b * max_blocks : nullptr;

  const int frag_row = lane & 15;       // q column (and K key-in-tile)
  const int frag_ko = (lane >> 4) * 8;  // k (d) offset
  const int g = lane >> 4;              // lane group

  // Q fragment: B[k=d][j=qrow] -- same per-lane bytes as v1's A fragment
  bf16x8_t q_frag[D / 32];
#pragma unroll
  for (int kc = 0; kc < D / 32; ++kc) {
    const int qrow = wm0 + frag_row;
    if (qrow < Sq) {
      q_frag[kc] = *reinterpret_cast<const bf16x8_t*>(
          q + q_base + (long)qrow * st.qs + kc * 32 + frag_ko);
    } else {
      q_frag[kc] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // running softmax state: ONE query column per lane (lane-local scalars)
  float m_run = -INFINITY, l_run = 0.f;
  // O^T accumulators: C[d][q] -- lane holds col q = frag_row,
  // rows d = dt*16 + 4g + r
  f32x4_t acc_o[D / 16];
#pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) acc_o[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  constexpr int PIECES = BLOCK_N * D / 8 / (NWAVES * 64);
  bf16x8_t kreg[KDIRECT ? 1 : PIECES], vreg[PIECES];

  auto stage_load = [&](int n0) {
#pragma unroll
    for (int i = 0; i < PIECES; ++i) {
      const int p = tid + i * NWAVES * 64;
      const int gkey = n0 + p / (D / 8);
      const int d8 = (p % (D / 8)) * 8;
      if (gkey < kv_len) {
        long off;
        long blk = 0;
        if (PAGED) {
          blk = btab[gkey / block_size];
          off = (blk * Hkv) * (long)block_size * D + kv_base +
                (long)(gkey % block_size) * D + d8;
        } else {
          off = kv_base + (long)gkey * st.ks + d8;
        }
        if (FP8KV && PAGED) {
          // e4m3 bytes + per-token scales -> bf16 fragments at stage time
          const long sidx = (blk * Hkv + hkv) * block_size +
                            (gkey % block_size);
          const float ksc = k_scale[sidx], vsc = v_scale[sidx];
          const int* k8 = reinterpret_cast<const int*>(
              (const unsigned char*)kvoid + off);
          const int* v8 = reinterpret_cast<const int*>(
              (const unsigned char*)vvoid + off);
          short kb[8], vb[8];
#pragma unroll
          for (int w = 0; w < 2; ++w) {
            v2_fp8x4_to_bf16(k8[w], ksc, kb + 4 * w);
            v2_fp8x4_to_bf16(v8[w], vsc, vb + 4 * w);
          }
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            kreg[i][e] = kb[e];
            vreg[i][e] = vb[e];
          }
        } else {
          if (!KDIRECT) kreg[i] = *reinterpret_cast<const bf16x8_t*>(k + off);
          vreg[i] = *reinterpret_cast<const bf16x8_t*>(v + off);
        }
      } else {
        if (!KDIRECT) kreg[i] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
        vreg[i] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int i = 0; i < PIECES; ++i) {
      const int p = tid + i * NWAVES * 64;
      const int key = p / (D / 8);
      const int d8 = (p % (D / 8)) * 8;
      if (!KDIRECT)
        *reinterpret_cast<bf16x8_t*>(&lds_k[buf][key * KSTRIDE + d8]) =
            kreg[i];
      const int kswz = key ^ (((d8 >> 3) & 7) << 3);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        lds_vt[buf][(d8 + e) * VSTRIDE + kswz] = vreg[i][e];
    }
  };

  // ---- per-tile pieces as lambdas (shared by both loop schedules) ---- //
  auto compute_qk = [&](int n0, int buf, f32x4_t (&acc_s)[BLOCK_N / 16]) {
#pragma unroll
    for (int t = 0; t < BLOCK_N / 16; ++t)
      acc_s[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};
    if (KDIRECT) {
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) {
        const int gkey = min(n0 + t * 16 + frag_row, kv_len - 1);
        const __hip_bfloat16* krow = k + kv_base + (long)gkey * st.ks;
#pragma unroll
        for (int kc = 0; kc < D / 32; ++kc) {
          const bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
              krow + kc * 32 + frag_ko);
          acc_s[t] = MFMA_16x16x32(kf, q_frag[kc], acc_s[t]);  // SWAPPED
        }
      }
    } else {
#pragma unroll
      for (int kc = 0; kc < D / 32; ++kc) {
#pragma unroll
        for (int t = 0; t < BLOCK_N / 16; ++t) {
          const bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
              &lds_k[buf][(t * 16 + frag_row) * KSTRIDE + kc * 32 + frag_ko]);
          acc_s[t] = MFMA_16x16x32(kf, q_frag[kc], acc_s[t]);  // SWAPPED
        }
      }
    }
  };

  // mask + in-register online softmax + P^T redistribution + PV, for the
  // tile whose S^T sits in acc_s and whose V tile is in lds_vt[buf]
  auto softmax_pv = [&](f32x4_t (&acc_s)[BLOCK_N / 16], int n0, int buf) {
    // lane's S values: key = n0 + 16t + 4g + r, q = wm0 + frag_row
    const int qrow = wm0 + frag_row;
    const bool tile_full =
        (n0 + BLOCK_N <= kv_len) &&
        (!CAUSAL || (n0 + BLOCK_N - 1 <= m0 + causal_off));
    float pv[BLOCK_N / 16][4];
    float rmax = -INFINITY;
    if (tile_full) {
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const float s = acc_s[t][r] * scale2;
          pv[t][r] = s;
          rmax = fmaxf(rmax, s);
        }
      }
    } else {
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int key = n0 + 16 * t + 4 * g + r;
          float s = acc_s[t][r] * scale2;
          bool valid = key < kv_len;
          if (CAUSAL) valid = valid && (key <= qrow + causal_off);
          s = valid ? s : -INFINITY;
          pv[t][r] = s;
          rmax = fmaxf(rmax, s);
        }
      }
    }
    // column max/sum: combine the 4 lane groups (2 shfl_xor vs v1's 4)
    rmax = fmaxf(rmax, __shfl_xor(rmax, 16, 64));
    rmax = fmaxf(rmax, __shfl_xor(rmax, 32, 64));
    const float m_new = fmaxf(m_run, rmax);
    const float alpha =
        (m_run == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_new);
    float rsum = 0.f;
#pragma unroll
    for (int t = 0; t < BLOCK_N / 16; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = (pv[t][r] == -INFINITY || m_new == -INFINITY)
                            ? 0.f
                            : __builtin_amdgcn_exp2f(pv[t][r] - m_new);
        pv[t][r] = p;
        rsum += p;
      }
    }
    rsum += __shfl_xor(rsum, 16, 64);
    rsum += __shfl_xor(rsum, 32, 64);
    l_run = l_run * alpha + rsum;
    m_run = m_new;

#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[dt][r] *= alpha;
    }

    // ---- P^T -> PV B-fragments via lane moves (no LDS round-trip) ---- //
    unsigned int pk01[BLOCK_N / 16], pk23[BLOCK_N / 16];
#pragma unroll
    for (int t = 0; t < BLOCK_N / 16; ++t) {
      pk01[t] = pack_bf16x2(pv[t][0], pv[t][1]);
      pk23[t] = pack_bf16x2(pv[t][2], pv[t][3]);
    }
    // B[k=key][j=q] fragment for kc: this lane (group g) holds keys
    // 32kc + 8g + [0..7]; tile t_sel = (g>>1) + 2kc, source groups
    // 2(g&1) and 2(g&1)+1 (r = e&3).
    const int col = frag_row;
#pragma unroll
    for (int kc = 0; kc < BLOCK_N / 32; ++kc) {
      // __shfl transports the SOURCE lane's register, and each source pair
      // serves one tile-0 and one tile-1 target: shuffle both, select
      const int gA = 2 * (g & 1);
      const int srcA = (gA << 4) | col;
      const int srcB = ((gA + 1) << 4) | col;
      const bool hiT = (g >> 1) != 0;
      unsigned int u[4];
      {
        const int a0 = __shfl((int)pk01[2 * kc], srcA, 64);
        const int a1 = __shfl((int)pk01[2 * kc + 1], srcA, 64);
        u[0] = (unsigned int)(hiT ? a1 : a0);
        const int b0 = __shfl((int)pk23[2 * kc], srcA, 64);
        const int b1 = __shfl((int)pk23[2 * kc + 1], srcA, 64);
        u[1] = (unsigned int)(hiT ? b1 : b0);
        const int c0 = __shfl((int)pk01[2 * kc], srcB, 64);
        const int c1 = __shfl((int)pk01[2 * kc + 1], srcB, 64);
        u[2] = (unsigned int)(hiT ? c1 : c0);
        const int d0 = __shfl((int)pk23[2 * kc], srcB, 64);
        const int d1 = __shfl((int)pk23[2 * kc + 1], srcB, 64);
        u[3] = (unsigned int)(hiT ? d1 : d0);
      }
      bf16x8_t pf;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        pf[2 * i] = (short)(u[i] & 0xFFFF);
        pf[2 * i + 1] = (short)(u[i] >> 16);
      }
      // O^T += V^T P^T: A = V^T[d][key] (same LDS reads as v1's B role)
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        const int vrow = dt * 16 + frag_row;
        const int vkey = (kc * 32 + frag_ko) ^ (((vrow >> 3) & 7) << 3);
        const bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
            &lds_vt[buf][vrow * VSTRIDE + vkey]);
        acc_o[dt] = MFMA_16x16x32(vf, pf, acc_o[dt]);
      }
    }
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (PIPE) {
    // T15 double-pipeline: QK^T of tile n+1 (MFMA) issues BEFORE the
    // softmax finish of tile n (VALU) -- independent instruction streams
    // the scheduler interleaves, hiding ~1/3 of the softmax cost. Costs a
    // second barrier per tile and one extra acc_s register set.
    f32x4_t s_prev[BLOCK_N / 16], s_cur[BLOCK_N / 16];
    compute_qk(0, 0, s_prev);
    if (BLOCK_N < kv_hi) stage_load(BLOCK_N);
    int n_prev = 0;
    int cur = 0;
    for (int n0 = BLOCK_N;; n0 += BLOCK_N) {
      const bool has_cur = n0 < kv_hi;
      if (has_cur) {
        __syncthreads();         // all PV reads of lds[cur^1] done
        stage_write(cur ^ 1);    // tile n0 -> other buffer
        __syncthreads();
        compute_qk(n0, cur ^ 1, s_cur);
        if (n0 + BLOCK_N < kv_hi) stage_load(n0 + BLOCK_N);
      }
      softmax_pv(s_prev, n_prev, cur);
      if (!has_cur) break;
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) s_prev[t] = s_cur[t];
      n_prev = n0;
      cur ^= 1;
    }
  } else {
    int cur = 0;
    for (int n0 = 0; n0 < kv_hi; n0 += BLOCK_N) {
      const bool has_next = n0 + BLOCK_N < kv_hi;
      if (has_next) stage_load(n0 + BLOCK_N);
      f32x4_t acc_s[BLOCK_N / 16];
      compute_qk(n0, cur, acc_s);
      softmax_pv(acc_s, n0, cur);
      if (has_next) {
        stage_write(cur ^ 1);
        __syncthreads();
        cur ^= 1;
      }
    }
  }

  // ---- epilogue: O^T / l, 4-element d-contiguous stores ---- //
  const int qrow = wm0 + frag_row;
  if (qrow < q_len) {
    const float inv_l = l_run > 0.f ? 1.0f / l_run : 0.f;
    __hip_bfloat16* dst = out + o_base + (long)qrow * st.os;
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      const int d0 = dt * 16 + 4 * g;
      __hip_bfloat16 pack[4];
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pack[r] = __float2bfloat16(acc_o[dt][r] * inv_l);
      *reinterpret_cast<uint2*>(dst + d0) =
          *reinterpret_cast<const uint2*>(pack);
    }
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
torch::Tensor attention_prefill_v2(torch::Tensor q, torch::Tensor k,
                                   torch::Tensor v, bool causal, double scale,
                                   c10::optional<torch::Tensor> seq_lens,
                                   bool bshd, bool kdirect, bool pipe) {
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attention: bf16 only");
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1);
  const int hdim = bshd ? 2 : 1, sdim = bshd ? 1 : 2;
  const int B = q.size(0), H = q.size(hdim), Sq = q.size(sdim), D = q.size(3);
  const int Hkv = k.size(hdim), Sk = k.size(sdim);
  TORCH_CHECK(H % Hkv == 0);
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(v.size(hdim) == Hkv && v.size(sdim) == Sk && v.size(3) == D);
  TORCH_CHECK(k.stride(0) == v.stride(0) && k.stride(1) == v.stride(1) &&
              k.stride(2) == v.stride(2));

  auto out = bshd ? torch::empty({B, Sq, H, D}, q.options())
                  : torch::empty({B, H, Sq, D}, q.options());
  AttnStrides2 st;
  st.qb = q.stride(0); st.qh = q.stride(hdim); st.qs = q.stride(sdim);
  st.kb = k.stride(0); st.kh = k.stride(hdim); st.ks = k.stride(sdim);
  st.ob = out.stride(0); st.oh = out.stride(hdim); st.os = out.stride(sdim);
  const int* sl = nullptr;
  torch::Tensor sl_t;
  if (seq_lens.has_value() && seq_lens->defined()) {
    sl_t = seq_lens->to(q.device(), at::kInt).contiguous();
    sl = sl_t.data_ptr<int>();
  }
  dim3 grid(B * H, (Sq + BLOCK_M - 1) / BLOCK_M);
  dim3 block(NWAVES * 64);
  hipStream_t stream_ = cmls::current_stream();
#define LAUNCH_ATTN2(DD, CC, SS, KD, PP)                                     \
  hipLaunchKernelGGL((attn_prefill_v2_kernel<DD, CC, SS, false, KD, PP>),    \
                     grid, block, 0,                                         \
                     stream_, (const __hip_bfloat16*)q.data_ptr(),           \
                     (const void*)k.data_ptr(),                              \
                     (const void*)v.data_ptr(), nullptr, nullptr,            \
                     (__hip_bfloat16*)out.data_ptr(), sl, nullptr, nullptr,  \
                     0, 0, st, B, H, Hkv, Sq, Sk, (float)scale)
#define LAUNCH_ATTN2_KD(DD, CC, SS)                                          \
  do { if (pipe) LAUNCH_ATTN2(DD, CC, SS, false, true);                      \
       else if (kdirect) LAUNCH_ATTN2(DD, CC, SS, true, false);              \
       else LAUNCH_ATTN2(DD, CC, SS, false, false); } while (0)
  if (D == 64) {
    if (causal) { if (sl) LAUNCH_ATTN2_KD(64, true, true); else LAUNCH_ATTN2_KD(64, true, false); }
    else        { if (sl) LAUNCH_ATTN2_KD(64, false, true); else LAUNCH_ATTN2_KD(64, false, false); }
  } else {
    if (causal) { if (sl) LAUNCH_ATTN2_KD(128, true, true); else LAUNCH_ATTN2_KD(128, true, false); }
    else        { if (sl) LAUNCH_ATTN2_KD(128, false, true); else LAUNCH_ATTN2_KD(128, false, false); }
  }
#undef LAUNCH_ATTN2_KD
#undef LAUNCH_ATTN2
  return out;
}

torch::Tensor attention_prefill_paged_v2(
    torch::Tensor q, torch::Tensor k_cache, torch::Tensor v_cache,
    torch::Tensor block_table, torch::Tensor kv_lens, torch::Tensor q_lens,
    double scale, c10::optional<torch::Tensor> k_scale,
    c10::optional<torch::Tensor> v_scale) {
  TORCH_CHECK(q.dim() == 4 && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.stride(3) == 1 && q.stride(2) == q.size(3));
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  const bool fp8kv = k_cache.scalar_type() == at::kByte ||
                     k_cache.scalar_type() == at::kFloat8_e4m3fn;
  const float* kscp = nullptr;
  const float* vscp = nullptr;
  if (fp8kv) {
    TORCH_CHECK(k_scale.has_value() && v_scale.has_value(),
                "fp8 kv cache needs k_scale/v_scale");
    TORCH_CHECK(k_scale->scalar_type() == at::kFloat &&
                k_scale->is_contiguous() && v_scale->is_contiguous());
    kscp = k_scale->data_ptr<float>();
    vscp = v_scale->data_ptr<float>();
  }
  const int B = q.size(0), Sq = q.size(1), H = q.size(2), D = q.size(3);
  const int Hkv = k_cache.size(1), BS = k_cache.size(2);
  const int max_blocks = block_table.size(1);
  TORCH_CHECK(D == 64 || D == 128);
  TORCH_CHECK(H % Hkv == 0);
  auto bt = block_table.to(q.device(), at::kInt).contiguous();
  auto kl = kv_lens.to(q.device(), at::kInt).contiguous();
  auto ql = q_lens.to(q.device(), at::kInt).contiguous();
  auto out = torch::empty({B, Sq, H, D}, q.options());
  AttnStrides2 st;
  st.qb = q.stride(0); st.qh = q.stride(2); st.qs = q.stride(1);
  st.kb = 0; st.kh = 0; st.ks = 0;
  st.ob = out.stride(0); st.oh = out.stride(2); st.os = out.stride(1);
  const int Sk = kl.max().item<int>();
  dim3 grid(B * H, (Sq + BLOCK_M - 1) / BLOCK_M);
  dim3 block(NWAVES * 64);
  hipStream_t stream_ = cmls::current_stream();
#define LAUNCH_PAGED2(DD, F8)                                                \
  hipLaunchKernelGGL(                                                        \
      (attn_prefill_v2_kernel<DD, true, true, true, false, false, F8>),      \
      grid, block, 0, stream_, (const __hip_bfloat16*)q.data_ptr(),          \
      (const void*)k_cache.data_ptr(), (const void*)v_cache.data_ptr(),      \
      kscp, vscp, (__hip_bfloat16*)out.data_ptr(), kl.data_ptr<int>(),       \
      ql.data_ptr<int>(), bt.data_ptr<int>(), BS, max_blocks,                \
      st, B, H, Hkv, Sq, Sk, (float)scale)
  if (D == 64) { if (fp8kv) LAUNCH_PAGED2(64, true); else LAUNCH_PAGED2(64, false); }
  else         { if (fp8kv) LAUNCH_PAGED2(128, true); else LAUNCH_PAGED2(128, false); }
#undef LAUNCH_PAGED2
  return out;
}
#endif  // CMLS_KERNEL_ONLY
