// Fused multi-head attention (prefill / encoder) for gfx950.
//
// One kernel: S = QK^T -> online softmax -> O = P V, never materializing
// S/P in HBM (the reference's Triton/vLLM engines own this step; SURVEY.md
// §2.6). Flash-attention-2 style tiling on MFMA:
//
//   grid  = (ceil(Sq/64), B*H);  block = 4 waves (256 threads)
//   each wave owns 16 query rows; the workgroup shares K/V tiles of 64 keys
//   staged in LDS (K row-major, V transposed at stage time so PV B-fragments
//   are contiguous ds_read_b128).
//
// MFMA: v_mfma_f32_16x16x32_bf16.  Fragment maps (cdna4 §3):
//   A[i][k]: lane l holds i = l%16, k = (l/16)*8 + e   (e = 0..7)
//   B[k][j]: lane l holds j = l%16, k = (l/16)*8 + e
//   C/D    : lane l holds col = l%16, row = (l/16)*4 + reg
//
// LDS rows are padded by 8 bf16 (stride 144 B): the 16-lane groups of
// ds_read_b128 then hit 16 distinct 16-B slots (bank-conflict-free; the
// unpadded 128 B stride is 8-way).
//
// Online softmax is the standard running (m, l) update; row statistics
// reduce over the 16 lanes of each C-layout row group via shfl_xor 1/2/4/8.
//
// Supports: head_dim 64/128, GQA (H_kv | H), causal masking, per-batch
// kv sequence lengths (padding masks). bf16 in/out, fp32 accumulate.
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;   // 4 VGPRs
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define MFMA_16x16x32(A, B, C) \
  __builtin_amdgcn_mfma_f32_16x16x32_bf16((A), (B), (C), 0, 0, 0)

constexpr int BLOCK_M = 128;  // query rows per workgroup
constexpr int BLOCK_N = 64;   // keys per tile
constexpr int NWAVES = 8;     // one 16-row M-slice per wave; 8 waves =
                              // 2 waves/SIMD so MFMA + staging latency hide
                              // across waves even at 1 block/CU
constexpr int PAD = 8;        // bf16 elements of row padding (16 B)

// row-group reduction: combine over the 16 lanes holding one C-layout row
__device__ __forceinline__ float rowgroup_max(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}
__device__ __forceinline__ float rowgroup_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
  return v;
}

struct AttnStrides {
  long qb, qh, qs;   // q/out: batch, head, seq strides (elements)
  long kb, kh, ks;   // k/v share layout
  long ob, oh, os;
};

template <int HEAD_DIM, bool CAUSAL, bool HAS_SEQLENS, bool PAGED = false>
__global__ __launch_bounds__(NWAVES * 64, 2) void attn_prefill_kernel(
    const __hip_bfloat16* __restrict__ q,   // [B, H, Sq, D] via strides
    const __hip_bfloat16* __restrict__ k,   // dense via strides, or paged
    const __hip_bfloat16* __restrict__ v,   //   cache [NB, Hkv, BS, D]
    __hip_bfloat16* __restrict__ out,
    const int* __restrict__ seq_lens,       // [B] kv lengths (or null)
    const int* __restrict__ q_lens,         // [B] query lengths (or null):
                                            // chunked prefill -- per-seq
                                            // valid q rows + causal offset
    const int* __restrict__ block_table,    // [B, max_blocks] when PAGED
    int block_size, int max_blocks,
    AttnStrides st,
    int B, int H, int Hkv, int Sq, int Sk, float scale) {
  // softmax in the exp2 domain: v_exp_f32 IS 2^x, so folding log2(e) into
  // the score scale deletes one full-rate VALU mul per exp call
  const float scale2 = scale * 1.4426950408889634f;
  constexpr int D = HEAD_DIM;
  constexpr int KSTRIDE = D + PAD;          // LDS K row stride (bf16)
  constexpr int VSTRIDE = BLOCK_N + PAD;    // LDS V^T row stride
  constexpr int PSTRIDE = BLOCK_N + PAD;

  // double-buffered K/V tiles: tile n+1's global loads are issued into
  // registers BEFORE computing on tile n (T14 issue-early / write-late) --
  // HBM latency hides under the MFMA/softmax work, and the ds_write of the
  // next tile lands in the other buffer (no barrier between compute and
  // write, only before the swapped buffer is read)
  __shared__ short lds_k[2][BLOCK_N * KSTRIDE];
  __shared__ short lds_vt[2][D * VSTRIDE];
  __shared__ short lds_p[NWAVES * 16 * PSTRIDE];

  // grid = (B*H, n_m_tiles): bh on x so the dispatcher's XCD round-robin
  // (XCD = linear_id % 8) decorrelates from the causal work skew, and heavy
  // (high-m) tiles dispatch FIRST so light blocks backfill the tail --
  // with m_tile on x, causal ran at full-attention wall time (measured 2x).
  const int bh = blockIdx.x;
  const int m_tile = (int)gridDim.y - 1 - (int)blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int hkv = h / (H / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  const int m0 = m_tile * BLOCK_M;          // first query row of the block
  const int wm0 = m0 + wave * 16;           // this wave's first query row

  const long q_base = (long)b * st.qb + (long)h * st.qh;
  const long kv_base = PAGED ? (long)hkv * block_size * D
                             : (long)b * st.kb + (long)hkv * st.kh;
  const long o_base = (long)b * st.ob + (long)h * st.oh;
  const int kv_len = HAS_SEQLENS ? min(seq_lens[b], Sk) : Sk;
  // per-seq query length: explicit q_lens (chunked prefill), else kv_len
  // when seq_lens is given (padded dense batch: rows beyond kv_len are
  // pad -- q_len = Sq here made causal_off NEGATIVE for short sequences
  // and truncated their history by the pad amount)
  const int q_len = q_lens ? q_lens[b] : (HAS_SEQLENS ? kv_len : Sq);
  // causal: the chunk's LAST query row attends up to the LAST key
  // (history offset = kv_len - q_len)
  const int causal_off = kv_len - q_len;
  const int kv_hi = CAUSAL ? min(kv_len, m0 + BLOCK_M + causal_off) : kv_len;
  const int* btab = PAGED ? block_table + (long)b * max_blocks : nullptr;

  // ---- load this wave's Q fragments (scaled once; fp32->bf16 later in S) --
  const int frag_row = lane & 15;           // i
  const int frag_ko = (lane >> 4) * 8;      // k offset
  bf16x8_t q_frag[D / 32];
#pragma unroll
  for (int kc = 0; kc < D / 32; ++kc) {
    const int qrow = wm0 + frag_row;
    if (qrow < Sq) {
      const __hip_bfloat16* src =
          q + q_base + (long)qrow * st.qs + kc * 32 + frag_ko;
      q_frag[kc] = *reinterpret_cast<const bf16x8_t*>(src);
    } else {
      q_frag[kc] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }

  // ---- running softmax state: 4 rows per lane (reg r -> row base+r) ------ //
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
  f32x4_t acc_o[D / 16];
#pragma unroll
  for (int dt = 0; dt < D / 16; ++dt) acc_o[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  // staging helpers: each thread owns PIECES 16-B pieces of the K and V
  // tiles (piece p -> key p/(D/8), d-offset (p%(D/8))*8)
  constexpr int PIECES = BLOCK_N * D / 8 / (NWAVES * 64);
  bf16x8_t kreg[PIECES], vreg[PIECES];

  auto stage_load = [&](int n0) {
#pragma unroll
    for (int i = 0; i < PIECES; ++i) {
      const int p = tid + i * NWAVES * 64;
      const int gkey = n0 + p / (D / 8);
      const int d8 = (p % (D / 8)) * 8;
      if (gkey < kv_len) {
        long off;
        if (PAGED) {
          const long blk = btab[gkey / block_size];
          off = (blk * Hkv) * (long)block_size * D + kv_base +
                (long)(gkey % block_size) * D + d8;
        } else {
          off = kv_base + (long)gkey * st.ks + d8;
        }
        kreg[i] = *reinterpret_cast<const bf16x8_t*>(k + off);
        vreg[i] = *reinterpret_cast<const bf16x8_t*>(v + off);
      } else {
        kreg[i] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
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
      *reinterpret_cast<bf16x8_t*>(&lds_k[buf][key * KSTRIDE + d8]) = kreg[i];
      // V scatter-transposed with the key-XOR bank swizzle (see PV reads)
      const int kswz = key ^ (((d8 >> 3) & 7) << 3);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        lds_vt[buf][(d8 + e) * VSTRIDE + kswz] = vreg[i][e];
    }
  };

  // ================= KV tile loop ================= //
  stage_load(0);
  stage_write(0);
  __syncthreads();
  int cur = 0;
  for (int n0 = 0; n0 < kv_hi; n0 += BLOCK_N) {
    // issue next tile's global loads now; they complete under this tile's
    // MFMA + softmax work
    const bool has_next = n0 + BLOCK_N < kv_hi;
    if (has_next) stage_load(n0 + BLOCK_N);

    // ---- S = Q K^T for this wave's 16 rows x 64 keys ---- //
    f32x4_t acc_s[BLOCK_N / 16];
#pragma unroll
    for (int t = 0; t < BLOCK_N / 16; ++t) acc_s[t] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kc = 0; kc < D / 32; ++kc) {
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) {
        // B[k][j] = K[j + 16t][k]: contiguous 8 bf16 at row j, col k-offset
        const bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
            &lds_k[cur][(t * 16 + frag_row) * KSTRIDE + kc * 32 + frag_ko]);
        acc_s[t] = MFMA_16x16x32(q_frag[kc], kf, acc_s[t]);
      }
    }

    // ---- mask + online softmax ---- //
    // interior tiles (every key valid for every query of the block) skip
    // masking entirely: the per-element compare+select chain was ~20% of
    // the kernel's VALU work (PMC: VALU instrs 10.5x MFMA instrs)
    const int row_base = wm0 + (lane >> 4) * 4;  // C layout: row = base + reg
    const int col_base = n0 + (lane & 15);       // col = base + 16t
    float pvals[4][BLOCK_N / 16];
    float alpha[4];
    const bool tile_full =
        (n0 + BLOCK_N <= kv_len) &&
        (!CAUSAL || (n0 + BLOCK_N - 1 <= m0 + causal_off));
    if (tile_full) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float rmax = -INFINITY;
#pragma unroll
        for (int t = 0; t < BLOCK_N / 16; ++t) {
          const float s = acc_s[t][r] * scale2;
          pvals[r][t] = s;
          rmax = fmaxf(rmax, s);
        }
        rmax = rowgroup_max(rmax);
        const float m_new = fmaxf(m_run[r], rmax);
        alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m_run[r] - m_new);
        float rsum = 0.f;
#pragma unroll
        for (int t = 0; t < BLOCK_N / 16; ++t) {
          const float p = __builtin_amdgcn_exp2f(pvals[r][t] - m_new);
          pvals[r][t] = p;
          rsum += p;
        }
        rsum = rowgroup_sum(rsum);
        l_run[r] = l_run[r] * alpha[r] + rsum;
        m_run[r] = m_new;
      }
    } else {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = row_base + r;
        float rmax = -INFINITY;
#pragma unroll
        for (int t = 0; t < BLOCK_N / 16; ++t) {
          float s = acc_s[t][r] * scale2;
          const int key = col_base + t * 16;
          bool valid = key < kv_len;
          if (CAUSAL) valid = valid && (key <= qrow + causal_off);
          s = valid ? s : -INFINITY;
          pvals[r][t] = s;
          rmax = fmaxf(rmax, s);
        }
        rmax = rowgroup_max(rmax);
        const float m_new = fmaxf(m_run[r], rmax);
        // all-masked rows keep m = -inf; exp() yields 0 contributions
        alpha[r] = (m_run[r] == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m_run[r] - m_new);
        float rsum = 0.f;
#pragma unroll
        for (int t = 0; t < BLOCK_N / 16; ++t) {
          const float p = (pvals[r][t] == -INFINITY || m_new == -INFINITY)
                              ? 0.f
                              : __builtin_amdgcn_exp2f(pvals[r][t] - m_new);
          pvals[r][t] = p;
          rsum += p;
        }
        rsum = rowgroup_sum(rsum);
        l_run[r] = l_run[r] * alpha[r] + rsum;
        m_run[r] = m_new;
      }
    }
    // rescale O by alpha (per C-layout row)
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[dt][r] *= alpha[r];
    }

    // ---- write P (bf16) to this wave's LDS slab, re-read as A fragments --
    short* pslab = &lds_p[wave * 16 * PSTRIDE];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int prow = (lane >> 4) * 4 + r;
#pragma unroll
      for (int t = 0; t < BLOCK_N / 16; ++t) {
        __hip_bfloat16 pb = __float2bfloat16(pvals[r][t]);
        pslab[prow * PSTRIDE + (lane & 15) + t * 16] =
            *reinterpret_cast<short*>(&pb);
      }
    }
    // cross-LANE hand-off through LDS within the wave: the compiler's
    // per-thread alias analysis cannot see it, so drain the wave's ds queue
    // before any lane reads another lane's P ("memory" orders the compiler's
    // ds ops around the asm; the MFMA consumes the loads by data dependency)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P V ---- //
#pragma unroll
    for (int kc = 0; kc < BLOCK_N / 32; ++kc) {
      const bf16x8_t pf = *reinterpret_cast<const bf16x8_t*>(
          &pslab[frag_row * PSTRIDE + kc * 32 + frag_ko]);
#pragma unroll
      for (int dt = 0; dt < D / 16; ++dt) {
        // B[k][j] = V[k + 32kc][j + 16dt] = vt[j + 16dt][k]: contiguous in k;
        // apply the staging key-XOR (8-aligned base ^ bits 3-5 stays
        // 16-B aligned)
        const int vrow = dt * 16 + frag_row;
        const int vkey = (kc * 32 + frag_ko) ^ (((vrow >> 3) & 7) << 3);
        const bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
            &lds_vt[cur][vrow * VSTRIDE + vkey]);
        acc_o[dt] = MFMA_16x16x32(pf, vf, acc_o[dt]);
      }
    }
    if (has_next) {
      stage_write(cur ^ 1);   // other buffer: no read/write hazard with cur
      __syncthreads();        // next iteration reads the freshly written buf
      cur ^= 1;
    }
  }

  // ---- epilogue: O / l, bf16 store ---- //
  const int row_base = wm0 + (lane >> 4) * 4;
  const int col = lane & 15;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = row_base + r;
    if (qrow >= q_len) continue;
    const float inv_l = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
    __hip_bfloat16* dst = out + o_base + (long)qrow * st.os;
#pragma unroll
    for (int dt = 0; dt < D / 16; ++dt) {
      dst[dt * 16 + col] = __float2bfloat16(acc_o[dt][r] * inv_l);
    }
  }
}

}  // namespace

torch::Tensor attention_prefill(torch::Tensor q, torch::Tensor k,
                                torch::Tensor v, bool causal, double scale,
                                c10::optional<torch::Tensor> seq_lens,
                                bool bshd) {
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4,
              "q/k/v must be 4-D");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attention: bf16 only");
  TORCH_CHECK(q.stride(3) == 1 && k.stride(3) == 1 && v.stride(3) == 1,
              "head_dim must be dense");
  // bshd: tensors are [B, S, H, D] (e.g. views into a merged QKV
  // projection) -- no transpose copies; default is [B, H, S, D]
  const int hdim = bshd ? 2 : 1, sdim = bshd ? 1 : 2;
  const int B = q.size(0), H = q.size(hdim), Sq = q.size(sdim),
            D = q.size(3);
  const int Hkv = k.size(hdim), Sk = k.size(sdim);
  TORCH_CHECK(H % Hkv == 0, "H must be a multiple of H_kv");
  TORCH_CHECK(D == 64 || D == 128, "head_dim must be 64 or 128");
  TORCH_CHECK(v.size(hdim) == Hkv && v.size(sdim) == Sk && v.size(3) == D);
  TORCH_CHECK(k.stride(0) == v.stride(0) && k.stride(1) == v.stride(1) &&
              k.stride(2) == v.stride(2), "k/v must share layout");

  auto out = bshd ? torch::empty({B, Sq, H, D}, q.options())
                  : torch::empty({B, H, Sq, D}, q.options());
  AttnStrides st;
  st.qb = q.stride(0); st.qh = q.stride(hdim); st.qs = q.stride(sdim);
  st.kb = k.stride(0); st.kh = k.stride(hdim); st.ks = k.stride(sdim);
  st.ob = out.stride(0); st.oh = out.stride(hdim); st.os = out.stride(sdim);
  const int* sl = nullptr;
  torch::Tensor sl_t;
  if (seq_lens.has_value() && seq_lens->defined()) {
    sl_t = seq_lens->to(q.device(), at::kInt).contiguous();
    TORCH_CHECK(sl_t.numel() == B, "seq_lens must be [B]");
    sl = sl_t.data_ptr<int>();
  }
  dim3 grid(B * H, (Sq + BLOCK_M - 1) / BLOCK_M);
  dim3 block(NWAVES * 64);
  hipStream_t stream_ = cmls::current_stream();

#define LAUNCH_ATTN(DD, CC, SS)                                              \
  hipLaunchKernelGGL((attn_prefill_kernel<DD, CC, SS>), grid, block, 0,      \
                     stream_,                                                \
                     (const __hip_bfloat16*)q.data_ptr(),                    \
                     (const __hip_bfloat16*)k.data_ptr(),                    \
                     (const __hip_bfloat16*)v.data_ptr(),                    \
                     (__hip_bfloat16*)out.data_ptr(), sl, nullptr, nullptr,  \
                     0, 0, st, B, H, Hkv, Sq, Sk, (float)scale)
  if (D == 64) {
    if (causal) { if (sl) LAUNCH_ATTN(64, true, true); else LAUNCH_ATTN(64, true, false); }
    else        { if (sl) LAUNCH_ATTN(64, false, true); else LAUNCH_ATTN(64, false, false); }
  } else {
    if (causal) { if (sl) LAUNCH_ATTN(128, true, true); else LAUNCH_ATTN(128, true, false); }
    else        { if (sl) LAUNCH_ATTN(128, false, true); else LAUNCH_ATTN(128, false, false); }
  }
#undef LAUNCH_ATTN
  return out;
}


torch::Tensor attention_prefill_paged(
    torch::Tensor q,            // [B, Sq, H, D] (bshd; strided views ok)
    torch::Tensor k_cache,      // [NB, Hkv, BS, D]
    torch::Tensor v_cache,
    torch::Tensor block_table,  // int32 [B, max_blocks]
    torch::Tensor kv_lens,      // int32 [B] total keys (history + chunk)
    torch::Tensor q_lens,       // int32 [B] valid query rows per sequence
    double scale) {
  // chunked-prefill attention: the chunk's queries attend causally to the
  // full paged history + the chunk itself (K/V already scattered into the
  // cache by kv_cache_write before this call).
  TORCH_CHECK(q.dim() == 4 && q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.stride(3) == 1 && q.stride(2) == q.size(3),
              "q heads must be dense");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  const int B = q.size(0), Sq = q.size(1), H = q.size(2), D = q.size(3);
  const int Hkv = k_cache.size(1), BS = k_cache.size(2);
  const int max_blocks = block_table.size(1);
  TORCH_CHECK(D == 64 || D == 128);
  TORCH_CHECK(H % Hkv == 0);
  auto bt = block_table.to(q.device(), at::kInt).contiguous();
  auto kl = kv_lens.to(q.device(), at::kInt).contiguous();
  auto ql = q_lens.to(q.device(), at::kInt).contiguous();
  auto out = torch::empty({B, Sq, H, D}, q.options());
  AttnStrides st;
  st.qb = q.stride(0); st.qh = q.stride(2); st.qs = q.stride(1);
  st.kb = 0; st.kh = 0; st.ks = 0;  // unused on the paged path
  st.ob = out.stride(0); st.oh = out.stride(2); st.os = out.stride(1);
  // Sk bound for the kv loop = max kv_len; kernel clamps per sequence
  const int Sk = kl.max().item<int>();
  dim3 grid(B * H, (Sq + BLOCK_M - 1) / BLOCK_M);
  dim3 block(NWAVES * 64);
  hipStream_t stream_ = cmls::current_stream();

#define LAUNCH_PAGED(DD)                                                     \
  hipLaunchKernelGGL((attn_prefill_kernel<DD, true, true, true>), grid,      \
                     block, 0, stream_,                                      \
                     (const __hip_bfloat16*)q.data_ptr(),                    \
                     (const __hip_bfloat16*)k_cache.data_ptr(),              \
                     (const __hip_bfloat16*)v_cache.data_ptr(),              \
                     (__hip_bfloat16*)out.data_ptr(), kl.data_ptr<int>(),    \
                     ql.data_ptr<int>(), bt.data_ptr<int>(), BS, max_blocks, \
                     st, B, H, Hkv, Sq, Sk, (float)scale)
  if (D == 64) LAUNCH_PAGED(64);
  else LAUNCH_PAGED(128);
#undef LAUNCH_PAGED
  return out;
}
