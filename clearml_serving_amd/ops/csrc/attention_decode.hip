// Paged-KV decode attention + KV-cache scatter for gfx950.
//
// The decode step of the native LLM engine (replaces the work the reference
// delegates to vLLM): one query token per sequence against a
// paged KV cache.
//
// attention_decode:
//   grid = (B, Hkv); block = 4 waves.
//   Each workgroup serves ALL GQ = H/Hkv query heads of one kv head, so K/V
//   rows stream from HBM once per kv head (not once per q head).
//   Within a wave, keys are processed 4 at a time by 16-lane groups: each
//   lane holds an 8-element slice of the 128-wide row (one bf16x8 = 16 B
//   load, coalesced across the group). Scores reduce over the group with 4
//   shfl_xor steps; the online-softmax state (m, l) is wave-uniform per
//   q head; per-lane O accumulators merge across groups (shfl) and waves
//   (LDS) at the end -- flash-decoding style combine.
//
// kv_cache_write: scatter [T, Hkv, D] new keys/values into
//   [num_blocks, Hkv, block_size, D] caches at slot_mapping[t].
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(2))) short bf16x2_t;

// values uniform within each 16-lane group: combining the 4 groups of a
// wave needs only xor 16 and 32
__device__ __forceinline__ float group_max4(float v) {
  v = fmaxf(v, __shfl_xor(v, 16, 64));
  v = fmaxf(v, __shfl_xor(v, 32, 64));
  return v;
}
__device__ __forceinline__ float group_sum4(float v) {
  v += __shfl_xor(v, 16, 64);
  v += __shfl_xor(v, 32, 64);
  return v;
}

// UNROLL/MINB default to the measured-best production point (4-deep key
// unroll, 148 VGPR -> 3 waves/SIMD for GQ=4).  <.., 2, 4> is the
// occupancy-first alternative (116 VGPR -> 4 waves/SIMD at half the loads
// in flight per wave); scripts/decode_ab.hip A/Bs them.
template <int D, int GQ, int UNROLL = 4, int MINB = 2>
__global__ __launch_bounds__(256, MINB) void attn_decode_kernel(
    const __hip_bfloat16* __restrict__ q,        // [B, H, D]
    const __hip_bfloat16* __restrict__ k_cache,  // [NB, Hkv, BS, D]
    const __hip_bfloat16* __restrict__ v_cache,  // [NB, Hkv, BS, D]
    const int* __restrict__ block_table,         // [B, max_blocks]
    const int* __restrict__ seq_lens,            // [B]
    __hip_bfloat16* __restrict__ out,            // [B, H, D]
    float* __restrict__ partial_o,               // [B, H, SPLITS, D] f32
    float* __restrict__ partial_ml,              // [B, H, SPLITS, 2] f32
    int splits, long q_bstride,
    int H, int Hkv, int block_size, int max_blocks, float scale) {
  // softmax in the exp2 domain (v_exp_f32 is 2^x; log2e folds into scale)
  const float scale2 = scale * 1.4426950408889634f;
  constexpr int NW = 4;        // waves
  constexpr int GROUPS = 4;    // 16-lane key groups per wave
  constexpr int EPL = D / 16;  // elements per lane (8 for D=128)

  const int b = blockIdx.x;
  const int hkv = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int group = lane >> 4;        // 0..3
  const int sub = lane & 15;          // d-slice owner
  const int split = blockIdx.z;
  const int total_keys = seq_lens[b];
  // chunk rounded to the 16-key workgroup iteration so splits don't overlap
  const int chunk = ((total_keys + splits - 1) / splits + 15) & ~15;
  const int key_lo = split * chunk;
  const int n_keys = min(total_keys, key_lo + chunk);
  if (key_lo >= total_keys && split > 0) {
    // empty split: publish a neutral partial (m=-inf skips it in combine)
    if (threadIdx.x < 2 * GQ && partial_ml) {
      const int g = threadIdx.x / 2;
      const int h = hkv * GQ + g;
      float* ml = partial_ml + (((long)b * H + h) * splits + split) * 2;
      ml[threadIdx.x & 1] = (threadIdx.x & 1) ? 0.f : -INFINITY;
    }
    return;
  }

  // q fragments: packed bf16 pairs feed v_dot2_f32_bf16 (2 MACs per
  // instruction in the K dot products; scale folds into the score later)
  bf16x2_t qp[GQ][EPL / 2];
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
    const int h = hkv * GQ + g;
    const __hip_bfloat16* qrow = q + (long)b * q_bstride + (long)h * D +
                                 sub * EPL;
#pragma unroll
    for (int e = 0; e < EPL / 2; ++e)
      qp[g][e] = *reinterpret_cast<const bf16x2_t*>(qrow + 2 * e);
  }

  float m_run[GQ], l_run[GQ], o_acc[GQ][EPL];
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
    m_run[g] = -INFINITY;
    l_run[g] = 0.f;
#pragma unroll
    for (int e = 0; e < EPL; ++e) o_acc[g][e] = 0.f;
  }

  const int* btab = block_table + (long)b * max_blocks;
  const long hk_off = (long)hkv * block_size * D;

  // keys processed 16 per workgroup iteration (wave w, group g -> key
  // it*16 + w*4 + g), UNROLLED 4 iterations deep: 8 x 16-B loads per lane
  // in flight (one K + one V per unrolled key) -- a single load in flight
  // was latency-bound at ~1.7 TB/s
  const int per_iter = NW * GROUPS;
  const int n_iters = (n_keys - key_lo + per_iter - 1) / per_iter;
  for (int it0 = 0; it0 < n_iters; it0 += UNROLL) {
    bf16x8 kvec[UNROLL], vvec[UNROLL];
    bool valid[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int key = key_lo + (it0 + u) * per_iter + wave * GROUPS + group;
      valid[u] = (it0 + u) < n_iters && key < n_keys;
      const int kc = min(key, n_keys - 1);  // clamped: always a mapped page
      const int blk = btab[kc / block_size];
      const long base = ((long)blk * Hkv) * block_size * D + hk_off +
                        (long)(kc % block_size) * D;
      kvec[u].u = *reinterpret_cast<const uint32x4*>(
          k_cache + base + sub * EPL);
      vvec[u].u = *reinterpret_cast<const uint32x4*>(
          v_cache + base + sub * EPL);
    }
    // scores for the whole 4-iteration chunk (4x16 = 64 wave keys)
    float score[UNROLL][GQ];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const bf16x2_t* kp = reinterpret_cast<const bf16x2_t*>(&kvec[u]);
#pragma unroll
      for (int g = 0; g < GQ; ++g) {
        float acc = 0.f;
#pragma unroll
        for (int e = 0; e < EPL / 2; ++e)
          acc = __builtin_amdgcn_fdot2_f32_bf16(qp[g][e], kp[e], acc, false);
        score[u][g] = acc;
      }
      // group-level dot reduction (16 lanes hold partial sums)
#pragma unroll
      for (int g = 0; g < GQ; ++g) {
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          score[u][g] += __shfl_xor(score[u][g], off, 64);
        score[u][g] = valid[u] ? score[u][g] * scale2 : -INFINITY;
      }
    }
    // ONE online-softmax update per chunk per head (rescaling O per
    // 4-key iteration was the serial VALU chain limiting bandwidth)
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      float cmax = score[0][g];
#pragma unroll
      for (int u = 1; u < UNROLL; ++u) cmax = fmaxf(cmax, score[u][g]);
      cmax = group_max4(cmax);
      if (cmax == -INFINITY) continue;
      const float m_new = fmaxf(m_run[g], cmax);
      const float alpha =
          (m_run[g] == -INFINITY) ? 0.f
                                  : __builtin_amdgcn_exp2f(m_run[g] - m_new);
      float psum = 0.f;
#pragma unroll
      for (int e = 0; e < EPL; ++e) o_acc[g][e] *= alpha;
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        const float p = valid[u] ? __builtin_amdgcn_exp2f(score[u][g] - m_new) : 0.f;
        psum += p;
#pragma unroll
        for (int e = 0; e < EPL; ++e)
          o_acc[g][e] += p * to_f32(vvec[u].h[e]);
      }
      l_run[g] = l_run[g] * alpha + group_sum4(psum);
      m_run[g] = m_new;
    }
  }

  // merge the 4 key groups within each wave: lanes with equal `sub` but
  // different `group` hold partial O for the same d-slice (m/l are already
  // wave-uniform)
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
#pragma unroll
    for (int e = 0; e < EPL; ++e) {
      o_acc[g][e] += __shfl_xor(o_acc[g][e], 16, 64);
      o_acc[g][e] += __shfl_xor(o_acc[g][e], 32, 64);
    }
  }

  // merge the 4 waves through LDS (flash-decoding combine)
  __shared__ float lds_m[NW][GQ], lds_l[NW][GQ];
  __shared__ float lds_o[NW][GQ][D];
  if (group == 0) {  // one group per wave writes its state
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      if (sub == 0) { lds_m[wave][g] = m_run[g]; lds_l[wave][g] = l_run[g]; }
#pragma unroll
      for (int e = 0; e < EPL; ++e) lds_o[wave][g][sub * EPL + e] = o_acc[g][e];
    }
  }
  __syncthreads();
  if (wave == 0 && group == 0) {
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      float m_tot = -INFINITY;
      for (int w = 0; w < NW; ++w) m_tot = fmaxf(m_tot, lds_m[w][g]);
      float l_tot = 0.f;
      float o_tot[EPL];
#pragma unroll
      for (int e = 0; e < EPL; ++e) o_tot[e] = 0.f;
      for (int w = 0; w < NW; ++w) {
        if (lds_m[w][g] == -INFINITY) continue;
        const float f = __builtin_amdgcn_exp2f(lds_m[w][g] - m_tot);
        l_tot += lds_l[w][g] * f;
#pragma unroll
        for (int e = 0; e < EPL; ++e)
          o_tot[e] += lds_o[w][g][sub * EPL + e] * f;
      }
      const int h = hkv * GQ + g;
      if (splits == 1) {
        __hip_bfloat16* dst = out + ((long)b * H + h) * D + sub * EPL;
        const float inv_l = l_tot > 0.f ? 1.0f / l_tot : 0.f;
#pragma unroll
        for (int e = 0; e < EPL; ++e)
          dst[e] = __float2bfloat16(o_tot[e] * inv_l);
      } else {
        // flash-decoding: publish unnormalized partial + (m, l)
        float* po = partial_o + (((long)b * H + h) * splits + split) * D +
                    sub * EPL;
#pragma unroll
        for (int e = 0; e < EPL; ++e) po[e] = o_tot[e];
        if (sub == 0) {
          float* ml = partial_ml + (((long)b * H + h) * splits + split) * 2;
          ml[0] = m_tot;
          ml[1] = l_tot;
        }
      }
    }
  }
}

// combine the per-split partials: grid (B, H), one wave per block
template <int D>
__global__ void decode_combine_kernel(
    const float* __restrict__ partial_o,   // [B, H, SPLITS, D]
    const float* __restrict__ partial_ml,  // [B, H, SPLITS, 2]
    __hip_bfloat16* __restrict__ out,      // [B, H, D]
    int splits, int H) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const long base = ((long)b * H + h) * splits;
  float m_tot = -INFINITY;
  for (int s = 0; s < splits; ++s)
    m_tot = fmaxf(m_tot, partial_ml[(base + s) * 2]);
  float l_tot = 0.f;
  float acc[(D + 63) / 64];
#pragma unroll
  for (int i = 0; i < (D + 63) / 64; ++i) acc[i] = 0.f;
  for (int s = 0; s < splits; ++s) {
    const float m = partial_ml[(base + s) * 2];
    if (m == -INFINITY) continue;
    // m values are in the exp2 (log2-probability) domain
    const float f = __builtin_amdgcn_exp2f(m - m_tot);
    l_tot += partial_ml[(base + s) * 2 + 1] * f;
    const float* po = partial_o + (base + s) * D;
#pragma unroll
    for (int i = 0; i < (D + 63) / 64; ++i) {
      const int d = i * 64 + threadIdx.x;
      if (d < D) acc[i] += po[d] * f;
    }
  }
  const float inv_l = l_tot > 0.f ? 1.0f / l_tot : 0.f;
  __hip_bfloat16* dst = out + ((long)b * H + h) * D;
#pragma unroll
  for (int i = 0; i < (D + 63) / 64; ++i) {
    const int d = i * 64 + threadIdx.x;
    if (d < D) dst[d] = __float2bfloat16(acc[i] * inv_l);
  }
}

// ---------------------------------------------------------------------- //
template <typename T>
__global__ void kv_cache_write_kernel(const T* __restrict__ knew,  // [T,Hkv,D]
                                      const T* __restrict__ vnew,
                                      T* __restrict__ k_cache,  // [NB,Hkv,BS,D]
                                      T* __restrict__ v_cache,
                                      const int* __restrict__ slots,  // [T]
                                      int hkv, int d, int block_size,
                                      long k_tstride, long v_tstride) {
  const int t = blockIdx.x;
  const int slot = slots[t];
  if (slot < 0) return;
  const long blk = slot / block_size;
  const int off = slot % block_size;
  for (int i = threadIdx.x; i < hkv * d; i += blockDim.x) {
    const int h = i / d;
    const int dd = i % d;
    const long dst =
        ((blk * hkv + h) * block_size + off) * (long)d + dd;
    k_cache[dst] = knew[(long)t * k_tstride + (long)h * d + dd];
    v_cache[dst] = vnew[(long)t * v_tstride + (long)h * d + dd];
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
torch::Tensor attention_decode(torch::Tensor q, torch::Tensor k_cache,
                               torch::Tensor v_cache,
                               torch::Tensor block_table,
                               torch::Tensor seq_lens, double scale) {
  TORCH_CHECK(q.dim() == 3, "q must be [B, H, D]");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "decode attention: bf16 only");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q heads must be dense");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(1), BS = k_cache.size(2);
  const int max_blocks = block_table.size(1);
  const int GQ = H / Hkv;
  TORCH_CHECK(H % Hkv == 0 && (D == 64 || D == 128));
  TORCH_CHECK(GQ == 1 || GQ == 2 || GQ == 4 || GQ == 8,
              "GQA group must be 1/2/4/8, got ", GQ);
  auto bt = block_table.to(q.device(), at::kInt).contiguous();
  auto sl = seq_lens.to(q.device(), at::kInt).contiguous();
  auto out = torch::empty_like(q);
  hipStream_t stream_ = cmls::current_stream();

  // flash-decoding split count: B*Hkv workgroups alone under-fill 256 CUs at
  // serving batch sizes; split the key range so ~4 workgroups land per CU
  int splits = 1;
  const long base_wg = (long)B * Hkv;
  if (base_wg < 1024) {
    splits = (int)std::min<long>((1024 + base_wg - 1) / base_wg, 16);
  }
  torch::Tensor partial_o, partial_ml;
  float* po = nullptr;
  float* pml = nullptr;
  if (splits > 1) {
    auto opts = q.options().dtype(at::kFloat);
    partial_o = torch::empty({B, H, splits, D}, opts);
    partial_ml = torch::empty({B, H, splits, 2}, opts);
    po = partial_o.data_ptr<float>();
    pml = partial_ml.data_ptr<float>();
  }
  dim3 grid(B, Hkv, splits);

#define LAUNCH_DEC(DD, GG)                                                   \
  hipLaunchKernelGGL((attn_decode_kernel<DD, GG>), grid, dim3(256), 0,       \
                     stream_, (const __hip_bfloat16*)q.data_ptr(),           \
                     (const __hip_bfloat16*)k_cache.data_ptr(),              \
                     (const __hip_bfloat16*)v_cache.data_ptr(),              \
                     bt.data_ptr<int>(), sl.data_ptr<int>(),                 \
                     (__hip_bfloat16*)out.data_ptr(), po, pml, splits,       \
                     q.stride(0), H, Hkv, BS, max_blocks, (float)scale)
#define LAUNCH_DEC4(DD, GG, UU, MM)                                          \
  hipLaunchKernelGGL((attn_decode_kernel<DD, GG, UU, MM>), grid, dim3(256),  \
                     0, stream_, (const __hip_bfloat16*)q.data_ptr(),        \
                     (const __hip_bfloat16*)k_cache.data_ptr(),              \
                     (const __hip_bfloat16*)v_cache.data_ptr(),              \
                     bt.data_ptr<int>(), sl.data_ptr<int>(),                 \
                     (__hip_bfloat16*)out.data_ptr(), po, pml, splits,       \
                     q.stride(0), H, Hkv, BS, max_blocks, (float)scale)
  if (D == 128) {
    if (GQ == 1) LAUNCH_DEC(128, 1);
    else if (GQ == 2) LAUNCH_DEC(128, 2);
    // GQ=4 (llama-3 shapes): occupancy-first variant measured 3-6% faster
    // at B=32-64, S=1-4k (scripts/decode_ab.hip, profiles/decode_ab.txt)
    else if (GQ == 4) LAUNCH_DEC4(128, 4, 2, 4);
    else LAUNCH_DEC(128, 8);
  } else {
    if (GQ == 1) LAUNCH_DEC(64, 1);
    else if (GQ == 2) LAUNCH_DEC(64, 2);
    else if (GQ == 4) LAUNCH_DEC(64, 4);
    else LAUNCH_DEC(64, 8);
  }
#undef LAUNCH_DEC
#undef LAUNCH_DEC4
  if (splits > 1) {
    dim3 cgrid(B, H);
    if (D == 128) {
      hipLaunchKernelGGL(decode_combine_kernel<128>, cgrid, dim3(64), 0,
                         stream_, po, pml, (__hip_bfloat16*)out.data_ptr(),
                         splits, H);
    } else {
      hipLaunchKernelGGL(decode_combine_kernel<64>, cgrid, dim3(64), 0,
                         stream_, po, pml, (__hip_bfloat16*)out.data_ptr(),
                         splits, H);
    }
  }
  return out;
}

void kv_cache_write(torch::Tensor knew, torch::Tensor vnew,
                    torch::Tensor k_cache, torch::Tensor v_cache,
                    torch::Tensor slot_mapping) {
  TORCH_CHECK(knew.dim() == 3, "knew must be [T, Hkv, D]");
  // token-strided views of a merged QKV projection are accepted
  TORCH_CHECK(knew.stride(2) == 1 && knew.stride(1) == knew.size(2));
  TORCH_CHECK(vnew.stride(2) == 1 && vnew.stride(1) == vnew.size(2));
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  const long kts = knew.stride(0), vts = vnew.stride(0);
  const int T = knew.size(0), Hkv = knew.size(1), D = knew.size(2);
  const int BS = k_cache.size(2);
  auto slots = slot_mapping.to(knew.device(), at::kInt).contiguous();
  TORCH_CHECK(slots.numel() == T);
  if (T == 0) return;
  hipStream_t stream_ = cmls::current_stream();
  const auto st = knew.scalar_type();
  const int block = std::min(256, Hkv * D);
  if (st == at::kBFloat16) {
    hipLaunchKernelGGL(kv_cache_write_kernel<__hip_bfloat16>, dim3(T),
                       dim3(block), 0, stream_,
                       (const __hip_bfloat16*)knew.data_ptr(),
                       (const __hip_bfloat16*)vnew.data_ptr(),
                       (__hip_bfloat16*)k_cache.data_ptr(),
                       (__hip_bfloat16*)v_cache.data_ptr(),
                       slots.data_ptr<int>(), Hkv, D, BS, kts, vts);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(kv_cache_write_kernel<__half>, dim3(T), dim3(block), 0,
                       stream_, (const __half*)knew.data_ptr(),
                       (const __half*)vnew.data_ptr(),
                       (__half*)k_cache.data_ptr(),
                       (__half*)v_cache.data_ptr(), slots.data_ptr<int>(),
                       Hkv, D, BS, kts, vts);
  } else {
    TORCH_CHECK(false, "kv_cache_write: unsupported dtype ", st);
  }
}
#endif  // CMLS_KERNEL_ONLY
