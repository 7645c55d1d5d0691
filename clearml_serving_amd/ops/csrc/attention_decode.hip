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
typedef __attribute__((ext_vector_type(2))) float f32x2_t;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16v2_t;

// 4 e4m3 bytes -> 2 packed-bf16 pairs (v_cvt_scalef32_pk_bf16_fp8: the
// word-select must be a literal constant, scale multiplies the result)
__device__ __forceinline__ void fp8x4_to_bf16x2x2(int w, float sc,
                                                  bf16x2_t* o) {
  bf16v2_t lo = __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(w, sc, false);
  bf16v2_t hi = __builtin_amdgcn_cvt_scalef32_pk_bf16_fp8(w, sc, true);
  o[0] = *reinterpret_cast<bf16x2_t*>(&lo);
  o[1] = *reinterpret_cast<bf16x2_t*>(&hi);
}

// 4 e4m3 bytes -> 4 f32, scaled (one v_cvt_scalef32_pk_f32_fp8 per pair)
__device__ __forceinline__ void fp8x4_to_f32_scaled(int w, float sc,
                                                    float* o) {
  f32x2_t lo = __builtin_amdgcn_cvt_scalef32_pk_f32_fp8(w, sc, false);
  f32x2_t hi = __builtin_amdgcn_cvt_scalef32_pk_f32_fp8(w, sc, true);
  o[0] = lo[0]; o[1] = lo[1]; o[2] = hi[0]; o[3] = hi[1];
}
typedef __attribute__((ext_vector_type(2))) unsigned int uint32x2_t;

// FP8KV: caches hold OCP-e4m3 bytes with per-token-per-head scales
// ([NB, Hkv, BS] f32) -- halves the KV HBM stream AND doubles the cached
// tokens per GB; scores/values dequantize inline (VALU headroom exists:
// the kernel is HBM-bound at 3.4-4.9 TB/s).
template <int D, int GQ, int UNROLL = 4, int MINB = 2, bool FP8KV = false>
__global__ __launch_bounds__(256, MINB) void attn_decode_kernel(
    const __hip_bfloat16* __restrict__ q,        // [B, H, D]
    const void* __restrict__ k_cache,            // [NB, Hkv, BS, D] bf16|e4m3
    const void* __restrict__ v_cache,
    const float* __restrict__ k_scale,           // [NB, Hkv, BS] (FP8KV)
    const float* __restrict__ v_scale,
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
    uint32x2_t k8v[FP8KV ? UNROLL : 1], v8v[FP8KV ? UNROLL : 1];
    float ksc[FP8KV ? UNROLL : 1], vsc[FP8KV ? UNROLL : 1];
    bool valid[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int key = key_lo + (it0 + u) * per_iter + wave * GROUPS + group;
      valid[u] = (it0 + u) < n_iters && key < n_keys;
      const int kc = min(key, n_keys - 1);  // clamped: always a mapped page
      const int blk = btab[kc / block_size];
      const long base = ((long)blk * Hkv) * block_size * D + hk_off +
                        (long)(kc % block_size) * D;
      if (FP8KV) {
        const unsigned char* kcp = (const unsigned char*)k_cache;
        const unsigned char* vcp = (const unsigned char*)v_cache;
        k8v[u] = *reinterpret_cast<const uint32x2_t*>(
            kcp + base + sub * EPL);
        v8v[u] = *reinterpret_cast<const uint32x2_t*>(
            vcp + base + sub * EPL);
        const long sidx = ((long)blk * Hkv + hkv) * block_size +
                          (kc % block_size);
        ksc[u] = k_scale[sidx];
        vsc[u] = v_scale[sidx];
      } else {
        const __hip_bfloat16* kcp = (const __hip_bfloat16*)k_cache;
        const __hip_bfloat16* vcp = (const __hip_bfloat16*)v_cache;
        kvec[u].u = *reinterpret_cast<const uint32x4*>(
            kcp + base + sub * EPL);
        vvec[u].u = *reinterpret_cast<const uint32x4*>(
            vcp + base + sub * EPL);
      }
    }
    // scores for the whole 4-iteration chunk (4x16 = 64 wave keys)
    float score[UNROLL][GQ];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      if (FP8KV) {
        // packed dequant (scale=1; k_scale folds into the score scale) ->
        // the same v_dot2_f32_bf16 dot as the bf16 cache path
        bf16x2_t kb[EPL / 2];
        const int* ki = reinterpret_cast<const int*>(&k8v[u]);
#pragma unroll
        for (int w = 0; w < EPL / 4; ++w)
          fp8x4_to_bf16x2x2(ki[w], 1.0f, kb + 2 * w);
#pragma unroll
        for (int g = 0; g < GQ; ++g) {
          float acc = 0.f;
#pragma unroll
          for (int e = 0; e < EPL / 2; ++e)
            acc = __builtin_amdgcn_fdot2_f32_bf16(qp[g][e], kb[e], acc,
                                                  false);
          score[u][g] = acc;
        }
      } else {
        const bf16x2_t* kp = reinterpret_cast<const bf16x2_t*>(&kvec[u]);
#pragma unroll
        for (int g = 0; g < GQ; ++g) {
          float acc = 0.f;
#pragma unroll
          for (int e = 0; e < EPL / 2; ++e)
            acc = __builtin_amdgcn_fdot2_f32_bf16(qp[g][e], kp[e], acc, false);
          score[u][g] = acc;
        }
      }
      // group-level dot reduction (16 lanes hold partial sums)
#pragma unroll
      for (int g = 0; g < GQ; ++g) {
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
          score[u][g] += __shfl_xor(score[u][g], off, 64);
        const float s2 = FP8KV ? scale2 * ksc[u] : scale2;
        score[u][g] = valid[u] ? score[u][g] * s2 : -INFINITY;
      }
    }
    // ONE online-softmax update per chunk per head (rescaling O per
    // 4-key iteration was the serial VALU chain limiting bandwidth)
    float vfq[FP8KV ? UNROLL : 1][FP8KV ? EPL : 1];
    if (FP8KV) {
      // dequant V ONCE per chunk (not per head): v_scale folds into the
      // packed conversion, so the per-head loop below is plain FMA
#pragma unroll
      for (int u = 0; u < UNROLL; ++u) {
        const int* vi = reinterpret_cast<const int*>(&v8v[u]);
#pragma unroll
        for (int w = 0; w < EPL / 4; ++w)
          fp8x4_to_f32_scaled(vi[w], vsc[u], vfq[u] + 4 * w);
      }
    }
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
        if (FP8KV) {
#pragma unroll
          for (int e = 0; e < EPL; ++e) o_acc[g][e] += p * vfq[u][e];
        } else {
#pragma unroll
          for (int e = 0; e < EPL; ++e)
            o_acc[g][e] += p * to_f32(vvec[u].h[e]);
        }
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

template <typename T>
__global__ void kv_cache_write_fp8_kernel(
    const T* __restrict__ knew,  // [T, Hkv, D] (token-strided views ok)
    const T* __restrict__ vnew,
    unsigned char* __restrict__ k8,  // [NB, Hkv, BS, D] e4m3
    unsigned char* __restrict__ v8,
    float* __restrict__ ks,          // [NB, Hkv, BS]
    float* __restrict__ vs,
    const int* __restrict__ slots, int hkv, int d, int block_size,
    long k_tstride, long v_tstride) {
  const int t = blockIdx.x;
  const int slot = slots[t];
  if (slot < 0) return;
  const long blk = slot / block_size;
  const int off = slot % block_size;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  // one wave per head: per-(token, head) absmax -> scale -> quantize
  for (int h = wave; h < hkv; h += (int)blockDim.x / 64) {
    const T* krow = knew + (long)t * k_tstride + (long)h * d;
    const T* vrow = vnew + (long)t * v_tstride + (long)h * d;
    float kamax = 0.f, vamax = 0.f;
    for (int i = lane; i < d; i += 64) {
      kamax = fmaxf(kamax, fabsf(to_f32(krow[i])));
      vamax = fmaxf(vamax, fabsf(to_f32(vrow[i])));
    }
    kamax = wave_reduce_max(kamax);
    vamax = wave_reduce_max(vamax);
    // POWER-OF-2 scales: v_cvt_scalef32_pk_* applies only the scale's
    // exponent (MX semantics), so pow2 makes the hardware dequant exact;
    // e4m3 is itself a float format, so the rounded-up scale costs no
    // relative precision (values keep >= half the exponent range)
    const float kscale = exp2f(ceilf(log2f(fmaxf(kamax, 1e-8f) / 448.0f)));
    const float vscale = exp2f(ceilf(log2f(fmaxf(vamax, 1e-8f) / 448.0f)));
    const long sidx = (blk * hkv + h) * block_size + off;
    if (lane == 0) {
      ks[sidx] = kscale;
      vs[sidx] = vscale;
    }
    const float kinv = 1.0f / kscale, vinv = 1.0f / vscale;
    unsigned char* kd = k8 + ((blk * hkv + h) * block_size + off) * (long)d;
    unsigned char* vd = v8 + ((blk * hkv + h) * block_size + off) * (long)d;
    for (int i = lane; i < d; i += 64) {
      const float kq = to_f32(krow[i]) * kinv;
      const float vq = to_f32(vrow[i]) * vinv;
      kd[i] = __builtin_amdgcn_cvt_pk_fp8_f32(kq, kq, 0, false) & 0xff;
      vd[i] = __builtin_amdgcn_cvt_pk_fp8_f32(vq, vq, 0, false) & 0xff;
    }
  }
}

}  // namespace

#ifndef CMLS_KERNEL_ONLY
torch::Tensor attention_decode(torch::Tensor q, torch::Tensor k_cache,
                               torch::Tensor v_cache,
                               torch::Tensor block_table,
                               torch::Tensor seq_lens, double scale,
                               c10::optional<torch::Tensor> k_scale,
                               c10::optional<torch::Tensor> v_scale) {
  TORCH_CHECK(q.dim() == 3, "q must be [B, H, D]");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "decode attention: bf16 only");
  TORCH_CHECK(q.stride(2) == 1 && q.stride(1) == q.size(2),
              "q heads must be dense");
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  const bool fp8kv = k_cache.scalar_type() == at::kByte ||
                     k_cache.scalar_type() == at::kFloat8_e4m3fn;
  const float* ksp = nullptr;
  const float* vsp = nullptr;
  if (fp8kv) {
    TORCH_CHECK(k_scale.has_value() && v_scale.has_value(),
                "fp8 KV cache needs k_scale/v_scale");
    TORCH_CHECK(k_scale->is_contiguous() && v_scale->is_contiguous());
    ksp = k_scale->data_ptr<float>();
    vsp = v_scale->data_ptr<float>();
  }
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int Hkv = k_cache.size(1), BS = k_cache.size(2);
  const int max_blocks = block_table.size(1);
  const int GQ = H / Hkv;
  TORCH_CHECK(H % Hkv == 0 && (D == 64 || D == 128));
  TORCH_CHECK(GQ >= 1 && GQ <= 8,
              "GQA group must be 1..8, got ", GQ);
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

#define LAUNCH_DEC(DD, GG, UU, MM, F8)                                       \
  hipLaunchKernelGGL((attn_decode_kernel<DD, GG, UU, MM, F8>), grid,         \
                     dim3(256), 0,                                           \
                     stream_, (const __hip_bfloat16*)q.data_ptr(),           \
                     (const void*)k_cache.data_ptr(),                        \
                     (const void*)v_cache.data_ptr(), ksp, vsp,              \
                     bt.data_ptr<int>(), sl.data_ptr<int>(),                 \
                     (__hip_bfloat16*)out.data_ptr(), po, pml, splits,       \
                     q.stride(0), H, Hkv, BS, max_blocks, (float)scale)
#define LAUNCH_DEC_F(DD, GG, UU, MM)                                         \
  do { if (fp8kv) LAUNCH_DEC(DD, GG, UU, MM, true);                          \
       else LAUNCH_DEC(DD, GG, UU, MM, false); } while (0)
  // GQ 3/5/6/7 (qwen2-7b is 28q/4kv = 7, qwen2-1.5b 12q/2kv = 6) run the
  // same kernel with shallower UNROLL: per-group accumulator arrays
  // (o_acc[GQ][8] + score[UNROLL][GQ]) scale VGPRs linearly with GQ
  if (D == 128) {
    if (GQ == 1) LAUNCH_DEC_F(128, 1, 4, 2);
    else if (GQ == 2) LAUNCH_DEC_F(128, 2, 4, 2);
    else if (GQ == 3) LAUNCH_DEC_F(128, 3, 4, 2);
    // GQ=4 (llama-3 shapes): occupancy-first variant measured 3-6% faster
    // at B=32-64, S=1-4k (scripts/decode_ab.hip, profiles/decode_ab.txt)
    else if (GQ == 4) LAUNCH_DEC_F(128, 4, 2, 4);
    else if (GQ == 5) LAUNCH_DEC_F(128, 5, 2, 2);
    else if (GQ == 6) LAUNCH_DEC_F(128, 6, 2, 2);
    else if (GQ == 7) LAUNCH_DEC_F(128, 7, 2, 2);
    else LAUNCH_DEC_F(128, 8, 4, 2);
  } else {
    if (GQ == 1) LAUNCH_DEC_F(64, 1, 4, 2);
    else if (GQ == 2) LAUNCH_DEC_F(64, 2, 4, 2);
    else if (GQ == 3) LAUNCH_DEC_F(64, 3, 4, 2);
    else if (GQ == 4) LAUNCH_DEC_F(64, 4, 4, 2);
    else if (GQ == 5) LAUNCH_DEC_F(64, 5, 4, 2);
    else if (GQ == 6) LAUNCH_DEC_F(64, 6, 4, 2);
    else if (GQ == 7) LAUNCH_DEC_F(64, 7, 4, 2);
    else LAUNCH_DEC_F(64, 8, 4, 2);
  }
#undef LAUNCH_DEC_F
#undef LAUNCH_DEC
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
                    torch::Tensor slot_mapping,
                    c10::optional<torch::Tensor> k_scale,
                    c10::optional<torch::Tensor> v_scale) {
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
  const bool fp8kv = k_cache.scalar_type() == at::kByte ||
                     k_cache.scalar_type() == at::kFloat8_e4m3fn;
  if (fp8kv) {
    TORCH_CHECK(k_scale.has_value() && v_scale.has_value(),
                "fp8 KV cache needs k_scale/v_scale");
    TORCH_CHECK(st == at::kBFloat16, "fp8 KV write: bf16 source only");
    hipLaunchKernelGGL(kv_cache_write_fp8_kernel<__hip_bfloat16>, dim3(T),
                       dim3(std::min(256, 64 * Hkv)), 0, stream_,
                       (const __hip_bfloat16*)knew.data_ptr(),
                       (const __hip_bfloat16*)vnew.data_ptr(),
                       (unsigned char*)k_cache.data_ptr(),
                       (unsigned char*)v_cache.data_ptr(),
                       k_scale->data_ptr<float>(),
                       v_scale->data_ptr<float>(),
                       slots.data_ptr<int>(), Hkv, D, BS, kts, vts);
    return;
  }
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
