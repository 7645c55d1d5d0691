// Token sampling: temperature + gumbel-argmax categorical draw.
//
// The categorical sample over softmax(logits/T) is taken with the gumbel-max
// trick in ONE pass over the vocabulary -- no softmax materialization, no
// sort: argmax(logits/T + G_i), G_i = -log(-log(U_i)). top-k/top-p filtering
// (when requested) happens before this kernel (torch.topk/sort composition in
// the C++ wrapper); the hot path (plain sampling, the serving default) is a
// single kernel.
//
// Occupancy: decode batches are small (B <= 64 rows), so the vocabulary is
// split across gridDim.y workgroups (B workgroups alone leave 3/4 of the 256
// CUs idle); per-split winners combine through a packed (value, index)
// 64-bit atomicMax -- deterministic, one extra 8-B atomic per workgroup.
//
// Replaces the sampling step the reference delegates to vLLM
// ("token-sampling kernel (top-k/top-p)").
#include "common.h"

namespace {

__device__ __forceinline__ unsigned int pcg_hash(unsigned int x) {
  x = x * 747796405u + 2891336453u;
  unsigned int w = ((x >> ((x >> 28u) + 4u)) ^ x) * 277803737u;
  return (w >> 22u) ^ w;
}

// order-preserving float -> uint32 (works for +/-, inf)
__device__ __forceinline__ unsigned int float_orderable(float f) {
  unsigned int u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

template <typename T>
__global__ void gumbel_argmax_kernel(const T* __restrict__ logits,
                                     unsigned long long* __restrict__ best,
                                     int rows, int vocab, float inv_temp,
                                     unsigned int seed) {
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* lrow = logits + (long)row * vocab;
  const int chunk = (vocab + gridDim.y - 1) / gridDim.y;
  const int lo = blockIdx.y * chunk;
  const int hi = min(vocab, lo + chunk);
  const unsigned int row_seed = pcg_hash(seed ^ ((unsigned)row * 9781u));

  float bv = -INFINITY;
  int bi = lo;
  for (int idx = lo + threadIdx.x; idx < hi; idx += blockDim.x) {
    const float lg = to_f32(lrow[idx]);
    if (lg == -INFINITY) continue;  // filtered token
    const unsigned int h = pcg_hash(row_seed + (unsigned)idx);
    const float u = ((float)h + 1.0f) * (1.0f / 4294967296.0f);
    const float g = -__logf(-__logf(u));
    const float s = lg * inv_temp + g;
    if (s > bv) { bv = s; bi = idx; }
  }
  // wave reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(bv, off, 64);
    const int oi = __shfl_xor(bi, off, 64);
    if (ov > bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
  }
  __shared__ float svals[16];
  __shared__ int sidx[16];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  if (lane == 0) { svals[wave] = bv; sidx[wave] = bi; }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = blockDim.x >> 6;
    for (int w = 1; w < nwaves; ++w) {
      if (svals[w] > bv || (svals[w] == bv && sidx[w] < bi)) {
        bv = svals[w];
        bi = sidx[w];
      }
    }
    const unsigned long long packed =
        ((unsigned long long)float_orderable(bv) << 32) |
        (unsigned int)(vocab - 1 - bi);  // tie -> lowest index wins
    atomicMax(&best[row], packed);
  }
}

__global__ void unpack_kernel(const unsigned long long* __restrict__ best,
                              long* __restrict__ out, int rows, int vocab) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row < rows) out[row] = vocab - 1 - (int)(best[row] & 0xFFFFFFFFu);
}

}  // namespace

torch::Tensor sample_top_k_top_p(torch::Tensor logits, double temperature,
                                 long top_k, double top_p, long seed) {
  TORCH_CHECK(logits.dim() == 2, "logits must be [B, V]");
  CHECK_LASTDIM_CONTIG(logits);
  const int rows = logits.size(0);
  const int vocab = logits.size(1);
  TORCH_CHECK(temperature > 0.0, "temperature must be > 0 (greedy is argmax)");

  torch::Tensor filtered = logits;
  // top-k / top-p filtering composed from library selection ops; the draw
  // itself is the hand-written kernel below.
  if (top_k > 0 && top_k < vocab) {
    auto kth = std::get<0>(filtered.topk(top_k, -1)).select(-1, top_k - 1)
                   .unsqueeze(-1);
    filtered = filtered.where(filtered >= kth,
                              torch::full({}, -INFINITY, filtered.options()));
  }
  if (top_p < 1.0) {
    auto sorted = filtered.sort(-1, /*descending=*/true);
    auto sl = std::get<0>(sorted);
    auto si = std::get<1>(sorted);
    auto probs = (sl.to(at::kFloat) / temperature).softmax(-1);
    // drop tokens whose inclusive cumulative prob exceeds p; keep the top
    // token unconditionally (matches the python reference in ops/__init__.py)
    auto cut = probs.cumsum(-1) > top_p;
    cut.index_put_({torch::indexing::Slice(), 0}, false);
    auto masked = sl.where(~cut, torch::full({}, -INFINITY, sl.options()));
    filtered = torch::full_like(filtered, -INFINITY)
                   .scatter(-1, si, masked);
  }
  filtered = filtered.contiguous();

  auto best = torch::zeros(
      {rows}, logits.options().dtype(at::kLong));  // packed (val, idx)
  auto out = torch::empty({rows}, logits.options().dtype(at::kLong));
  const int vsplits =
      std::max(1, std::min(16, 1024 / std::max(rows, 1)));
  dim3 grid(rows, vsplits);
  const int block = 256;
  hipStream_t stream_ = cmls::current_stream();
  const auto st = filtered.scalar_type();
  const float inv_t = 1.0f / (float)temperature;
  auto* bptr = reinterpret_cast<unsigned long long*>(best.data_ptr<long>());
  if (st == at::kBFloat16) {
    hipLaunchKernelGGL(gumbel_argmax_kernel<__hip_bfloat16>, grid,
                       dim3(block), 0, stream_,
                       (const __hip_bfloat16*)filtered.data_ptr(), bptr,
                       rows, vocab, inv_t, (unsigned int)seed);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(gumbel_argmax_kernel<__half>, grid, dim3(block), 0,
                       stream_, (const __half*)filtered.data_ptr(), bptr,
                       rows, vocab, inv_t, (unsigned int)seed);
  } else {
    auto f = filtered.to(at::kFloat).contiguous();
    hipLaunchKernelGGL(gumbel_argmax_kernel<float>, grid, dim3(block), 0,
                       stream_, (const float*)f.data_ptr(), bptr, rows,
                       vocab, inv_t, (unsigned int)seed);
  }
  hipLaunchKernelGGL(unpack_kernel, dim3((rows + 63) / 64), dim3(64), 0,
                     stream_, bptr, out.data_ptr<long>(), rows, vocab);
  return out;
}
