#include "hip/hip_runtime.h"
// Token sampling: temperature + gumbel-argmax categorical draw.
//
// The categorical sample over softmax(logits/T) is taken with the gumbel-max
// trick in ONE pass over the vocabulary -- no softmax materialization, no
// sort: argmax(logits/T + G_i), G_i = -log(-log(U_i)). top-k/top-p filtering
// (when requested) happens before this kernel (torch.topk/sort composition in
// the C++ wrapper); the hot path (plain sampling, the serving default) is a
// single kernel.
//
// Replaces the sampling step the reference delegates to vLLM
// ("token-sampling kernel (top-k/top-p)").
#include "common.h"

namespace {

// counter-based RNG: one 32-bit hash per (seed, row, col) -- statistically
// adequate for sampling draws (not for cryptography).
__device__ __forceinline__ unsigned int pcg_hash(unsigned int x) {
  x = x * 747796405u + 2891336453u;
  unsigned int w = ((x >> ((x >> 28u) + 4u)) ^ x) * 277803737u;
  return (w >> 22u) ^ w;
}

template <typename T>
__global__ void gumbel_argmax_kernel(const T* __restrict__ logits,
                                     long* __restrict__ out, int rows,
                                     int vocab, float inv_temp,
                                     unsigned int seed) {
  extern __shared__ float tmp[];
  const int row = blockIdx.x;
  if (row >= rows) return;
  const T* lrow = logits + (long)row * vocab;

  float best = -INFINITY;
  int best_idx = 0;
  for (int idx = threadIdx.x; idx < vocab; idx += blockDim.x) {
    const float lg = to_f32(lrow[idx]);
    if (lg == -INFINITY) continue;  // filtered token
    unsigned int h = pcg_hash(seed ^ pcg_hash((unsigned)row * 9781u + idx));
    // u in (0, 1): avoid exactly 0
    const float u = ((float)h + 1.0f) * (1.0f / 4294967296.0f);
    const float g = -__logf(-__logf(u));
    const float s = lg * inv_temp + g;
    if (s > best) { best = s; best_idx = idx; }
  }
  // block argmax: pack value+lane, reduce via shfl then LDS
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int nwaves = blockDim.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(best_idx, off, 64);
    if (ov > best) { best = ov; best_idx = oi; }
  }
  float* vals = tmp;
  int* idxs = (int*)(tmp + 16);
  if (lane == 0) { vals[wave] = best; idxs[wave] = best_idx; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < nwaves; ++w) {
      if (vals[w] > best) { best = vals[w]; best_idx = idxs[w]; }
    }
    out[row] = best_idx;
  }
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

  auto out = torch::empty({rows}, logits.options().dtype(at::kLong));
  const int block = 256;
  const int smem = 16 * sizeof(float) + 16 * sizeof(int);
  hipStream_t stream_ = cmls::current_stream();
  const auto st = filtered.scalar_type();
  const float inv_t = 1.0f / (float)temperature;
  if (st == at::kBFloat16) {
    hipLaunchKernelGGL(gumbel_argmax_kernel<__hip_bfloat16>, dim3(rows),
                       dim3(block), smem, stream_,
                       (const __hip_bfloat16*)filtered.data_ptr(),
                       out.data_ptr<long>(), rows, vocab, inv_t,
                       (unsigned int)seed);
  } else if (st == at::kHalf) {
    hipLaunchKernelGGL(gumbel_argmax_kernel<__half>, dim3(rows), dim3(block),
                       smem, stream_,
                       (const __half*)filtered.data_ptr(),
                       out.data_ptr<long>(), rows, vocab, inv_t,
                       (unsigned int)seed);
  } else {
    auto f = filtered.to(at::kFloat).contiguous();
    hipLaunchKernelGGL(gumbel_argmax_kernel<float>, dim3(rows), dim3(block),
                       smem, stream_, (const float*)f.data_ptr(),
                       out.data_ptr<long>(), rows, vocab, inv_t,
                       (unsigned int)seed);
  }
  return out;
}
