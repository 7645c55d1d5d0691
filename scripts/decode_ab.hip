// A/B harness for the paged decode attention kernel variants
// (clearml_serving_amd/ops/csrc/attention_decode.hip):
//   A = <D, GQ, UNROLL=4, MINB=2>  production: 8 loads in flight/lane,
//       148 VGPR -> 3 waves/SIMD at GQ=4
//   B = <D, GQ, UNROLL=2, MINB=4>  occupancy-first: 4 loads in flight,
//       116 VGPR -> 4 waves/SIMD
// Times both at serving-shaped configs and cross-checks outputs.
// Build: hipcc --offload-arch=gfx950 -O3 scripts/decode_ab.hip -o /tmp/decode_ab
#define CMLS_KERNEL_ONLY
#include "../clearml_serving_amd/ops/csrc/attention_decode.hip"

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CK(x)                                                       \
  do {                                                              \
    hipError_t e = (x);                                             \
    if (e != hipSuccess) {                                          \
      fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e),   \
              __LINE__);                                            \
      exit(1);                                                      \
    }                                                               \
  } while (0)

constexpr int D = 128, GQ = 4, HKV = 8, BS = 16;
constexpr int H = HKV * GQ;

struct Buffers {
  __hip_bfloat16 *q, *k, *v, *out;
  int *btab, *slens;
  float *po, *pml;
  int B, S, nblocks, max_blocks, splits;
};

static int pick_splits(int B) {
  long base_wg = (long)B * HKV;
  if (base_wg >= 1024) return 1;
  return (int)std::min<long>((1024 + base_wg - 1) / base_wg, 16);
}

Buffers make(int B, int S) {
  Buffers bu;
  bu.B = B;
  bu.S = S;
  bu.max_blocks = (S + BS - 1) / BS;
  bu.nblocks = B * bu.max_blocks;
  bu.splits = pick_splits(B);
  CK(hipMalloc(&bu.q, (size_t)B * H * D * 2));
  CK(hipMalloc(&bu.k, (size_t)bu.nblocks * HKV * BS * D * 2));
  CK(hipMalloc(&bu.v, (size_t)bu.nblocks * HKV * BS * D * 2));
  CK(hipMalloc(&bu.out, (size_t)B * H * D * 2));
  CK(hipMalloc(&bu.btab, (size_t)B * bu.max_blocks * 4));
  CK(hipMalloc(&bu.slens, (size_t)B * 4));
  CK(hipMalloc(&bu.po, (size_t)B * H * bu.splits * D * 4));
  CK(hipMalloc(&bu.pml, (size_t)B * H * bu.splits * 2 * 4));

  std::vector<__hip_bfloat16> hq((size_t)B * H * D);
  srand(42);
  for (auto& x : hq) x = __float2bfloat16((rand() / (float)RAND_MAX - .5f));
  CK(hipMemcpy(bu.q, hq.data(), hq.size() * 2, hipMemcpyHostToDevice));
  std::vector<__hip_bfloat16> hk((size_t)bu.nblocks * HKV * BS * D);
  for (auto& x : hk) x = __float2bfloat16((rand() / (float)RAND_MAX - .5f));
  CK(hipMemcpy(bu.k, hk.data(), hk.size() * 2, hipMemcpyHostToDevice));
  for (auto& x : hk) x = __float2bfloat16((rand() / (float)RAND_MAX - .5f));
  CK(hipMemcpy(bu.v, hk.data(), hk.size() * 2, hipMemcpyHostToDevice));
  std::vector<int> bt((size_t)B * bu.max_blocks);
  for (int b = 0; b < B; ++b)
    for (int i = 0; i < bu.max_blocks; ++i)
      bt[(size_t)b * bu.max_blocks + i] = b * bu.max_blocks + i;
  CK(hipMemcpy(bu.btab, bt.data(), bt.size() * 4, hipMemcpyHostToDevice));
  std::vector<int> sl(B, S);
  CK(hipMemcpy(bu.slens, sl.data(), sl.size() * 4, hipMemcpyHostToDevice));
  return bu;
}

template <int UNROLL, int MINB>
void run_variant(const Buffers& bu) {
  dim3 grid(bu.B, HKV, bu.splits);
  float scale = 1.f / sqrtf((float)D);
  hipLaunchKernelGGL((attn_decode_kernel<D, GQ, UNROLL, MINB>), grid,
                     dim3(256), 0, 0, bu.q, bu.k, bu.v, bu.btab, bu.slens,
                     bu.out, bu.splits > 1 ? bu.po : nullptr,
                     bu.splits > 1 ? bu.pml : nullptr, bu.splits,
                     (long)H * D, H, HKV, BS, bu.max_blocks, scale);
  if (bu.splits > 1) {
    hipLaunchKernelGGL(decode_combine_kernel<D>, dim3(bu.B, H), dim3(64), 0,
                       0, bu.po, bu.pml, bu.out, bu.splits, H);
  }
}

template <int UNROLL, int MINB>
double bench(const Buffers& bu, int iters) {
  for (int i = 0; i < 10; ++i) run_variant<UNROLL, MINB>(bu);
  CK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  CK(hipEventCreate(&t0));
  CK(hipEventCreate(&t1));
  CK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) run_variant<UNROLL, MINB>(bu);
  CK(hipEventRecord(t1));
  CK(hipEventSynchronize(t1));
  float ms;
  CK(hipEventElapsedTime(&ms, t0, t1));
  CK(hipEventDestroy(t0));
  CK(hipEventDestroy(t1));
  return ms / iters;
}

int main() {
  int cfgs[][2] = {{8, 4096}, {32, 1024}, {32, 4096}, {64, 1024},
                   {64, 4096}, {64, 8192}};
  printf("%-14s %8s %12s %12s %8s %10s\n", "config", "splits", "A(u4,b2)",
         "B(u2,b4)", "B/A", "maxdiff");
  for (auto& c : cfgs) {
    int B = c[0], S = c[1];
    Buffers bu = make(B, S);
    // cross-check outputs first
    std::vector<__hip_bfloat16> oa((size_t)B * H * D), ob(oa.size());
    run_variant<4, 2>(bu);
    CK(hipDeviceSynchronize());
    CK(hipMemcpy(oa.data(), bu.out, oa.size() * 2, hipMemcpyDeviceToHost));
    run_variant<2, 4>(bu);
    CK(hipDeviceSynchronize());
    CK(hipMemcpy(ob.data(), bu.out, ob.size() * 2, hipMemcpyDeviceToHost));
    float md = 0;
    for (size_t i = 0; i < oa.size(); ++i)
      md = fmaxf(md, fabsf(__bfloat162float(oa[i]) - __bfloat162float(ob[i])));
    double a = bench<4, 2>(bu, 100), b = bench<2, 4>(bu, 100);
    // bytes: K+V streamed once per kv head's workgroup
    double gb = (double)B * HKV * S * D * 2 * 2 / 1e9;
    printf("B=%-3d S=%-6d %6d %7.3fms/%5.0fGB/s %7.3fms/%5.0fGB/s %6.2fx %8.4f\n",
           B, S, bu.splits, a, gb / (a / 1e3), b, gb / (b / 1e3), a / b, md);
    CK(hipFree(bu.q)); CK(hipFree(bu.k)); CK(hipFree(bu.v));
    CK(hipFree(bu.out)); CK(hipFree(bu.btab)); CK(hipFree(bu.slens));
    CK(hipFree(bu.po)); CK(hipFree(bu.pml));
  }
  return 0;
}
