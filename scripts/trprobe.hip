// Empirical semantics probe for ds_read_b64_tr_b16 on gfx950.
// LDS holds element-index values; dump which elements land in each lane.
#include <hip/hip_runtime.h>

#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short bf16x4_t;

__global__ void probe(short* out, int mode) {
  __shared__ short lds[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  int idx;
  if (mode == 0) idx = lane * 4;                       // linear 8B per lane
  else if (mode == 1) idx = (lane & 15) * 4 + (lane >> 4) * 64;
  else idx = (lane >> 4) * 64;                         // group-uniform
  unsigned addr = (unsigned)(unsigned long long)&lds[idx];
  bf16x4_t v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr) : "memory");
  for (int e = 0; e < 4; ++e) out[threadIdx.x * 4 + e] = v[e];
}

int main() {
  short* out;
  hipMalloc(&out, 64 * 4 * sizeof(short));
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out, mode);
    short host[256];
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, host[l * 4],
             host[l * 4 + 1], host[l * 4 + 2], host[l * 4 + 3]);
    }
  }
  hipFree(out);
  return 0;
}
