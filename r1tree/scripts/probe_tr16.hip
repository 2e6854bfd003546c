// Reverse-engineer ds_read_b64_tr_b16: fill LDS with element-index encodings
// and dump what each lane's 4 elements actually are, for (a) uniform base
// address and (b) per-lane addresses base + (lane&3)*32.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

typedef __hip_bfloat16 bf16;
using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

__global__ void tr_probe(float* out, int mode) {
  __shared__ __attribute__((aligned(16))) bf16 lds[4096];
  const int lane = threadIdx.x & 63;
  // fill: element e -> lo byte in pass 0, hi byte in pass 1 handled by host:
  // encode e as lo + hi*256 via two kernel calls (mode&2)
  for (int e = lane; e < 4096; e += 64)
    lds[e] = (bf16)(float)((mode & 2) ? (e >> 8) : (e & 255));
  __builtin_amdgcn_s_barrier();
  unsigned addr0 = (unsigned)(uintptr_t)(
      (__attribute__((address_space(3))) bf16*)lds);
  unsigned a = addr0;
  if (mode & 1) a += (lane & 3) * 32;     // per-lane row step within cluster
  u32x2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(r) : "v"(a));
  __builtin_amdgcn_sched_barrier(0);
  union { u32x2 u; __attribute__((ext_vector_type(4))) __bf16 v; } c;
  c.u = r;
  for (int j = 0; j < 4; ++j)
    out[lane * 4 + j] = __bfloat162float((bf16)c.v[j]);
}

int main() {
  float* d;
  hipMalloc(&d, 64 * 4 * sizeof(float));
  float lo[256], hi[256];
  for (int mode = 0; mode < 2; ++mode) {
    hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(lo, d, sizeof lo, hipMemcpyDeviceToHost);
    hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, d, mode | 2);
    hipMemcpy(hi, d, sizeof hi, hipMemcpyDeviceToHost);
    printf("=== mode %d (%s) ===\n", mode,
           mode ? "addr += (lane&3)*32" : "uniform addr");
    for (int l = 0; l < 20; ++l) {
      printf("lane %2d: ", l);
      for (int j = 0; j < 4; ++j)
        printf("%5d", (int)lo[l * 4 + j] + 256 * (int)hi[l * 4 + j]);
      printf("\n");
    }
    printf("lane 32: ");
    for (int j = 0; j < 4; ++j)
      printf("%5d", (int)lo[32 * 4 + j] + 256 * (int)hi[32 * 4 + j]);
    printf("\nlane 48: ");
    for (int j = 0; j < 4; ++j)
      printf("%5d", (int)lo[48 * 4 + j] + 256 * (int)hi[48 * 4 + j]);
    printf("\n");
  }
  return 0;
}
