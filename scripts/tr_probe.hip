// Probe ds_read_b64_tr_b16 semantics on gfx950: fill LDS with element
// indices, read with several per-lane address patterns, dump what each
// lane's 4 output elements contain. Standalone: hipcc -o tr_probe
// scripts/tr_probe.hip --offload-arch=gfx950 && ./tr_probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef unsigned short u16;
typedef unsigned long long u64;

__global__ void probe(u16* out, int pattern) {
  __shared__ __attribute__((aligned(16))) u16 lds[4096];
  int tid = threadIdx.x;  // 64 lanes
  for (int i = tid; i < 4096; i += 64) lds[i] = (u16)i;
  __syncthreads();
  unsigned addr;
  // LDS byte offset of element idx:
  auto off = [&](int idx) {
    return (unsigned)(
        (char*)(__attribute__((address_space(3))) char*)(&lds[idx]) -
        (char*)(__attribute__((address_space(3))) char*)(&lds[0]));
  };
  unsigned base = off(0);
  if (pattern == 0) addr = base + tid * 8;           // lane-linear 8B
  else if (pattern == 1) addr = base + (tid & 15) * 8;  // same per group? no: 16 lanes span 128B
  else if (pattern == 2) addr = base + (tid >> 4) * 128;  // uniform within group
  else addr = base + (tid & 15) * 8 + (tid >> 4) * 128;
  u64 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
  __builtin_amdgcn_sched_barrier(0);
  out[tid * 4 + 0] = (u16)(v & 0xffff);
  out[tid * 4 + 1] = (u16)((v >> 16) & 0xffff);
  out[tid * 4 + 2] = (u16)((v >> 32) & 0xffff);
  out[tid * 4 + 3] = (u16)((v >> 48) & 0xffff);
}

int main() {
  u16* d;
  hipMalloc(&d, 64 * 4 * sizeof(u16));
  u16 h[256];
  for (int pat = 0; pat < 4; ++pat) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, pat);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("pattern %d:\n", pat);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
    }
  }
  hipFree(d);
  return 0;
}
