// Empirical probe of ds_read_b64_tr_b16 addressing on gfx950:
// LDS filled with identity (lds_u16[i] = i); each lane supplies an address
// per SCHEME; we dump the 4 u16 elements every lane receives.
#include <hip/hip_runtime.h>
#include <stdio.h>
#include <stdint.h>

__global__ void tr16_probe(unsigned short* out, int scheme) {
  __shared__ __align__(16) unsigned short lds[2048];
  int tid = threadIdx.x;
  for (int i = tid; i < 2048; i += 64) lds[i] = (unsigned short)i;
  // escape hatch: the tr16 asm below only sees an INTEGER offset, so LLVM
  // would otherwise prove the array dead and delete the fill
  asm volatile("" : : "v"((unsigned)(uintptr_t)&lds[0]) : "memory");
  __syncthreads();
  unsigned addr = 0;  // BYTE address into LDS
  int l = tid;
  switch (scheme) {
    case 0: addr = 0; break;                       // uniform
    case 1: addr = l * 2; break;                   // per-lane element
    case 2: addr = l * 8; break;                   // per-lane 8B
    case 3: addr = (l & 15) * 8 + (l >> 4) * 128; break;
    case 4: addr = (l & 15) * 2 + (l >> 4) * 128; break;
  }
  uint2 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr) : "memory");
  __builtin_amdgcn_sched_barrier(0);
  out[tid * 4 + 0] = (unsigned short)(v.x & 0xffff);
  out[tid * 4 + 1] = (unsigned short)(v.x >> 16);
  out[tid * 4 + 2] = (unsigned short)(v.y & 0xffff);
  out[tid * 4 + 3] = (unsigned short)(v.y >> 16);
}

int main() {
  unsigned short* out;
  hipMalloc(&out, 64 * 4 * 2);
  unsigned short host[256];
  for (int s = 0; s <= 4; ++s) {
    hipLaunchKernelGGL(tr16_probe, dim3(1), dim3(64), 0, 0, out, s);
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("scheme %d:\n", s);
    for (int l = 0; l < 64; l += 1) {
      printf("  l%02d: %4d %4d %4d %4d%s", l, host[l*4], host[l*4+1],
             host[l*4+2], host[l*4+3], (l % 4 == 3) ? "\n" : "   ");
    }
    printf("\n");
  }
  return 0;
}
