// Probe ds_read_b64_tr_b16 semantics on gfx950: fill LDS with identity
// pattern lds[i] = i (ushort), read with different per-lane addresses, dump
// what each lane's 4 elements contain.
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void tr_probe(unsigned short* out, int variant) {
  __shared__ unsigned short lds[2048];
  const int l = threadIdx.x;
  for (int i = l; i < 2048; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  unsigned base = (unsigned)(unsigned long long)(&lds[0]);
  unsigned addr;
  switch (variant) {
    case 0: addr = base; break;                                   // uniform
    case 1: addr = base + 2u * ((l & 15) + ((l >> 4) * 64)); break;  // doc map
    case 2: addr = base + 8u * l; break;                          // like b64
    default: addr = base + 2u * (l & 15); break;                  // col only
  }
  unsigned r0, r1;
  asm volatile("ds_read_b64_tr_b16 v[30:31], %2\n\ts_waitcnt lgkmcnt(0)\n\tv_mov_b32 %0, v30\n\tv_mov_b32 %1, v31"
               : "=v"(r0), "=v"(r1) : "v"(addr) : "v30", "v31", "memory");
  out[(variant * 64 + l) * 4 + 0] = (unsigned short)(r0 & 0xffff);
  out[(variant * 64 + l) * 4 + 1] = (unsigned short)(r0 >> 16);
  out[(variant * 64 + l) * 4 + 2] = (unsigned short)(r1 & 0xffff);
  out[(variant * 64 + l) * 4 + 3] = (unsigned short)(r1 >> 16);
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 4 * 64 * 4 * sizeof(unsigned short));
  for (int v = 0; v < 4; ++v) hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, d, v);
  unsigned short h[4 * 64 * 4];
  hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int v = 0; v < 4; ++v) {
    printf("variant %d:\n", v);
    for (int l = 0; l < 64; l += 1) {
      printf("l%02d:[%4d %4d %4d %4d] ", l, h[(v*64+l)*4], h[(v*64+l)*4+1], h[(v*64+l)*4+2], h[(v*64+l)*4+3]);
      if (l % 4 == 3) printf("\n");
    }
  }
  return 0;
}
