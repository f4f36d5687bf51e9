// Semantics probe for gfx950 ds_read_b64_tr_b16.
// LDS filled with lds[i] = i (u16). Each lane reads 64b at its own address;
// dump the 4 u16 element indices each lane receives, for several address maps.
#include <hip/hip_runtime.h>
#include <cstdio>

__device__ __forceinline__ uint2 tr16(unsigned off_bytes) {
  uint2 r;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(r) : "v"(off_bytes));
  return r;
}

__global__ void probe(unsigned short* out, int mapsel) {
  __shared__ unsigned short lds[8192];
  for (int i = threadIdx.x; i < 8192; i += blockDim.x) lds[i] = i;
  __syncthreads();
  const int l = threadIdx.x;
  unsigned base = (unsigned)(unsigned long long)lds;
  unsigned elem;
  switch (mapsel) {
    case 0: elem = l * 4; break;                      // linear: lane*8B
    case 1: elem = (l & 15) * 4 + (l >> 4) * 64; break; // guide image: (l&15)+j*16+(l>>4)*64 view
    case 2: elem = (l & 15) * 4; break;               // same 64B window per 16-lane group
    default: elem = l * 8; break;                     // strided 16B
  }
  uint2 r = tr16(base + elem * 2);
  union { uint2 u; unsigned short s[4]; } c; c.u = r;
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = c.s[j];
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * 2);
  unsigned short h[256];
  for (int m = 0; m < 4; ++m) {
    probe<<<1, 64>>>(d, m);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("map %d:\n", m);
    for (int l = 0; l < 64; ++l) {
      printf("L%02d:[%4d %4d %4d %4d] ", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
      if (l % 4 == 3) printf("\n");
    }
  }
  hipFree(d);
  return 0;
}
