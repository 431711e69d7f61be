// Empirical probe of ds_read_b64_tr_b16 lane->element mapping on gfx950.
// LDS is filled with lds[i] = i; each lane passes addr = base + lane*4
// elements (8B); the returned 4 u16 values per lane reveal the transpose
// pattern exactly.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 v4bf;

__global__ void probe(unsigned short* out, int addr_mode) {
  __shared__ unsigned short lds[2048];
  int t = threadIdx.x;
  for (int i = t; i < 2048; i += 64) lds[i] = (unsigned short)i;
  __syncthreads();
  int el;
  switch (addr_mode) {
    case 0: el = t * 4; break;                 // linear 4-el per lane
    case 1: el = (t >> 4) * 256 + (t & 15) * 4; break;
    case 2: el = t * 8; break;
    default: el = 0;
  }
  __attribute__((address_space(3))) v4bf* p =
      (__attribute__((address_space(3))) v4bf*)(uintptr_t)(lds + el);
  v4bf v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p);
  union { v4bf v; unsigned short u[4]; } c;
  c.v = v;
  for (int j = 0; j < 4; ++j) out[t * 4 + j] = c.u[j];
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * 2);
  unsigned short h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d (lane: got elements)\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  L%02d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
      if ((l & 15) == 15) printf("\n");
    }
  }
  hipFree(d);
  return 0;
}
