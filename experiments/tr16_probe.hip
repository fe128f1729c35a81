// Empirical semantics probe for gfx950 ds_read_b64_tr_b16: fill LDS with
// identifiable raw u16 bit patterns (value = element index), have each of
// 64 lanes read 8 bytes at a chosen address, and dump what arrives where.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) unsigned short u16x4;

__global__ void probe(unsigned short* out, int mode) {
  __shared__ unsigned short lds[1024];
  const int t = threadIdx.x;
  for (int i = t; i < 1024; i += blockDim.x) lds[i] = (unsigned short)i;
  __syncthreads();
  if (t >= 64) return;  // one wave
  unsigned base = (unsigned)(unsigned long long)
      (__attribute__((address_space(3))) unsigned short*)&lds[0];
  unsigned addr;
  switch (mode) {
    case 0: addr = base + t * 8; break;              // lane-linear 8B
    case 1: addr = base + (t & 15) * 8 + (t >> 4) * 128; break;
    case 2: addr = base + (t & 15) * 64; break;      // 16 lanes, 64B stride
    default: addr = base + t * 16; break;
  }
  u16x4 v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\ns_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
  out[t * 4 + 0] = v[0];
  out[t * 4 + 1] = v[1];
  out[t * 4 + 2] = v[2];
  out[t * 4 + 3] = v[3];
}

int main() {
  unsigned short* d;
  hipMalloc(&d, 64 * 4 * 2);
  unsigned short h[256];
  for (int mode = 0; mode < 4; ++mode) {
    hipMemset(d, 0xff, 512);
    probe<<<1, 256>>>(d, mode);
    hipMemcpy(h, d, 512, hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("  lane %2d: %4d %4d %4d %4d\n", l, h[l*4], h[l*4+1], h[l*4+2],
             h[l*4+3]);
      if (l == 19 && mode != 2) { printf("  ...\n"); l = 47; }
    }
  }
  return 0;
}
