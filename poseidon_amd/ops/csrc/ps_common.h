// Common helpers for poseidon_amd CDNA4 (gfx950) kernels.
// Hardware model per /opt/skills/guides/MI355X_MICROARCH.md:
//   256 CUs in 8 XCDs, wave64, 160 KiB LDS/CU, MFMA bf16 2.5 PF dense /
//   f32-in 155 TF, HBM3E ~6.3 TB/s achievable.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>
#include <cstdio>

#define PS_HIP_CHECK(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      fprintf(stderr, "HIP error %s at %s:%d: %s\n", hipGetErrorString(_e),  \
              __FILE__, __LINE__, #expr);                                    \
      abort();                                                               \
    }                                                                        \
  } while (0)

namespace ps {

constexpr int kWave = 64;  // CDNA wavefront width (never 32)

__host__ __device__ inline int cdiv(int a, int b) { return (a + b - 1) / b; }
__host__ __device__ inline int64_t cdiv64(int64_t a, int64_t b) { return (a + b - 1) / b; }

// Grid-stride elementwise launch geometry: cap blocks, stride the rest
// (guide §6 Guideline 11).
inline dim3 ew_grid(int64_t n, int block = 256) {
  int64_t blocks = cdiv64(n, block);
  if (blocks > 2048) blocks = 2048;
  return dim3((unsigned)blocks);
}

// vector types
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

using bf16 = __hip_bfloat16;

// Philox4x32-10 counter-based RNG (graph-safe, reproducible): replaces
// curand uniform for dropout masks + stochastic pooling
// (reference math_functions.cu:379-421).
__device__ inline void philox_round(uint32_t& c0, uint32_t& c1, uint32_t& c2,
                                    uint32_t& c3, uint32_t k0, uint32_t k1) {
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t h0 = __umulhi(M0, c0), l0 = M0 * c0;
  uint32_t h1 = __umulhi(M1, c2), l1 = M1 * c2;
  uint32_t n0 = h1 ^ c1 ^ k0, n1 = l1, n2 = h0 ^ c3 ^ k1, n3 = l0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

__device__ inline uint4 philox4(uint64_t seed, uint64_t offset, uint32_t idx) {
  uint32_t c0 = idx, c1 = (uint32_t)offset, c2 = (uint32_t)(offset >> 32), c3 = 0;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
  const uint32_t B0 = 0x9E3779B9u, B1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += B0; k1 += B1;
  }
  return make_uint4(c0, c1, c2, c3);
}

__device__ inline float u32_to_uniform(uint32_t x) {
  return (x >> 8) * (1.0f / 16777216.0f);  // [0,1)
}

__device__ inline float to_f32(float x) { return x; }
__device__ inline float to_f32(__bf16 x) { return (float)x; }
__device__ inline void from_f32(float v, float& out) { out = v; }
__device__ inline void from_f32(float v, __bf16& out) { out = (__bf16)v; }

}  // namespace ps
