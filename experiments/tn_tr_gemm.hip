// Standalone experiment: TN GEMM (both operands K-major) with glds staging
// of the NATURAL [BK][cols] K-major tile images and ds_read_b64_tr_b16
// hardware-transpose fragment reads -- no per-thread register transpose,
// no guarded loads, no LDS write conflicts.
//
// C[M,N] = A^T @ B, A [K][ldA>=M], B [K][ldB>=N], bf16 in / f32 out.
// Eligibility: M % 64 == 0 (BM), N % 64 == 0 (BN), K % 64 == 0 (BK),
// ldA/ldB % 8 == 0, 16B-aligned bases.
//
// Probe-verified tr semantics (experiments/tr16_probe.hip): within each
// 16-lane group, lane l loads 4 contiguous bf16 at its own 8B-aligned
// address; lane j then RECEIVES element j of each of the four 16-element
// chunks of the group's concatenated 64 elements. With lane l pointing at
// row kk+l/4, col nb+4*(l%4) of a row-major [k][col] image, lane j ends
// holding col nb+j of rows kk..kk+3 -- exactly an MFMA operand quarter.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define BM 64
#define BN 64
#define BK 64

// glds: stage [BK rows][64 cols] from K-major global rows (contiguous cols)
__device__ inline void stage_kmaj(__bf16* lds, const __bf16* src, int64_t ld,
                                  int k0, int c0, int wid, int lane) {
  // row length 64 bf16 = 128 B = 8 lanes x 16 B; 8 rows per 1 KB chunk
  const int r_in = lane >> 3;        // 0..7 row within chunk
  const int slot = lane & 7;         // 16B slot within row
#pragma unroll
  for (int ci = wid; ci < BK / 8; ci += 4) {
    const int row = ci * 8 + r_in;
    const __bf16* g = src + (int64_t)(k0 + row) * ld + c0 + slot * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)g,
        (__attribute__((address_space(3))) void*)(lds + ci * 512), 16, 0, 0);
  }
}

__device__ inline bf16x8 tr_frag(unsigned lds_base, int kk, int cb, int l,
                                 int ldt /*row stride elems*/) {
  // lane l of its 16-group: rows kk + l/4 (+4), col cb + 4*(l%4)
  const unsigned a1 =
      lds_base + (unsigned)(((kk + (l >> 2)) * ldt + cb + 4 * (l & 3)) * 2);
  const unsigned a2 = a1 + 4u * ldt * 2u;
  bf16x4 v1, v2;
  // "=&v": early-clobber -- insn 1 writes v1 BEFORE insn 2 consumes a2,
  // so the allocator must not overlap them (silent corruption otherwise)
  asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
               "ds_read_b64_tr_b16 %1, %3\n\t"
               "s_waitcnt lgkmcnt(0)"
               : "=&v"(v1), "=&v"(v2) : "v"(a1), "v"(a2));
  bf16x8 f;
#pragma unroll
  for (int i = 0; i < 4; ++i) { f[i] = v1[i]; f[4 + i] = v2[i]; }
  return f;
}

__global__ __launch_bounds__(256)
void gemm_tn_tr(const __bf16* __restrict__ A, const __bf16* __restrict__ B,
                float* __restrict__ C, int M, int N, int K, int64_t ldA,
                int64_t ldB, int64_t ldC, int splitk, int kchunk) {
  __shared__ __attribute__((aligned(16))) __bf16 a_lds[2][BK * BM];
  __shared__ __attribute__((aligned(16))) __bf16 b_lds[2][BK * BN];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  // 4 waves as 2x2: wave covers 32x32 = 2x2 fragments of 16x16
  const int wm = (wid >> 1) * 32;
  const int wn = (wid & 1) * 32;
  const int m0 = blockIdx.y * BM;
  const int n0 = blockIdx.x * BN;
  int k_begin = 0, k_end = K;
  if (splitk > 1) {
    k_begin = blockIdx.z * kchunk;
    k_end = min(K, k_begin + kchunk);
    if (k_begin >= k_end) return;
  }
  const int l = lane & 15, q = lane >> 4;

  f32x4 acc[2][2] = {};
  stage_kmaj(a_lds[0], A, ldA, k_begin, m0, wid, lane);
  stage_kmaj(b_lds[0], B, ldB, k_begin, n0, wid, lane);
  __syncthreads();
  unsigned ab[2] = {
      (unsigned)(unsigned long long)(__attribute__((address_space(3))) __bf16*)
          a_lds[0],
      (unsigned)(unsigned long long)(__attribute__((address_space(3))) __bf16*)
          a_lds[1]};
  unsigned bb[2] = {
      (unsigned)(unsigned long long)(__attribute__((address_space(3))) __bf16*)
          b_lds[0],
      (unsigned)(unsigned long long)(__attribute__((address_space(3))) __bf16*)
          b_lds[1]};
  int cur = 0;
  for (int k0 = k_begin; k0 < k_end; k0 += BK) {
    if (k0 + BK < k_end) {
      stage_kmaj(a_lds[cur ^ 1], A, ldA, k0 + BK, m0, wid, lane);
      stage_kmaj(b_lds[cur ^ 1], B, ldB, k0 + BK, n0, wid, lane);
    }
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int f = 0; f < 2; ++f)
        af[f] = tr_frag(ab[cur], kk + q * 8, wm + f * 16, l, BM);
#pragma unroll
      for (int f = 0; f < 2; ++f)
        bf[f] = tr_frag(bb[cur], kk + q * 8, wn + f * 16, l, BN);
#pragma unroll
      for (int fm = 0; fm < 2; ++fm)
#pragma unroll
        for (int fn = 0; fn < 2; ++fn)
          acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fm], bf[fn], acc[fm][fn], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }
  // epilogue: C map col = lane&15, row = (lane>>4)*4 + r
#pragma unroll
  for (int fm = 0; fm < 2; ++fm)
#pragma unroll
    for (int fn = 0; fn < 2; ++fn) {
      const int col = n0 + wn + fn * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = m0 + wm + fm * 16 + (lane >> 4) * 4 + r;
        if (splitk > 1)
          atomicAdd(&C[(int64_t)row * ldC + col], acc[fm][fn][r]);
        else
          C[(int64_t)row * ldC + col] = acc[fm][fn][r];
      }
    }
}

static double bench_shape(int M, int N, int K, int reps) {
  std::vector<float> ha((size_t)K * M), hb((size_t)K * N);
  srand(1);
  for (auto& v : ha) v = (rand() % 1000 - 500) / 500.0f;
  for (auto& v : hb) v = (rand() % 1000 - 500) / 500.0f;
  std::vector<__bf16> haf(ha.size()), hbf(hb.size());
  for (size_t i = 0; i < ha.size(); ++i) haf[i] = (__bf16)ha[i];
  for (size_t i = 0; i < hb.size(); ++i) hbf[i] = (__bf16)hb[i];
  __bf16 *dA, *dB;
  float* dC;
  (void)hipMalloc(&dA, haf.size() * 2);
  (void)hipMalloc(&dB, hbf.size() * 2);
  (void)hipMalloc(&dC, (size_t)M * N * 4);
  (void)hipMemcpy(dA, haf.data(), haf.size() * 2, hipMemcpyHostToDevice);
  (void)hipMemcpy(dB, hbf.data(), hbf.size() * 2, hipMemcpyHostToDevice);

  int tiles = (M / BM) * (N / BN);
  int sk = 1, kchunk = K;
  if (tiles < 384) {
    int want = (512 + tiles - 1) / tiles;
    int maxsk = (K + 255) / 256;
    sk = want < maxsk ? want : maxsk;
    if (sk < 1) sk = 1;
    kchunk = ((K / sk + BK - 1) / BK) * BK;
    sk = (K + kchunk - 1) / kchunk;
  }
  dim3 grid(N / BN, M / BM, sk);
  if (sk > 1) (void)hipMemset(dC, 0, (size_t)M * N * 4);
  gemm_tn_tr<<<grid, 256>>>(dA, dB, dC, M, N, K, M, N, N, sk, kchunk);
  (void)hipDeviceSynchronize();

  // correctness spot check on a few entries
  std::vector<float> hc((size_t)M * N);
  (void)hipMemcpy(hc.data(), dC, hc.size() * 4, hipMemcpyDeviceToHost);
  double maxerr = 0;
  for (int s = 0; s < 40; ++s) {
    int i = rand() % M, j = rand() % N;
    double ref = 0;
    for (int k = 0; k < K; ++k)
      ref += (float)(__bf16)ha[(size_t)k * M + i] *
             (float)(__bf16)hb[(size_t)k * N + j];
    double e = fabs(ref - hc[(size_t)i * N + j]) / (fabs(ref) + 1.0);
    if (e > maxerr) maxerr = e;
  }

  // clock ramp + timing
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  for (int i = 0; i < 20; ++i)
    gemm_tn_tr<<<grid, 256>>>(dA, dB, dC, M, N, K, M, N, N, sk, kchunk);
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(e0);
  for (int i = 0; i < reps; ++i) {
    if (sk > 1) (void)hipMemsetAsync(dC, 0, (size_t)M * N * 4);
    gemm_tn_tr<<<grid, 256>>>(dA, dB, dC, M, N, K, M, N, N, sk, kchunk);
  }
  (void)hipEventRecord(e1);
  (void)hipEventSynchronize(e1);
  float ms;
  (void)hipEventElapsedTime(&ms, e0, e1);
  double tf = 2.0 * M * N * K * reps / (ms / 1e3) / 1e12;
  printf("tn_tr %dx%dx%d sk=%d: %.1f TF/s (%.3f ms)  maxrelerr=%.2e\n", M, N,
         K, sk, tf, ms / reps, maxerr);
  (void)hipFree(dA);
  (void)hipFree(dB);
  (void)hipFree(dC);
  return tf;
}

int main() {
  bench_shape(64, 576, 1605632, 10);    // VGG conv1_2 wgrad
  bench_shape(512, 4608, 25088, 10);    // VGG conv5 wgrad
  bench_shape(128, 832, 25088, 10);     // inception-ish wgrad (N%64==0)
  bench_shape(4096, 4096, 4096, 10);    // square reference
  bench_shape(4096, 9216, 256, 10);     // fc6 wgrad (batch-K)
  return 0;
}
