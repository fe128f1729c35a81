// Elementwise / neuron-layer kernels (grid-stride, float4-vectorized where
// the tail allows). Replaces relu/sigmoid/tanh/bnll/dropout/threshold/power
// CUDA kernels (reference src/caffe/layers/*_layer.cu) and the
// math_functions.cu elementwise set.

#include "ps_common.h"

namespace ps {

// ---------------------------------------------------------------------------
// neuron fwd/bwd
// ---------------------------------------------------------------------------

template <typename T>
__global__ void relu_fwd_k(const T* x, T* y, int64_t n, float slope) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    from_f32(v > 0.f ? v : v * slope, y[i]);
  }
}

template <typename T>
__global__ void relu_bwd_k(const T* x, const T* dy, T* dx, int64_t n, float slope) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float g = to_f32(dy[i]);
    from_f32(to_f32(x[i]) > 0.f ? g : g * slope, dx[i]);
  }
}

// vec8 bf16 relu forms (activation tensors are N*H*W*C with C % 8 == 0 in
// practice; scalar kernels below remain the fallback + f32/CPU-parity path)
__global__ void relu_fwd_v8_k(const __bf16* __restrict__ x,
                              __bf16* __restrict__ y, int64_t nvec,
                              float slope) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 v = ((const bf16x8*)x)[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)v[j];
      o[j] = (__bf16)(f > 0.f ? f : f * slope);
    }
    ((bf16x8*)y)[i] = o;
  }
}

__global__ void relu_bwd_v8_k(const __bf16* __restrict__ x,
                              const __bf16* __restrict__ dy,
                              __bf16* __restrict__ dx, int64_t nvec,
                              float slope) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 xv = ((const bf16x8*)x)[i], dyv = ((const bf16x8*)dy)[i], o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = (float)dyv[j];
      o[j] = (__bf16)((float)xv[j] > 0.f ? g : g * slope);
    }
    ((bf16x8*)dx)[i] = o;
  }
}

template <typename T>
__global__ void sigmoid_fwd_k(const T* x, T* y, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    from_f32(1.0f / (1.0f + __expf(-to_f32(x[i]))), y[i]);
}

template <typename T>
__global__ void sigmoid_bwd_k(const T* y, const T* dy, T* dx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float yv = to_f32(y[i]);
    from_f32(to_f32(dy[i]) * yv * (1.0f - yv), dx[i]);
  }
}

template <typename T>
__global__ void tanh_fwd_k(const T* x, T* y, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    from_f32(tanhf(to_f32(x[i])), y[i]);
}

template <typename T>
__global__ void tanh_bwd_k(const T* y, const T* dy, T* dx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float yv = to_f32(y[i]);
    from_f32(to_f32(dy[i]) * (1.0f - yv * yv), dx[i]);
  }
}

template <typename T>
__global__ void bnll_fwd_k(const T* x, T* y, int64_t n) {
  // log(1+e^x), stable (bnll_layer.cu)
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    from_f32(v > 0 ? v + log1pf(__expf(-v)) : log1pf(__expf(v)), y[i]);
  }
}

template <typename T>
__global__ void bnll_bwd_k(const T* x, const T* dy, T* dx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = to_f32(x[i]);
    from_f32(to_f32(dy[i]) / (1.0f + __expf(-v)), dx[i]);
  }
}

// OFFDEV reads the philox offset from device memory so a hipGraph replay
// draws fresh masks each iteration (the host-value form freezes under replay)
template <typename T, bool OFFDEV>
__global__ void dropout_fwd_k(const T* x, T* y, uint8_t* mask, int64_t n,
                              float ratio, float scale, uint64_t seed,
                              uint64_t offset, const unsigned long long* off_dev) {
  if (OFFDEV) offset = off_dev[0];
  for (int64_t i4 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i4 * 4 < n; i4 += (int64_t)gridDim.x * blockDim.x) {
    uint4 r = philox4(seed, offset, (uint32_t)i4);
    uint32_t rv[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t i = i4 * 4 + j;
      if (i < n) {
        bool keep = u32_to_uniform(rv[j]) >= ratio;
        mask[i] = keep;
        from_f32(keep ? to_f32(x[i]) * scale : 0.0f, y[i]);
      }
    }
  }
}

template <typename T>
__global__ void dropout_bwd_k(const T* dy, const uint8_t* mask, T* dx,
                              int64_t n, float scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    from_f32(mask[i] ? to_f32(dy[i]) * scale : 0.0f, dx[i]);
}

template <typename T>
__global__ void threshold_fwd_k(const T* x, T* y, int64_t n, float thr) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    from_f32(to_f32(x[i]) > thr ? 1.0f : 0.0f, y[i]);
}

// pairwise eltwise max with argmax mask (ELTWISE MAX over >=2 blobs runs
// this iteratively, like the reference's MaxForward, eltwise_layer.cu:11)
template <typename T>
__global__ void eltwise_max_fwd_k(const T* a, const T* b, T* y,
                                  uint8_t* mask, int64_t n, int idx_b) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float av = to_f32(a[i]), bv = to_f32(b[i]);
    if (bv > av) {
      mask[i] = (uint8_t)idx_b;
      from_f32(bv, y[i]);
    } else {
      from_f32(av, y[i]);
    }
  }
}

template <typename T>
__global__ void eltwise_max_bwd_k(const T* dy, const uint8_t* mask, T* dx,
                                  int64_t n, int idx) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    from_f32(mask[i] == idx ? to_f32(dy[i]) : 0.0f, dx[i]);
}

// contrastive-loss per-pair terms (contrastive_loss_layer.cu:49 CLLForward):
// legacy=false variant: sim pairs contribute d^2, dissim max(margin-d, 0)^2
__global__ void contrastive_fwd_k(const float* dist_sq, const float* sim,
                                  float* loss, int64_t n, float margin,
                                  int legacy) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (sim[i] != 0.0f) {
      loss[i] = dist_sq[i];
    } else if (legacy) {
      float m = margin - dist_sq[i];
      loss[i] = m > 0.f ? m : 0.f;
    } else {
      float m = margin - sqrtf(dist_sq[i]);
      m = m > 0.f ? m : 0.f;
      loss[i] = m * m;
    }
  }
}

// column-sum: out[c] += sum_r in[r][c] for in[R][C] row-major -- the bias
// gradient for both linear (dy[M][N]) and NHWC conv (dy[(N*OH*OW)][C]).
// Each block owns a row slab; within the block, 64-lane groups sweep 64
// consecutive columns (coalesced) with 4 row-phases, LDS-reduce the 4
// partials per column, then one atomicAdd per (block, column)
// (guide §6 Guideline 12: partial-reduce first, few atomics).
template <typename T>
__global__ void colsum_k(const T* __restrict__ in, float* __restrict__ out,
                         int64_t R, int C) {
  typedef T vec2 __attribute__((ext_vector_type(2)));
  __shared__ float part[4][128];
  const int lane = threadIdx.x & 63;
  const int rg = threadIdx.x >> 6;  // 4 row groups
  int64_t rows_per = (R + gridDim.x - 1) / gridDim.x;
  int64_t r0 = (int64_t)blockIdx.x * rows_per;
  int64_t r1 = min(R, r0 + rows_per);
  const bool v2 = (C % 2) == 0;
  const int span = v2 ? 128 : 64;
  for (int c0 = 0; c0 < C; c0 += span) {
    float acc0 = 0.f, acc1 = 0.f;
    if (v2) {
      int c = c0 + lane * 2;
      if (c + 1 < C || c < C) {
        for (int64_t r = r0 + rg; r < r1; r += 4) {
          vec2 v = *reinterpret_cast<const vec2*>(&in[r * C + c]);
          acc0 += to_f32(v[0]);
          acc1 += to_f32(v[1]);
        }
      }
      part[rg][lane * 2] = acc0;
      part[rg][lane * 2 + 1] = acc1;
    } else {
      int c = c0 + lane;
      if (c < C)
        for (int64_t r = r0 + rg; r < r1; r += 4)
          acc0 += to_f32(in[r * C + c]);
      part[rg][lane] = acc0;
    }
    __syncthreads();
    if (rg < 2) {
      int idx = rg * 64 + lane;  // 128 partial slots
      int c = c0 + idx;
      if (idx < span && c < C) {
        float v = part[0][idx] + part[1][idx] + part[2][idx] + part[3][idx];
        if (gridDim.x == 1)
          out[c] += v;
        else
          atomicAdd(&out[c], v);
      }
    }
    __syncthreads();
  }
}

// Flat-vector colsum: whole-wave 16 B loads regardless of C (the banded
// kernel above degrades to C*sizeof(T) bytes per wave transaction when
// C < 2*64 lanes, e.g. conv bias grads with C<=128). Each thread walks the
// matrix as a flat vec stream; its column phase cycles with period
// PH = C/gcd(grid_stride*VEC, C), so it accumulates into PH*VEC REGISTERS
// and touches LDS only once at flush (a per-load LDS-atomic variant
// serialized on slot contention and measured slower than the banded
// kernel). PH > 4 falls back to the banded kernel.
template <typename T, int VEC, int PH>
__global__ void colsum_flat_k(const T* __restrict__ in,
                              float* __restrict__ out, int64_t R, int C,
                              int rot) {
  extern __shared__ float part[];  // C floats
  for (int c = threadIdx.x; c < C; c += blockDim.x) part[c] = 0.f;
  __syncthreads();
  typedef T vecT __attribute__((ext_vector_type(VEC)));
  const int64_t nvec = R * C / VEC;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int c0 = (int)((i * VEC) % C);  // multiple of VEC (C % VEC == 0)
  float acc[PH][VEC];
#pragma unroll
  for (int q = 0; q < PH; ++q)
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[q][j] = 0.f;
  int p = 0;
  // NOTE r2: a dual-outstanding-load variant and a 4096-block grid were
  // both tried for the 68.9%-wave-parked profile and REGRESSED (PH=3
  // launches 77 -> 141 us on AlexNet conv bias: doubling the block count
  // doubles the terminal atomicAdd traffic onto C addresses, and the
  // second load stream spills the PH accumulator registers).
  for (; i < nvec; i += stride) {
    vecT v = *((const vecT*)in + i);
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[p][j] += to_f32(v[j]);
    if (PH > 1 && ++p == PH) p = 0;
  }
#pragma unroll
  for (int q = 0; q < PH; ++q) {
    int c = c0 + q * rot;
    while (c >= C) c -= C;
#pragma unroll
    for (int j = 0; j < VEC; ++j) atomicAdd(&part[c + j], acc[q][j]);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += blockDim.x)
    atomicAdd(&out[c], part[c]);
}

static inline int colsum_gcd(int a, int b) {
  while (b) { int t = a % b; a = b; b = t; }
  return a;
}

static inline dim3 colsum_flat_grid(int64_t nvec) {
  int64_t blocks = cdiv64(nvec, 256);
  if (blocks > 2048) blocks = 2048;
  return dim3((unsigned)blocks);
}


// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

#define PS_EW_LAUNCH(name, ...)                                         \
  name<<<ew_grid(n), 256, 0, s>>>(__VA_ARGS__)

// ---------------------------------------------------------------------------
// Multi-tensor bias colsum: one launch covers every deferred conv bias
// gradient. 16 B vector loads; thread t owns column-chunk (t % CV) and
// row-phase (t / CV), so ALL 256 threads stream the slab even for C=16
// reduce layers (a lane-per-chunk form stranded 62/64 lanes there and a
// 4 B banded form re-read slabs per band: both measured ~3x SLOWER than
// the launches they replaced). Flush via LDS atomics once per block,
// then one global atomicAdd per column. Dynamic LDS = max C floats.
struct ColsumDesc {
  const void* dy;
  float* db;
  int64_t R;
  int64_t rows_per;
  int C;
};
struct ColsumChunk {
  int t;
  int64_t off;
};

template <typename T>
__global__ void colsum_mt_k(const ColsumDesc* __restrict__ descs,
                            const ColsumChunk* __restrict__ chunks) {
  constexpr int VEC = 16 / (int)sizeof(T);
  typedef T vecT __attribute__((ext_vector_type(VEC)));
  extern __shared__ float part[];  // max C floats
  const ColsumChunk ck = chunks[blockIdx.x];
  const ColsumDesc d = descs[ck.t];
  const T* in = (const T*)d.dy;
  const int C = d.C;
  const int CV = C / VEC;
  for (int c = threadIdx.x; c < C; c += 256) part[c] = 0.f;
  __syncthreads();
  const int nph = 256 / CV;  // row phases; threads >= nph*CV idle
  const int t = (int)threadIdx.x;
  const int64_t r1 = min(d.R, ck.off + d.rows_per);
  if (nph == 0) {
    // C > 2048 elements: single-phase strided over column chunks
    for (int cv = t; cv < CV; cv += 256) {
      float acc[VEC];
#pragma unroll
      for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
      for (int64_t r = ck.off; r < r1; ++r) {
        vecT v = *(((const vecT*)&in[r * C]) + cv);
#pragma unroll
        for (int j = 0; j < VEC; ++j) acc[j] += to_f32(v[j]);
      }
#pragma unroll
      for (int j = 0; j < VEC; ++j)
        atomicAdd(&part[cv * VEC + j], acc[j]);
    }
  } else if (t < nph * CV) {
    const int cv = t % CV;
    const int ph = t / CV;
    float acc[VEC];
#pragma unroll
    for (int j = 0; j < VEC; ++j) acc[j] = 0.f;
    for (int64_t r = ck.off + ph; r < r1; r += nph) {
      vecT v = *(((const vecT*)&in[r * C]) + cv);
#pragma unroll
      for (int j = 0; j < VEC; ++j) acc[j] += to_f32(v[j]);
    }
#pragma unroll
    for (int j = 0; j < VEC; ++j)
      atomicAdd(&part[cv * VEC + j], acc[j]);
  }
  __syncthreads();
  for (int c = threadIdx.x; c < C; c += 256)
    atomicAdd(&d.db[c], part[c]);
}

extern "C" {

#define PS_DEF_UNARY(opname, kern)                                            \
  void ps_##opname##_f32(const float* a, float* b, int64_t n, hipStream_t s) {\
    PS_EW_LAUNCH(kern<float>, a, b, n);                                       \
  }                                                                           \
  void ps_##opname##_bf16(const void* a, void* b, int64_t n, hipStream_t s) { \
    PS_EW_LAUNCH(kern<__bf16>, (const __bf16*)a, (__bf16*)b, n);              \
  }

void ps_threshold_fwd_f32(const float* x, float* y, int64_t n, float thr,
                          hipStream_t s) {
  PS_EW_LAUNCH(threshold_fwd_k<float>, x, y, n, thr);
}
void ps_threshold_fwd_bf16(const void* x, void* y, int64_t n, float thr,
                           hipStream_t s) {
  PS_EW_LAUNCH(threshold_fwd_k<__bf16>, (const __bf16*)x, (__bf16*)y, n, thr);
}
void ps_eltwise_max_fwd_f32(const float* a, const float* b, float* y,
                            uint8_t* mask, int64_t n, int idx_b,
                            hipStream_t s) {
  PS_EW_LAUNCH(eltwise_max_fwd_k<float>, a, b, y, mask, n, idx_b);
}
void ps_eltwise_max_fwd_bf16(const void* a, const void* b, void* y,
                             uint8_t* mask, int64_t n, int idx_b,
                             hipStream_t s) {
  PS_EW_LAUNCH(eltwise_max_fwd_k<__bf16>, (const __bf16*)a, (const __bf16*)b,
               (__bf16*)y, mask, n, idx_b);
}
void ps_eltwise_max_bwd_f32(const float* dy, const uint8_t* mask, float* dx,
                            int64_t n, int idx, hipStream_t s) {
  PS_EW_LAUNCH(eltwise_max_bwd_k<float>, dy, mask, dx, n, idx);
}
void ps_eltwise_max_bwd_bf16(const void* dy, const uint8_t* mask, void* dx,
                             int64_t n, int idx, hipStream_t s) {
  PS_EW_LAUNCH(eltwise_max_bwd_k<__bf16>, (const __bf16*)dy, mask,
               (__bf16*)dx, n, idx);
}
void ps_contrastive_fwd_f32(const float* dist_sq, const float* sim,
                            float* loss, int64_t n, float margin, int legacy,
                            hipStream_t s) {
  PS_EW_LAUNCH(contrastive_fwd_k, dist_sq, sim, loss, n, margin, legacy);
}

PS_DEF_UNARY(sigmoid_fwd, sigmoid_fwd_k)
PS_DEF_UNARY(tanh_fwd, tanh_fwd_k)
PS_DEF_UNARY(bnll_fwd, bnll_fwd_k)

#define PS_DEF_BINARY(opname, kern)                                           \
  void ps_##opname##_f32(const float* a, const float* b, float* c, int64_t n, \
                         hipStream_t s) {                                     \
    PS_EW_LAUNCH(kern<float>, a, b, c, n);                                    \
  }                                                                           \
  void ps_##opname##_bf16(const void* a, const void* b, void* c, int64_t n,   \
                          hipStream_t s) {                                    \
    PS_EW_LAUNCH(kern<__bf16>, (const __bf16*)a, (const __bf16*)b, (__bf16*)c, n); \
  }

PS_DEF_BINARY(sigmoid_bwd, sigmoid_bwd_k)
PS_DEF_BINARY(tanh_bwd, tanh_bwd_k)
PS_DEF_BINARY(bnll_bwd, bnll_bwd_k)

void ps_relu_fwd_f32(const float* x, float* y, int64_t n, float slope, hipStream_t s) {
  PS_EW_LAUNCH(relu_fwd_k<float>, x, y, n, slope);
}
void ps_relu_fwd_bf16(const void* x, void* y, int64_t n, float slope, hipStream_t s) {
  if ((n & 7) == 0)
    relu_fwd_v8_k<<<ew_grid(n / 8), 256, 0, s>>>((const __bf16*)x,
                                                 (__bf16*)y, n / 8, slope);
  else
    PS_EW_LAUNCH(relu_fwd_k<__bf16>, (const __bf16*)x, (__bf16*)y, n, slope);
}
void ps_relu_bwd_f32(const float* x, const float* dy, float* dx, int64_t n,
                     float slope, hipStream_t s) {
  PS_EW_LAUNCH(relu_bwd_k<float>, x, dy, dx, n, slope);
}
void ps_relu_bwd_bf16(const void* x, const void* dy, void* dx, int64_t n,
                      float slope, hipStream_t s) {
  if ((n & 7) == 0)
    relu_bwd_v8_k<<<ew_grid(n / 8), 256, 0, s>>>(
        (const __bf16*)x, (const __bf16*)dy, (__bf16*)dx, n / 8, slope);
  else
    PS_EW_LAUNCH(relu_bwd_k<__bf16>, (const __bf16*)x, (const __bf16*)dy,
                 (__bf16*)dx, n, slope);
}

void ps_dropout_fwd_f32(const float* x, float* y, uint8_t* mask, int64_t n,
                        float ratio, uint64_t seed, uint64_t offset, hipStream_t s) {
  dropout_fwd_k<float, false><<<ew_grid((n + 3) / 4), 256, 0, s>>>(
      x, y, mask, n, ratio, 1.0f / (1.0f - ratio), seed, offset, nullptr);
}
void ps_dropout_fwd_bf16(const void* x, void* y, uint8_t* mask, int64_t n,
                         float ratio, uint64_t seed, uint64_t offset, hipStream_t s) {
  dropout_fwd_k<__bf16, false><<<ew_grid((n + 3) / 4), 256, 0, s>>>(
      (const __bf16*)x, (__bf16*)y, mask, n, ratio, 1.0f / (1.0f - ratio),
      seed, offset, nullptr);
}
void ps_dropout_fwd_f32_offdev(const float* x, float* y, uint8_t* mask,
                               int64_t n, float ratio, uint64_t seed,
                               const void* off_dev, hipStream_t s) {
  dropout_fwd_k<float, true><<<ew_grid((n + 3) / 4), 256, 0, s>>>(
      x, y, mask, n, ratio, 1.0f / (1.0f - ratio), seed, 0,
      (const unsigned long long*)off_dev);
}
void ps_dropout_fwd_bf16_offdev(const void* x, void* y, uint8_t* mask,
                                int64_t n, float ratio, uint64_t seed,
                                const void* off_dev, hipStream_t s) {
  dropout_fwd_k<__bf16, true><<<ew_grid((n + 3) / 4), 256, 0, s>>>(
      (const __bf16*)x, (__bf16*)y, mask, n, ratio, 1.0f / (1.0f - ratio),
      seed, 0, (const unsigned long long*)off_dev);
}
void ps_dropout_bwd_f32(const float* dy, const uint8_t* mask, float* dx,
                        int64_t n, float ratio, hipStream_t s) {
  PS_EW_LAUNCH(dropout_bwd_k<float>, dy, mask, dx, n, 1.0f / (1.0f - ratio));
}
void ps_dropout_bwd_bf16(const void* dy, const uint8_t* mask, void* dx,
                         int64_t n, float ratio, hipStream_t s) {
  PS_EW_LAUNCH(dropout_bwd_k<__bf16>, (const __bf16*)dy, mask, (__bf16*)dx, n,
               1.0f / (1.0f - ratio));
}

static inline dim3 colsum_grid(int64_t R, int C) {
  // row slabs: fill the chip but keep >= ~64 rows per block
  int64_t blocks = R / 64;
  if (blocks < 1) blocks = 1;
  if (blocks > 1024) blocks = 1024;
  return dim3((unsigned)blocks);
}
#define PS_COLSUM_FLAT(T, VEC, inexpr)                                        \
  do {                                                                        \
    dim3 grid_ = colsum_flat_grid(R * C / VEC);                               \
    int rot_ = (int)(((int64_t)grid_.x * 256 * VEC) % C);                     \
    int ph_ = rot_ ? C / colsum_gcd(rot_, C) : 1;                             \
    if (ph_ == 1)                                                             \
      colsum_flat_k<T, VEC, 1><<<grid_, 256, C * 4, s>>>(inexpr, out, R, C,   \
                                                         rot_);              \
    else if (ph_ == 2)                                                        \
      colsum_flat_k<T, VEC, 2><<<grid_, 256, C * 4, s>>>(inexpr, out, R, C,   \
                                                         rot_);              \
    else if (ph_ == 3)                                                        \
      colsum_flat_k<T, VEC, 3><<<grid_, 256, C * 4, s>>>(inexpr, out, R, C,   \
                                                         rot_);              \
    else if (ph_ == 4)                                                        \
      colsum_flat_k<T, VEC, 4><<<grid_, 256, C * 4, s>>>(inexpr, out, R, C,   \
                                                         rot_);              \
    else                                                                      \
      colsum_k<T><<<colsum_grid(R, C), 256, 0, s>>>(inexpr, out, R, C);       \
  } while (0)

void ps_colsum_f32(const float* in, float* out, int64_t R, int C, hipStream_t s) {
  // flat path: big matrices (amortizes the LDS flush) OR small-R wide-C
  // (the banded kernel's row-slab grid collapses to R/64 blocks: an fc
  // bias grad at R=256 ran 4 workgroups)
  if (C % 4 == 0 && C <= 8192 && (R * C >= (8 << 20) || (R < 1024 && C >= 512)))
    PS_COLSUM_FLAT(float, 4, in);
  else
    colsum_k<float><<<colsum_grid(R, C), 256, 0, s>>>(in, out, R, C);
}
void ps_colsum_bf16(const void* in, float* out, int64_t R, int C, hipStream_t s) {
  if (C % 8 == 0 && C <= 8192 && (R * C >= (8 << 20) || (R < 1024 && C >= 512)))
    PS_COLSUM_FLAT(__bf16, 8, (const __bf16*)in);
  else
    colsum_k<__bf16><<<colsum_grid(R, C), 256, 0, s>>>((const __bf16*)in, out, R, C);
}

void ps_colsum_mt(const void* descs, const void* chunks, int nchunks,
                  int bf16, int max_c, hipStream_t s) {
  if (nchunks <= 0) return;
  const size_t lds = (size_t)max_c * sizeof(float);
  if (bf16)
    colsum_mt_k<__bf16><<<dim3((unsigned)nchunks), 256, lds, s>>>(
        (const ColsumDesc*)descs, (const ColsumChunk*)chunks);
  else
    colsum_mt_k<float><<<dim3((unsigned)nchunks), 256, lds, s>>>(
        (const ColsumDesc*)descs, (const ColsumChunk*)chunks);
}

}  // extern "C"

}  // namespace ps
