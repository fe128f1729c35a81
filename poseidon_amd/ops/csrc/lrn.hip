// LRN (cross-channel local response normalization), NHWC layout.
// In NHWC the sliding channel window is CONTIGUOUS memory -- each thread
// computes one (n,h,w,c) output with a c-window loop over adjacent elements
// (vs the reference's strided channel walk, lrn_layer.cu:10-160).
//
//   scale_i = 1 + (alpha/size) * sum_{j in window(i)} x_j^2
//   y_i     = x_i * scale_i^(-beta)
//   dx_i    = dy_i * scale_i^(-beta)
//             - (2*alpha*beta/size) * x_i * sum_{j in window(i)} dy_j*y_j/scale_j

#include "ps_common.h"

namespace ps {

// sc^(-beta) with a fast path for the universal beta = 0.75:
// sc^(-3/4) = rsqrt(sc) * sqrt(rsqrt(sc)) -- two hardware transcendentals
// instead of __powf's ~20-op expansion (PMC: the v8 kernels were 66-78%
// issue-stalled on the powf dependency chain, ~300 VALU/iteration).
__device__ inline float lrn_pow_negbeta(float sc, float beta) {
  if (beta == 0.75f) {
    const float r = rsqrtf(sc);
    return r * sqrtf(r);
  }
  return __powf(sc, -beta);
}

template <typename T>
__global__ void lrn_fwd_k(const T* x, T* y, float* scale, int64_t rows, int C,
                          int size, float alpha_over_n, float beta) {
  const int pre = (size - 1) / 2;
  int64_t total = rows * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % C;
    int64_t row = i / C;
    const T* xr = x + row * C;
    int c0 = max(c - pre, 0), c1 = min(c - pre + size, C);
    float ss = 0.f;
    for (int j = c0; j < c1; ++j) {
      float v = to_f32(xr[j]);
      ss += v * v;
    }
    float sc = 1.0f + alpha_over_n * ss;
    scale[i] = sc;
    from_f32(to_f32(xr[c]) * lrn_pow_negbeta(sc, beta), y[i]);
  }
}

// backward pass 1: ratio = dy*y/scale (one divide per element instead of
// one per window entry)
template <typename T>
__global__ void lrn_ratio_k(const T* y, const float* scale, const T* dy,
                            float* ratio, int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x)
    ratio[i] = to_f32(dy[i]) * to_f32(y[i]) / scale[i];
}

// backward pass 2: dx = dy*scale^-beta - cr * x * window_sum(ratio)
template <typename T>
__global__ void lrn_bwd_k(const T* x, const float* scale, const T* dy,
                          const float* ratio, T* dx, int64_t rows, int C,
                          int size, float cache_ratio, float beta) {
  const int pre = (size - 1) / 2;
  const int post = size - 1 - pre;
  int64_t total = rows * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % C;
    int64_t row = i / C;
    int64_t base = row * C;
    int c0 = max(c - post, 0), c1 = min(c + pre + 1, C);
    float acc = 0.f;
    for (int j = c0; j < c1; ++j) acc += ratio[base + j];
    from_f32(to_f32(dy[i]) * lrn_pow_negbeta(scale[i], beta)
                 - cache_ratio * to_f32(x[i]) * acc,
             dx[i]);
  }
}

// ---------------------------------------------------------------------------
// Row-block variants (C <= 256, the common case): a 256-thread block stages
// rpb = 256/C whole pixel rows in LDS, so the channel window reads hit LDS
// instead of re-reading global memory, and there is no per-element i%C /
// i/C (64-bit IDIV was the dominant cost of the flat kernels above).
// Backward additionally fuses the ratio pass (dy*y/scale) into the same
// kernel via LDS, removing the ratio workspace round-trip entirely.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void lrn_fwd_rows_k(const T* __restrict__ x, T* __restrict__ y,
                               float* __restrict__ scale, int64_t rows, int C,
                               int size, float alpha_over_n, float beta,
                               int rpb) {
  extern __shared__ float xs[];  // rpb * C squared inputs
  const int pre = (size - 1) / 2;
  const int t = threadIdx.x;
  const int lr = t / C, lc = t - lr * C;  // one IDIV per thread, not per elt
  const bool active = lr < rpb;
  for (int64_t r0 = (int64_t)blockIdx.x * rpb; r0 < rows;
       r0 += (int64_t)gridDim.x * rpb) {
    const int64_t row = r0 + lr;
    const bool ok = active && row < rows;
    const int64_t idx = row * C + lc;
    float v = 0.f;
    if (ok) v = to_f32(x[idx]);
    if (active) xs[lr * C + lc] = v * v;
    __syncthreads();
    if (ok) {
      const int c0 = max(lc - pre, 0), c1 = min(lc - pre + size, C);
      const float* xr = xs + lr * C;
      float ss = 0.f;
      for (int j = c0; j < c1; ++j) ss += xr[j];
      const float sc = 1.0f + alpha_over_n * ss;
      scale[idx] = sc;
      from_f32(v * lrn_pow_negbeta(sc, beta), y[idx]);
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void lrn_bwd_rows_k(const T* __restrict__ x,
                               const float* __restrict__ scale,
                               const T* __restrict__ dy,
                               const T* __restrict__ y, T* __restrict__ dx,
                               int64_t rows, int C, int size,
                               float cache_ratio, float beta, int rpb) {
  extern __shared__ float rs[];  // rpb * C ratios
  const int pre = (size - 1) / 2;
  const int post = size - 1 - pre;
  const int t = threadIdx.x;
  const int lr = t / C, lc = t - lr * C;
  const bool active = lr < rpb;
  for (int64_t r0 = (int64_t)blockIdx.x * rpb; r0 < rows;
       r0 += (int64_t)gridDim.x * rpb) {
    const int64_t row = r0 + lr;
    const bool ok = active && row < rows;
    const int64_t idx = row * C + lc;
    float dyv = 0.f, scv = 1.f, xv = 0.f;
    if (ok) {
      dyv = to_f32(dy[idx]);
      scv = scale[idx];
      xv = to_f32(x[idx]);
    }
    if (active) rs[lr * C + lc] = ok ? dyv * to_f32(y[idx]) / scv : 0.f;
    __syncthreads();
    if (ok) {
      const int c0 = max(lc - post, 0), c1 = min(lc + pre + 1, C);
      const float* rr = rs + lr * C;
      float acc = 0.f;
      for (int j = c0; j < c1; ++j) acc += rr[j];
      from_f32(dyv * lrn_pow_negbeta(scv, beta) - cache_ratio * xv * acc,
               dx[idx]);
    }
    __syncthreads();
  }
}

static inline dim3 lrn_grid(int64_t rows, int rpb) {
  int64_t blocks = cdiv64(rows, rpb);
  if (blocks > (64 << 10)) blocks = 64 << 10;
  return dim3((unsigned)blocks);
}

// ---------------------------------------------------------------------------
// Vec8 halo-register variants (C % 8 == 0, odd size, pre <= 4): each thread
// owns 8 contiguous channels of one pixel, loads them as ONE 16 B vector
// plus a few scalar halo loads (L1 hits -- neighbours just fetched them),
// and keeps the whole window arithmetic in registers. No LDS, no barriers,
// no stored scale tensor (backward recomputes it from x, trading a little
// ALU for 4 B/elt of HBM writes+reads). The row-block kernels above remain
// the fallback (they store scale) for odd channel counts.
// ---------------------------------------------------------------------------

template <typename T> struct LrnV8;
template <> struct LrnV8<float> {
  typedef f32x4 half_t;
  __device__ static inline void load8(const float* p, float* out) {
    f32x4 a = *(const f32x4*)p, b = *(const f32x4*)(p + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) { out[j] = a[j]; out[4 + j] = b[j]; }
  }
  __device__ static inline void store8(float* p, const float* v) {
    f32x4 a, b;
#pragma unroll
    for (int j = 0; j < 4; ++j) { a[j] = v[j]; b[j] = v[4 + j]; }
    *(f32x4*)p = a;
    *(f32x4*)(p + 4) = b;
  }
};
template <> struct LrnV8<__bf16> {
  __device__ static inline void load8(const __bf16* p, float* out) {
    bf16x8 v = *(const bf16x8*)p;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = (float)v[j];
  }
  __device__ static inline void store8(__bf16* p, const float* v) {
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (__bf16)v[j];
    *(bf16x8*)p = o;
  }
};

template <typename T, int PRE>
__global__ void lrn_fwd_v8_k(const T* __restrict__ x, T* __restrict__ y,
                             int64_t rows, int C, float alpha_over_n,
                             float beta, int gpr, int rpb) {
  const int t = threadIdx.x;
  const int lr = t / gpr, lg = t - lr * gpr;  // once per thread
  const int cb = lg * 8;
  const bool active = lr < rpb;
  for (int64_t r0 = (int64_t)blockIdx.x * rpb; r0 < rows;
       r0 += (int64_t)gridDim.x * rpb) {
    const int64_t row = r0 + lr;
    const bool ok = active && row < rows;
    float xv[8] = {};
    if (ok) LrnV8<T>::load8(x + row * C + cb, xv);
    float sq[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) sq[j] = xv[j] * xv[j];
    // halo x^2 from the NEIGHBOR LANES' registers (same row: lg +- 1),
    // not from memory -- the scalar halo loads of the previous version
    // quadrupled the memory request rate (4 extra 2 B requests per 16 B
    // vector) and capped the kernel at ~1.1 TB/s. Window clipping at the
    // channel edges = zeroed halo (lg boundaries).
    float lh[PRE], rh[PRE];
#pragma unroll
    for (int j = 0; j < PRE; ++j) {
      lh[j] = __shfl_up(sq[8 - PRE + j], 1);
      rh[j] = __shfl_down(sq[j], 1);
    }
    if (lg == 0)
#pragma unroll
      for (int j = 0; j < PRE; ++j) lh[j] = 0.f;
    if (lg == gpr - 1)
#pragma unroll
      for (int j = 0; j < PRE; ++j) rh[j] = 0.f;
    if (!ok) continue;
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float ss = 0.f;
#pragma unroll
      for (int k = 0; k <= 2 * PRE; ++k) {
        const int m = j + k;  // xs-space [0, 8+2P): [halo][sq][halo]
        ss += (m < PRE) ? lh[m] : (m < PRE + 8) ? sq[m - PRE]
                                                : rh[m - PRE - 8];
      }
      out[j] = xv[j] * lrn_pow_negbeta(1.0f + alpha_over_n * ss, beta);
    }
    LrnV8<T>::store8(y + row * C + cb, out);
  }
}

template <typename T, int PRE>
__global__ void lrn_bwd_v8_k(const T* __restrict__ x, const T* __restrict__ y,
                             const T* __restrict__ dy, T* __restrict__ dx,
                             int64_t rows, int C, float alpha_over_n,
                             float cache_ratio, float beta, int gpr, int rpb) {
  const int t = threadIdx.x;
  const int lr = t / gpr, lg = t - lr * gpr;
  const int cb = lg * 8;
  const bool active = lr < rpb;
  for (int64_t r0 = (int64_t)blockIdx.x * rpb; r0 < rows;
       r0 += (int64_t)gridDim.x * rpb) {
    const int64_t row = r0 + lr;
    const bool ok = active && row < rows;
    const int64_t base = row * C + cb;
    float xv[8] = {}, yv[8] = {}, dyv[8] = {};
    if (ok) {
      LrnV8<T>::load8(x + base, xv);
      LrnV8<T>::load8(y + base, yv);
      LrnV8<T>::load8(dy + base, dyv);
    }
    float sq[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) sq[j] = xv[j] * xv[j];
    // x^2 halo reaches 2P (scale of the halo ratio positions), y/dy halo P
    float lhx[2 * PRE], rhx[2 * PRE], lhy[PRE], rhy[PRE], lhd[PRE], rhd[PRE];
#pragma unroll
    for (int j = 0; j < 2 * PRE; ++j) {
      lhx[j] = __shfl_up(sq[8 - 2 * PRE + j], 1);
      rhx[j] = __shfl_down(sq[j], 1);
    }
#pragma unroll
    for (int j = 0; j < PRE; ++j) {
      lhy[j] = __shfl_up(yv[8 - PRE + j], 1);
      rhy[j] = __shfl_down(yv[j], 1);
      lhd[j] = __shfl_up(dyv[8 - PRE + j], 1);
      rhd[j] = __shfl_down(dyv[j], 1);
    }
    if (lg == 0) {
#pragma unroll
      for (int j = 0; j < 2 * PRE; ++j) lhx[j] = 0.f;
#pragma unroll
      for (int j = 0; j < PRE; ++j) lhy[j] = lhd[j] = 0.f;
    }
    if (lg == gpr - 1) {
#pragma unroll
      for (int j = 0; j < 2 * PRE; ++j) rhx[j] = 0.f;
#pragma unroll
      for (int j = 0; j < PRE; ++j) rhy[j] = rhd[j] = 0.f;
    }
    if (!ok) continue;
    // xs-space index m in [-2P, 8+2P) -> value
    auto XS = [&](int m) {
      return (m < 0) ? lhx[m + 2 * PRE]
                     : (m < 8) ? sq[m] : rhx[m - 8];
    };
    // scale at positions [-P, 8+P), ratio = dy*y/scale there
    float ratio[8 + 2 * PRE], scc[8];
#pragma unroll
    for (int i2 = 0; i2 < 8 + 2 * PRE; ++i2) {
      const int m = i2 - PRE;
      float ss = 0.f;
#pragma unroll
      for (int k = -PRE; k <= PRE; ++k) ss += XS(m + k);
      const float sc = 1.0f + alpha_over_n * ss;
      const float r = rsqrtf(sc);
      if (m >= 0 && m < 8) scc[m] = sc;
      const float ym = (m < 0) ? lhy[m + PRE] : (m < 8) ? yv[m] : rhy[m - 8];
      const float dm = (m < 0) ? lhd[m + PRE] : (m < 8) ? dyv[m] : rhd[m - 8];
      ratio[i2] = dm * ym * (r * r);  // 1/sc via rsqrt^2 (no v_div chain)
    }
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float acc = 0.f;
#pragma unroll
      for (int k = 0; k <= 2 * PRE; ++k) acc += ratio[j + k];
      out[j] = dyv[j] * lrn_pow_negbeta(scc[j], beta)
               - cache_ratio * xv[j] * acc;
    }
    LrnV8<T>::store8(dx + row * C + cb, out);
  }
}

template <typename T>
static bool lrn_v8_ok(int C, int size) {
  const int pre = (size - 1) / 2;
  // pre <= 2: the shfl halo reaches one neighbor lane (8 channels); wider
  // windows fall back to the LDS row-block kernels
  return (C % 8 == 0) && C >= 8 && (C / 8) <= 256 && (size % 2 == 1) &&
         pre >= 1 && pre <= 2;
}

#define LRN_V8_DISPATCH(kernel, T, ...)                                   \
  do {                                                                    \
    const int pre_ = (size - 1) / 2;                                      \
    const int gpr_ = C / 8;                                               \
    const int rpb_ = 256 / gpr_;                                          \
    dim3 grid_ = lrn_grid(rows, rpb_);                                    \
    switch (pre_) {                                                       \
      case 1: kernel<T, 1><<<grid_, 256, 0, s>>>(__VA_ARGS__, gpr_, rpb_); break; \
      default: kernel<T, 2><<<grid_, 256, 0, s>>>(__VA_ARGS__, gpr_, rpb_); break; \
    }                                                                     \
  } while (0)

extern "C" {

int ps_lrn_v8_ok(int C, int size) { return lrn_v8_ok<float>(C, size) ? 1 : 0; }

void ps_lrn_fwd_v8_f32(const float* x, float* y, int64_t rows, int C,
                       int size, float alpha, float beta, hipStream_t s) {
  const float an = alpha / size;
  LRN_V8_DISPATCH(lrn_fwd_v8_k, float, x, y, rows, C, an, beta);
}
void ps_lrn_fwd_v8_bf16(const void* x, void* y, int64_t rows, int C,
                        int size, float alpha, float beta, hipStream_t s) {
  const float an = alpha / size;
  LRN_V8_DISPATCH(lrn_fwd_v8_k, __bf16, (const __bf16*)x, (__bf16*)y, rows, C,
                  an, beta);
}
void ps_lrn_bwd_v8_f32(const float* x, const float* y, const float* dy,
                       float* dx, int64_t rows, int C, int size, float alpha,
                       float beta, hipStream_t s) {
  const float an = alpha / size;
  const float cr = 2.0f * alpha * beta / size;
  LRN_V8_DISPATCH(lrn_bwd_v8_k, float, x, y, dy, dx, rows, C, an, cr, beta);
}
void ps_lrn_bwd_v8_bf16(const void* x, const void* y, const void* dy,
                        void* dx, int64_t rows, int C, int size, float alpha,
                        float beta, hipStream_t s) {
  const float an = alpha / size;
  const float cr = 2.0f * alpha * beta / size;
  LRN_V8_DISPATCH(lrn_bwd_v8_k, __bf16, (const __bf16*)x, (const __bf16*)y,
                  (const __bf16*)dy, (__bf16*)dx, rows, C, an, cr, beta);
}

void ps_lrn_fwd_f32(const float* x, float* y, float* scale, int64_t rows,
                    int C, int size, float alpha, float beta, hipStream_t s) {
  if (C <= 256) {
    int rpb = 256 / C;
    lrn_fwd_rows_k<float><<<lrn_grid(rows, rpb), 256, rpb * C * 4, s>>>(
        x, y, scale, rows, C, size, alpha / size, beta, rpb);
  } else {
    lrn_fwd_k<float><<<ew_grid(rows * C), 256, 0, s>>>(
        x, y, scale, rows, C, size, alpha / size, beta);
  }
}
void ps_lrn_fwd_bf16(const void* x, void* y, float* scale, int64_t rows,
                     int C, int size, float alpha, float beta, hipStream_t s) {
  if (C <= 256) {
    int rpb = 256 / C;
    lrn_fwd_rows_k<__bf16><<<lrn_grid(rows, rpb), 256, rpb * C * 4, s>>>(
        (const __bf16*)x, (__bf16*)y, scale, rows, C, size, alpha / size,
        beta, rpb);
  } else {
    lrn_fwd_k<__bf16><<<ew_grid(rows * C), 256, 0, s>>>(
        (const __bf16*)x, (__bf16*)y, scale, rows, C, size, alpha / size,
        beta);
  }
}
void ps_lrn_bwd_f32(const float* x, const float* y, const float* scale,
                    const float* dy, float* dx, float* ratio_ws, int64_t rows,
                    int C, int size, float alpha, float beta, hipStream_t s) {
  const float cr = 2.0f * alpha * beta / size;
  if (C <= 256) {
    int rpb = 256 / C;
    lrn_bwd_rows_k<float><<<lrn_grid(rows, rpb), 256, rpb * C * 4, s>>>(
        x, scale, dy, y, dx, rows, C, size, cr, beta, rpb);
  } else {
    lrn_ratio_k<float><<<ew_grid(rows * C), 256, 0, s>>>(y, scale, dy,
                                                         ratio_ws, rows * C);
    lrn_bwd_k<float><<<ew_grid(rows * C), 256, 0, s>>>(
        x, scale, dy, ratio_ws, dx, rows, C, size, cr, beta);
  }
}
void ps_lrn_bwd_bf16(const void* x, const void* y, const float* scale,
                     const void* dy, void* dx, float* ratio_ws, int64_t rows,
                     int C, int size, float alpha, float beta, hipStream_t s) {
  const float cr = 2.0f * alpha * beta / size;
  if (C <= 256) {
    int rpb = 256 / C;
    lrn_bwd_rows_k<__bf16><<<lrn_grid(rows, rpb), 256, rpb * C * 4, s>>>(
        (const __bf16*)x, scale, (const __bf16*)dy, (const __bf16*)y,
        (__bf16*)dx, rows, C, size, cr, beta, rpb);
  } else {
    lrn_ratio_k<__bf16><<<ew_grid(rows * C), 256, 0, s>>>(
        (const __bf16*)y, scale, (const __bf16*)dy, ratio_ws, rows * C);
    lrn_bwd_k<__bf16><<<ew_grid(rows * C), 256, 0, s>>>(
        (const __bf16*)x, scale, (const __bf16*)dy, ratio_ws, (__bf16*)dx,
        rows, C, size, cr, beta);
  }
}

}  // extern "C"

}  // namespace ps
