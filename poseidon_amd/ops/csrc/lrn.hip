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
    from_f32(to_f32(xr[c]) * __powf(sc, -beta), y[i]);
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
    from_f32(to_f32(dy[i]) * __powf(scale[i], -beta)
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
      from_f32(v * __powf(sc, -beta), y[idx]);
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
      from_f32(dyv * __powf(scv, -beta) - cache_ratio * xv * acc, dx[idx]);
    }
    __syncthreads();
  }
}

static inline dim3 lrn_grid(int64_t rows, int rpb) {
  int64_t blocks = cdiv64(rows, rpb);
  if (blocks > (64 << 10)) blocks = 64 << 10;
  return dim3((unsigned)blocks);
}

extern "C" {

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
