#include "hip/hip_runtime.h"
// LRN (cross-channel local response normalization), NHWC layout.
// In NHWC the sliding channel window is CONTIGUOUS memory: one thread owns
// one (n,h,w) pixel and walks the channel window with a running sum -- each
// x element is read O(1) times instead of O(size) (vs the reference's
// strided channel walk, lrn_layer.cu:10-160). Parallelism = N*H*W threads
// (AlexNet norm1: 774k threads, plenty for 256 CUs).
//
//   scale_i = 1 + (alpha/size) * sum_{j in window(i)} x_j^2
//   y_i     = x_i * scale_i^(-beta)
//   dx_i    = dy_i * scale_i^(-beta)
//             - (2*alpha*beta/size) * x_i * sum_{j: i in window(j)} dy_j*y_j/scale_j

#include "ps_common_hip.h"

namespace ps {

template <typename T>
__global__ void lrn_fwd_k(const T* __restrict__ x, T* __restrict__ y,
                          float* __restrict__ scale, int64_t rows, int C,
                          int size, float alpha_over_n, float beta) {
  const int pre = (size - 1) / 2;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < rows; row += (int64_t)gridDim.x * blockDim.x) {
    const T* xr = x + row * C;
    T* yr = y + row * C;
    float* sr = scale + row * C;
    float ss = 0.f;
    // prime: window of c=0 covers [0, size-pre)
    int head = size - pre;  // first channel NOT yet included
    for (int j = 0; j < head && j < C; ++j) {
      float v = to_f32(xr[j]);
      ss += v * v;
    }
    for (int c = 0; c < C; ++c) {
      float sc = 1.0f + alpha_over_n * ss;
      sr[c] = sc;
      from_f32(to_f32(xr[c]) * __powf(sc, -beta), yr[c]);
      // slide: add c+head, drop c-pre
      int add = c + head;
      if (add < C) {
        float v = to_f32(xr[add]);
        ss += v * v;
      }
      int drop = c - pre;
      if (drop >= 0) {
        float v = to_f32(xr[drop]);
        ss -= v * v;
      }
    }
  }
}

template <typename T>
__global__ void lrn_bwd_k(const T* __restrict__ x, const T* __restrict__ y,
                          const float* __restrict__ scale,
                          const T* __restrict__ dy, T* __restrict__ dx,
                          int64_t rows, int C, int size, float cache_ratio,
                          float beta) {
  const int pre = (size - 1) / 2;
  const int post = size - 1 - pre;
  // i's accumulator sums over j in [i-post, i+pre]
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       row < rows; row += (int64_t)gridDim.x * blockDim.x) {
    int64_t base = row * C;
    float acc = 0.f;
    int head = pre + 1;  // first j NOT yet included for i=0 window [0, pre]
    for (int j = 0; j < head && j < C; ++j) {
      int64_t jj = base + j;
      acc += to_f32(dy[jj]) * to_f32(y[jj]) / scale[jj];
    }
    for (int c = 0; c < C; ++c) {
      int64_t i = base + c;
      float sc = scale[i];
      from_f32(to_f32(dy[i]) * __powf(sc, -beta)
                   - cache_ratio * to_f32(x[i]) * acc,
               dx[i]);
      int add = c + head;
      if (add < C) {
        int64_t jj = base + add;
        acc += to_f32(dy[jj]) * to_f32(y[jj]) / scale[jj];
      }
      int drop = c - post;
      if (drop >= 0) {
        int64_t jj = base + drop;
        acc -= to_f32(dy[jj]) * to_f32(y[jj]) / scale[jj];
      }
    }
  }
}

extern "C" {

void ps_lrn_fwd_f32(const float* x, float* y, float* scale, int64_t rows,
                    int C, int size, float alpha, float beta, hipStream_t s) {
 hipLaunchKernelGGL(( lrn_fwd_k<float>), dim3(ew_grid(rows)), dim3(256), 0, s, 
      x, y, scale, rows, C, size, alpha / size, beta);
}
void ps_lrn_fwd_bf16(const void* x, void* y, float* scale, int64_t rows,
                     int C, int size, float alpha, float beta, hipStream_t s) {
 hipLaunchKernelGGL(( lrn_fwd_k<__bf16>), dim3(ew_grid(rows)), dim3(256), 0, s, 
      (const __bf16*)x, (__bf16*)y, scale, rows, C, size, alpha / size, beta);
}
void ps_lrn_bwd_f32(const float* x, const float* y, const float* scale,
                    const float* dy, float* dx, int64_t rows, int C, int size,
                    float alpha, float beta, hipStream_t s) {
 hipLaunchKernelGGL(( lrn_bwd_k<float>), dim3(ew_grid(rows)), dim3(256), 0, s, 
      x, y, scale, dy, dx, rows, C, size, 2.0f * alpha * beta / size, beta);
}
void ps_lrn_bwd_bf16(const void* x, const void* y, const float* scale,
                     const void* dy, void* dx, int64_t rows, int C, int size,
                     float alpha, float beta, hipStream_t s) {
 hipLaunchKernelGGL(( lrn_bwd_k<__bf16>), dim3(ew_grid(rows)), dim3(256), 0, s, 
      (const __bf16*)x, (const __bf16*)y, scale, (const __bf16*)dy,
      (__bf16*)dx, rows, C, size, 2.0f * alpha * beta / size, beta);
}

}  // extern "C"

}  // namespace ps
