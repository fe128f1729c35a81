#include "hip/hip_runtime.h"
// Row softmax + fused softmax-with-loss, one wave per row with shuffle
// reductions (vs the reference's 6-pass kernel chain, softmax_layer.cu:14-84,
// and its CPU-fallback loss layer, softmax_loss_layer.cu:12-22).
// Rows = N*H*W (NHWC spatial softmax) or batch (classifier logits);
// the reduced dim (channels/classes) is contiguous.

#include "ps_common_hip.h"

namespace ps {

__device__ inline float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, 64));
  return __shfl(v, 0, 64);
}

__device__ inline float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_down(v, off, 64);
  return __shfl(v, 0, 64);
}

// 4 waves per block, one row per wave
template <typename T>
__global__ void softmax_rows_k(const T* x, T* y, int64_t rows, int C) {
  int64_t row = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* xr = x + row * C;
  T* yr = y + row * C;
  float m = -3.4e38f;
  for (int c = lane; c < C; c += 64) m = fmaxf(m, to_f32(xr[c]));
  m = wave_max(m);
  float sum = 0.f;
  for (int c = lane; c < C; c += 64) sum += __expf(to_f32(xr[c]) - m);
  sum = wave_sum(sum);
  float inv = 1.0f / sum;
  for (int c = lane; c < C; c += 64)
    from_f32(__expf(to_f32(xr[c]) - m) * inv, yr[c]);
}

// dx = (dy - sum_c(dy*y)) * y per row
template <typename T>
__global__ void softmax_bwd_rows_k(const T* y, const T* dy, T* dx,
                                   int64_t rows, int C) {
  int64_t row = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* yr = y + row * C;
  const T* dyr = dy + row * C;
  T* dxr = dx + row * C;
  float dot = 0.f;
  for (int c = lane; c < C; c += 64) dot += to_f32(dyr[c]) * to_f32(yr[c]);
  dot = wave_sum(dot);
  for (int c = lane; c < C; c += 64)
    from_f32((to_f32(dyr[c]) - dot) * to_f32(yr[c]), dxr[c]);
}

// fused: prob + per-row NLL, loss accumulated into loss_out[0] (pre-zeroed),
// final normalization by batch on the host side wrapper (adds /n).
template <typename T, typename LT>
__global__ void softmax_loss_fwd_k(const T* x, const LT* labels, T* prob,
                                   float* loss_out, int64_t rows, int C) {
  int64_t row = (int64_t)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  const T* xr = x + row * C;
  T* pr = prob + row * C;
  float m = -3.4e38f;
  for (int c = lane; c < C; c += 64) m = fmaxf(m, to_f32(xr[c]));
  m = wave_max(m);
  float sum = 0.f;
  for (int c = lane; c < C; c += 64) sum += __expf(to_f32(xr[c]) - m);
  sum = wave_sum(sum);
  float inv = 1.0f / sum;
  for (int c = lane; c < C; c += 64)
    from_f32(__expf(to_f32(xr[c]) - m) * inv, pr[c]);
  if (lane == 0) {
    int lbl = (int)to_f32(labels[row]);
    lbl = min(max(lbl, 0), C - 1);
    float logp = to_f32(xr[lbl]) - m - __logf(sum);
    atomicAdd(loss_out, -logp);
  }
}

// dx = (prob - onehot) * w  (w = loss_weight / batch)
template <typename T, typename LT>
__global__ void softmax_loss_bwd_k(const T* prob, const LT* labels, T* dx,
                                   int64_t rows, int C, float w) {
  int64_t total = rows * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = i / C;
    int c = i % C;
    int lbl = (int)to_f32(labels[row]);
    float p = to_f32(prob[i]);
    from_f32((p - (c == lbl ? 1.0f : 0.0f)) * w, dx[i]);
  }
}

extern "C" {

void ps_softmax_rows_f32(const float* x, float* y, int64_t rows, int C,
                         hipStream_t s) {
 hipLaunchKernelGGL(( softmax_rows_k<float>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, x, y, rows, C);
}
void ps_softmax_rows_bf16(const void* x, void* y, int64_t rows, int C,
                          hipStream_t s) {
 hipLaunchKernelGGL(( softmax_rows_k<__bf16>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, 
      (const __bf16*)x, (__bf16*)y, rows, C);
}
void ps_softmax_bwd_rows_f32(const float* y, const float* dy, float* dx,
                             int64_t rows, int C, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_bwd_rows_k<float>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, y, dy, dx, rows, C);
}
void ps_softmax_bwd_rows_bf16(const void* y, const void* dy, void* dx,
                              int64_t rows, int C, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_bwd_rows_k<__bf16>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, 
      (const __bf16*)y, (const __bf16*)dy, (__bf16*)dx, rows, C);
}
void ps_softmax_loss_fwd_f32(const float* x, const float* labels, float* prob,
                             float* loss_out, int64_t rows, int C, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_loss_fwd_k<float, float>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, 
      x, labels, prob, loss_out, rows, C);
}
void ps_softmax_loss_fwd_bf16(const void* x, const float* labels, void* prob,
                              float* loss_out, int64_t rows, int C, hipStream_t s) {
  // labels stay fp32: class indices above 256 are not exactly representable
  // in bf16's 8-bit mantissa
 hipLaunchKernelGGL(( softmax_loss_fwd_k<__bf16, float>), dim3(cdiv64(rows, 4)), dim3(256), 0, s, 
      (const __bf16*)x, labels, (__bf16*)prob, loss_out, rows, C);
}
void ps_softmax_loss_bwd_f32(const float* prob, const float* labels, float* dx,
                             int64_t rows, int C, float w, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_loss_bwd_k<float, float>), dim3(ew_grid(rows * C)), dim3(256), 0, s, 
      prob, labels, dx, rows, C, w);
}
void ps_softmax_loss_bwd_bf16(const void* prob, const float* labels, void* dx,
                              int64_t rows, int C, float w, hipStream_t s) {
 hipLaunchKernelGGL(( softmax_loss_bwd_k<__bf16, float>), dim3(ew_grid(rows * C)), dim3(256), 0, s, 
      (const __bf16*)prob, labels, (__bf16*)dx, rows, C, w);
}

}  // extern "C"

}  // namespace ps
