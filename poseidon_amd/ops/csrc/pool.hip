// Pooling kernels, NHWC (channels-last) layout -- the MI355X-native choice:
// the channel dim is contiguous so every thread's window reads are coalesced
// across lanes along C.
//
// Geometry matches Caffe exactly (ceil-mode output size computed on the
// host; AVE pool_size counts near-side padding and clips the far side to
// H+pad -- reference src/caffe/layers/pooling_layer.cu:12-330). Backward is
// gather-form (no atomics), like the reference's MaxPoolBackward.

#include "ps_common.h"
#include "ps_api.h"

namespace ps {

// one thread per (n, oh, ow, V-chunk of channels): mask stores the
// WINDOW-LOCAL argmax index kh*kw_w + kw as u8 (4x less mask traffic than a
// spatial int; the layer converts to Caffe's bottom-spatial-index semantics
// if the optional mask top is requested). Channel chunks vectorize every
// global access (NHWC: adjacent channels adjacent in memory).
template <typename T, int V>
__global__ void maxpool_fwd_k(const T* x, T* y, uint8_t* mask, PoolGeom g) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  typedef uint8_t mvec_t __attribute__((ext_vector_type(V)));
  const int CV = g.C / V;
  int64_t total = (int64_t)g.N * g.Ho * g.Wo * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t t = i / CV;
    int ow = t % g.Wo; t /= g.Wo;
    int oh = t % g.Ho;
    int n = t / g.Ho;
    int hb = oh * g.sh - g.ph, wb = ow * g.sw - g.pw;
    int h0 = max(hb, 0), w0 = max(wb, 0);
    int h1 = min(hb + g.kh, g.H), w1 = min(wb + g.kw, g.W);
    float best[V];
    mvec_t bidx;
#pragma unroll
    for (int e = 0; e < V; ++e) {
      best[e] = -3.4e38f;
      bidx[e] = (uint8_t)((h0 - hb) * g.kw + (w0 - wb));
    }
    for (int h = h0; h < h1; ++h)
      for (int w = w0; w < w1; ++w) {
        vec_t v = *reinterpret_cast<const vec_t*>(
            &x[(((int64_t)n * g.H + h) * g.W + w) * g.C + cv * V]);
        uint8_t li = (uint8_t)((h - hb) * g.kw + (w - wb));
#pragma unroll
        for (int e = 0; e < V; ++e) {
          float f = to_f32(v[e]);
          if (f > best[e]) { best[e] = f; bidx[e] = li; }
        }
      }
    vec_t out;
#pragma unroll
    for (int e = 0; e < V; ++e) {
      T o;
      from_f32(best[e], o);
      out[e] = o;
    }
    *reinterpret_cast<vec_t*>(&y[i * V]) = out;
    *reinterpret_cast<mvec_t*>(&mask[i * V]) = bidx;
  }
}

// gather: one thread per (n, h, w, V-chunk) scans covering windows
template <typename T, int V>
__global__ void maxpool_bwd_k(const T* dy, const uint8_t* mask, T* dx, PoolGeom g) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  typedef uint8_t mvec_t __attribute__((ext_vector_type(V)));
  const int CV = g.C / V;
  int64_t total = (int64_t)g.N * g.H * g.W * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t t = i / CV;
    int w = t % g.W; t /= g.W;
    int h = t % g.H;
    int n = t / g.H;
    int oh0 = (h + g.ph < g.kh) ? 0 : (h + g.ph - g.kh) / g.sh + 1;
    int oh1 = min((h + g.ph) / g.sh + 1, g.Ho);
    int ow0 = (w + g.pw < g.kw) ? 0 : (w + g.pw - g.kw) / g.sw + 1;
    int ow1 = min((w + g.pw) / g.sw + 1, g.Wo);
    float acc[V];
#pragma unroll
    for (int e = 0; e < V; ++e) acc[e] = 0.f;
    for (int oh = oh0; oh < oh1; ++oh)
      for (int ow = ow0; ow < ow1; ++ow) {
        int64_t oi = ((((int64_t)n * g.Ho + oh) * g.Wo + ow) * g.C) + cv * V;
        int local = (h - (oh * g.sh - g.ph)) * g.kw + (w - (ow * g.sw - g.pw));
        mvec_t mv = *reinterpret_cast<const mvec_t*>(&mask[oi]);
        vec_t dv = *reinterpret_cast<const vec_t*>(&dy[oi]);
#pragma unroll
        for (int e = 0; e < V; ++e)
          if ((int)mv[e] == local) acc[e] += to_f32(dv[e]);
      }
    vec_t out;
#pragma unroll
    for (int e = 0; e < V; ++e) {
      T o;
      from_f32(acc[e], o);
      out[e] = o;
    }
    *reinterpret_cast<vec_t*>(&dx[i * V]) = out;
  }
}

__device__ inline int ave_pool_size(int oh, int ow, const PoolGeom& g) {
  int hstart = oh * g.sh - g.ph, wstart = ow * g.sw - g.pw;
  int hend = min(hstart + g.kh, g.H + g.ph);
  int wend = min(wstart + g.kw, g.W + g.pw);
  return (hend - hstart) * (wend - wstart);
}

template <typename T>
__global__ void avepool_fwd_k(const T* x, T* y, PoolGeom g) {
  int64_t total = (int64_t)g.N * g.Ho * g.Wo * g.C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % g.C;
    int64_t t = i / g.C;
    int ow = t % g.Wo; t /= g.Wo;
    int oh = t % g.Ho;
    int n = t / g.Ho;
    int psize = ave_pool_size(oh, ow, g);
    int h0 = max(oh * g.sh - g.ph, 0), w0 = max(ow * g.sw - g.pw, 0);
    int h1 = min(oh * g.sh - g.ph + g.kh, g.H);
    int w1 = min(ow * g.sw - g.pw + g.kw, g.W);
    float acc = 0.f;
    for (int h = h0; h < h1; ++h)
      for (int w = w0; w < w1; ++w)
        acc += to_f32(x[(((int64_t)n * g.H + h) * g.W + w) * g.C + c]);
    from_f32(acc / psize, y[i]);
  }
}

template <typename T>
__global__ void avepool_bwd_k(const T* dy, T* dx, PoolGeom g) {
  int64_t total = (int64_t)g.N * g.H * g.W * g.C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % g.C;
    int64_t t = i / g.C;
    int w = t % g.W; t /= g.W;
    int h = t % g.H;
    int n = t / g.H;
    int oh0 = (h + g.ph < g.kh) ? 0 : (h + g.ph - g.kh) / g.sh + 1;
    int oh1 = min((h + g.ph) / g.sh + 1, g.Ho);
    int ow0 = (w + g.pw < g.kw) ? 0 : (w + g.pw - g.kw) / g.sw + 1;
    int ow1 = min((w + g.pw) / g.sw + 1, g.Wo);
    float acc = 0.f;
    for (int oh = oh0; oh < oh1; ++oh)
      for (int ow = ow0; ow < ow1; ++ow) {
        int64_t oi = (((int64_t)n * g.Ho + oh) * g.Wo + ow) * g.C + c;
        acc += to_f32(dy[oi]) / ave_pool_size(oh, ow, g);
      }
    from_f32(acc, dx[i]);
  }
}

// stochastic pooling (pooling_layer.cu:81-160): train picks an element with
// probability proportional to its (nonnegative) activation; test is the
// activation-weighted average. mask reuses the maxpool backward.
template <typename T>
__global__ void stochpool_fwd_train_k(const T* x, T* y, uint8_t* mask, PoolGeom g,
                                      uint64_t seed) {
  int64_t total = (int64_t)g.N * g.Ho * g.Wo * g.C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % g.C;
    int64_t t = i / g.C;
    int ow = t % g.Wo; t /= g.Wo;
    int oh = t % g.Ho;
    int n = t / g.Ho;
    int h0 = max(oh * g.sh - g.ph, 0), w0 = max(ow * g.sw - g.pw, 0);
    int h1 = min(oh * g.sh - g.ph + g.kh, g.H);
    int w1 = min(ow * g.sw - g.pw + g.kw, g.W);
    float sum = 0.f;
    for (int h = h0; h < h1; ++h)
      for (int w = w0; w < w1; ++w)
        sum += to_f32(x[(((int64_t)n * g.H + h) * g.W + w) * g.C + c]);
    // one uniform per output element
    uint32_t r = philox4(seed, (uint64_t)i, (uint32_t)(i >> 32)).x;
    float thresh = (r >> 8) * (1.0f / 16777216.0f) * sum;
    float cum = 0.f;
    float pick = 0.f;
    int hb = oh * g.sh - g.ph, wb = ow * g.sw - g.pw;
    int pick_idx = (h0 - hb) * g.kw + (w0 - wb);
    bool done = false;
    for (int h = h0; h < h1 && !done; ++h)
      for (int w = w0; w < w1 && !done; ++w) {
        float v = to_f32(x[(((int64_t)n * g.H + h) * g.W + w) * g.C + c]);
        cum += v;
        pick = v; pick_idx = (h - hb) * g.kw + (w - wb);
        if (cum >= thresh) done = true;
      }
    from_f32(pick, y[i]);
    mask[i] = (uint8_t)pick_idx;
  }
}

template <typename T>
__global__ void stochpool_fwd_test_k(const T* x, T* y, PoolGeom g) {
  int64_t total = (int64_t)g.N * g.Ho * g.Wo * g.C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = i % g.C;
    int64_t t = i / g.C;
    int ow = t % g.Wo; t /= g.Wo;
    int oh = t % g.Ho;
    int n = t / g.Ho;
    int h0 = max(oh * g.sh - g.ph, 0), w0 = max(ow * g.sw - g.pw, 0);
    int h1 = min(oh * g.sh - g.ph + g.kh, g.H);
    int w1 = min(ow * g.sw - g.pw + g.kw, g.W);
    float num = 0.f, den = 0.f;
    for (int h = h0; h < h1; ++h)
      for (int w = w0; w < w1; ++w) {
        float v = to_f32(x[(((int64_t)n * g.H + h) * g.W + w) * g.C + c]);
        num += v * v;
        den += v;
      }
    from_f32(den > 0 ? num / den : 0.f, y[i]);
  }
}

extern "C" {

#define PS_POOL_LAUNCH(kern, count, ...) \
  kern<<<ew_grid(count), 256, 0, s>>>(__VA_ARGS__)
#define POOL_COMMA ,

void ps_maxpool_fwd_f32(const float* x, float* y, uint8_t* mask, const PoolGeom* g,
                        hipStream_t s) {
  int64_t rows = (int64_t)g->N * g->Ho * g->Wo;
  if (g->C % 4 == 0)
    PS_POOL_LAUNCH(maxpool_fwd_k<float POOL_COMMA 4>, rows * (g->C / 4), x, y, mask, *g);
  else
    PS_POOL_LAUNCH(maxpool_fwd_k<float POOL_COMMA 1>, rows * g->C, x, y, mask, *g);
}
void ps_maxpool_fwd_bf16(const void* x, void* y, uint8_t* mask, const PoolGeom* g,
                         hipStream_t s) {
  int64_t rows = (int64_t)g->N * g->Ho * g->Wo;
  if (g->C % 8 == 0)
    PS_POOL_LAUNCH(maxpool_fwd_k<__bf16 POOL_COMMA 8>, rows * (g->C / 8),
                   (const __bf16*)x, (__bf16*)y, mask, *g);
  else
    PS_POOL_LAUNCH(maxpool_fwd_k<__bf16 POOL_COMMA 1>, rows * g->C,
                   (const __bf16*)x, (__bf16*)y, mask, *g);
}
void ps_maxpool_bwd_f32(const float* dy, const uint8_t* mask, float* dx,
                        const PoolGeom* g, hipStream_t s) {
  int64_t rows = (int64_t)g->N * g->H * g->W;
  if (g->C % 4 == 0)
    PS_POOL_LAUNCH(maxpool_bwd_k<float POOL_COMMA 4>, rows * (g->C / 4), dy, mask, dx, *g);
  else
    PS_POOL_LAUNCH(maxpool_bwd_k<float POOL_COMMA 1>, rows * g->C, dy, mask, dx, *g);
}
void ps_maxpool_bwd_bf16(const void* dy, const uint8_t* mask, void* dx,
                         const PoolGeom* g, hipStream_t s) {
  int64_t rows = (int64_t)g->N * g->H * g->W;
  if (g->C % 8 == 0)
    PS_POOL_LAUNCH(maxpool_bwd_k<__bf16 POOL_COMMA 8>, rows * (g->C / 8),
                   (const __bf16*)dy, mask, (__bf16*)dx, *g);
  else
    PS_POOL_LAUNCH(maxpool_bwd_k<__bf16 POOL_COMMA 1>, rows * g->C,
                   (const __bf16*)dy, mask, (__bf16*)dx, *g);
}
void ps_avepool_fwd_f32(const float* x, float* y, const PoolGeom* g, hipStream_t s) {
  PS_POOL_LAUNCH(avepool_fwd_k<float>, (int64_t)g->N * g->Ho * g->Wo * g->C, x, y, *g);
}
void ps_avepool_fwd_bf16(const void* x, void* y, const PoolGeom* g, hipStream_t s) {
  PS_POOL_LAUNCH(avepool_fwd_k<__bf16>, (int64_t)g->N * g->Ho * g->Wo * g->C,
                 (const __bf16*)x, (__bf16*)y, *g);
}
void ps_avepool_bwd_f32(const float* dy, float* dx, const PoolGeom* g, hipStream_t s) {
  PS_POOL_LAUNCH(avepool_bwd_k<float>, (int64_t)g->N * g->H * g->W * g->C, dy, dx, *g);
}
void ps_avepool_bwd_bf16(const void* dy, void* dx, const PoolGeom* g, hipStream_t s) {
  PS_POOL_LAUNCH(avepool_bwd_k<__bf16>, (int64_t)g->N * g->H * g->W * g->C,
                 (const __bf16*)dy, (__bf16*)dx, *g);
}
void ps_stochpool_fwd_train_f32(const float* x, float* y, uint8_t* mask,
                                const PoolGeom* g, uint64_t seed, hipStream_t s) {
  PS_POOL_LAUNCH(stochpool_fwd_train_k<float>, (int64_t)g->N * g->Ho * g->Wo * g->C,
                 x, y, mask, *g, seed);
}
void ps_stochpool_fwd_test_f32(const float* x, float* y, const PoolGeom* g,
                               hipStream_t s) {
  PS_POOL_LAUNCH(stochpool_fwd_test_k<float>, (int64_t)g->N * g->Ho * g->Wo * g->C,
                 x, y, *g);
}

}  // extern "C"

}  // namespace ps
