// im2col / col2im for NHWC activations, producing the TRANSPOSED column
// matrix colT[(n,oh,ow)][k] with k = (g*KH*KW + kh*KW + kw)*Cg + cg --
// K-contiguous rows that feed the NT MFMA GEMM directly (conv fwd/wgrad),
// with per-group k-slices contiguous.
//
// Replaces reference src/caffe/util/im2col.cu:12-144 (which is NCHW and
// column-major for cuBLAS). col2im is gather-form (no atomics), like the
// reference's col2im_gpu_kernel.

#include "ps_common.h"
#include "ps_api.h"

namespace ps {

// one thread per (np, g, khw, chunk-of-V-channels); V=16B vectors when the
// group channel count allows (guide Guideline 13: always vectorize)
template <typename T, int V>
__global__ void im2col_nhwc_k(const T* __restrict__ x, T* __restrict__ colT,
                              ConvGeom g, int ldcol) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  const int Cg = g.C / g.G;
  const int CV = Cg / V;
  const int KHW = g.kh * g.kw;
  const int Kcol = g.G * KHW * Cg;
  const int64_t NP = (int64_t)g.N * g.Ho * g.Wo;
  int64_t total = NP * g.G * KHW * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t t = i / CV;
    int khw = t % KHW; t /= KHW;
    int grp = t % g.G;
    int64_t np = t / g.G;
    int ow = np % g.Wo;
    int64_t t2 = np / g.Wo;
    int oh = t2 % g.Ho;
    int n = t2 / g.Ho;
    int kkh = khw / g.kw, kkw = khw % g.kw;
    int ih = oh * g.sh - g.ph + kkh;
    int iw = ow * g.sw - g.pw + kkw;
    T* dst = colT + np * (int64_t)ldcol
             + ((int64_t)grp * KHW + khw) * Cg + cv * V;
    vec_t v = {};
    if (ih >= 0 && ih < g.H && iw >= 0 && iw < g.W) {
      const T* src = x + (((int64_t)n * g.H + ih) * g.W + iw) * g.C
                     + grp * Cg + cv * V;
      v = *reinterpret_cast<const vec_t*>(src);
    }
    *reinterpret_cast<vec_t*>(dst) = v;
  }
}

// Row-run variant for G==1: in NHWC, for a fixed kh the whole (kw, c) span
// of a patch row is CONTIGUOUS in x (kw*C + c walks iw*C + c). One thread
// per (np, kh) copies kw*C elems with a 16B-vector loop -- the fast path
// for small-C first layers (conv1: C=3, kw*C=33) where per-channel chunks
// cannot vectorize.
template <typename T, int VBYTES>
__global__ void im2col_nhwc_rowrun_k(const T* __restrict__ x,
                                     T* __restrict__ colT, ConvGeom g,
                                     int ldcol) {
  constexpr int V = VBYTES / sizeof(T);
  // element-aligned only: run starts (np*Kcol, kh*RUN) are not 16B-aligned
  // for odd K (conv1 Kcol=363); clang emits the widest legal loads
  typedef T vec_t __attribute__((ext_vector_type(V), aligned(sizeof(T))));
  const int RUN = g.kw * g.C;  // elems per (np, kh)
  const int Kcol = g.kh * RUN;
  const int64_t NP = (int64_t)g.N * g.Ho * g.Wo;
  int64_t total = NP * g.kh;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int kkh = i % g.kh;
    int64_t np = i / g.kh;
    int ow = np % g.Wo;
    int64_t t2 = np / g.Wo;
    int oh = t2 % g.Ho;
    int n = t2 / g.Ho;
    int ih = oh * g.sh - g.ph + kkh;
    int iw0 = ow * g.sw - g.pw;
    T* dst = colT + np * (int64_t)ldcol + (int64_t)kkh * RUN;
    if (ih < 0 || ih >= g.H) {
      for (int e = 0; e < RUN; ++e) dst[e] = (T)0.0f;
      continue;
    }
    const T* src = x + (((int64_t)n * g.H + ih) * g.W + iw0) * g.C;
    // valid elem range within the run: iw in [0, W)
    int e_lo = iw0 < 0 ? -iw0 * g.C : 0;
    int e_hi = min(RUN, (g.W - iw0) * g.C);
    int e = 0;
    for (; e < e_lo; ++e) dst[e] = (T)0.0f;
    for (; e + V <= e_hi; e += V)
      *reinterpret_cast<vec_t*>(&dst[e]) =
          *reinterpret_cast<const vec_t*>(&src[e]);
    for (; e < e_hi; ++e) dst[e] = src[e];
    for (; e < RUN; ++e) dst[e] = (T)0.0f;
  }
}


// exact small-divisor division via f32 reciprocal (operands < 2^24)
__device__ inline int fdiv_fix2(int x2, int d, float inv, int& rem) {
  int q = (int)((float)x2 * inv);
  rem = x2 - q * d;
  if (rem < 0) { --q; rem += d; }
  else if (rem >= d) { ++q; rem -= d; }
  return q;
}

// LDS row-staged variant for small-C first layers (G==1, C % 8 != 0 --
// conv1-style). The rowrun kernel above issues unaligned 16 B ops on 66 B
// runs (every op splits into two requests) and re-reads overlapping
// windows from HBM; here a block stages the kh input rows of one (n, oh)
// in LDS once (coalesced), then emits the (Wo x Kcol) output span as
// GLOBALLY 16 B-ALIGNED vector stores, decoding (ow, kh, r) incrementally
// (one f32-reciprocal divide per 8 elements, no per-element IDIV).
template <typename T>
__global__ void im2col_rowstage_k(const T* __restrict__ x,
                                  T* __restrict__ colT, ConvGeom g,
                                  int rsw, int ldcol) {
  extern __shared__ char smem[];
  T* xs = (T*)smem;  // [kh][rsw]; staged elem s <-> iw_elem = s - pw*C
  constexpr int V = 16 / (int)sizeof(T);
  typedef T uvec_t __attribute__((ext_vector_type(V), aligned(sizeof(T))));
  typedef T avec_t __attribute__((ext_vector_type(V)));
  const int RUN = g.kw * g.C;
  const int Kcol = g.kh * RUN;
  const int WC = g.W * g.C;
  const int row_elems = g.Wo * Kcol;  // contiguous colT span per (n, oh)
  const float inv_Kcol = 1.0f / Kcol, inv_RUN = 1.0f / RUN;
  for (int64_t bo = blockIdx.x; bo < (int64_t)g.N * g.Ho; bo += gridDim.x) {
    const int oh = (int)(bo % g.Ho);
    const int n = (int)(bo / g.Ho);
    // stage kh rows (zero rows for padded ih; left/right iw pad as zeros)
    for (int kh = 0; kh < g.kh; ++kh) {
      const int ih = oh * g.sh - g.ph + kh;
      T* row = xs + kh * rsw;
      if (ih < 0 || ih >= g.H) {
        for (int e = threadIdx.x; e < rsw; e += blockDim.x) row[e] = (T)0.0f;
      } else {
        const T* src = x + (((int64_t)n * g.H + ih) * g.W) * g.C - g.pw * g.C;
        for (int e = threadIdx.x * V; e < rsw; e += blockDim.x * V) {
          uvec_t v;
#pragma unroll
          for (int j2 = 0; j2 < V; ++j2) {
            const int ie = e + j2 - g.pw * g.C;
            v[j2] = (ie >= 0 && ie < WC && e + j2 < rsw) ? src[e + j2 + 0]
                                                         : (T)0.0f;
          }
          if (e + V <= rsw)
            *reinterpret_cast<uvec_t*>(&row[e]) = v;
          else
            for (int j2 = 0; e + j2 < rsw; ++j2) row[e + j2] = v[j2];
        }
      }
    }
    __syncthreads();
    if (ldcol != Kcol) {
      // padded rows (ldcol % V == 0): each colT row is [Kcol valid)[pad]
      // and V-aligned on its own, so emit row-by-row with no head/tail
      T* out0 = colT + bo * (int64_t)g.Wo * ldcol;
      const int pieces = ldcol / V;
      const int total_p = g.Wo * pieces;
      for (int pi = threadIdx.x; pi < total_p; pi += blockDim.x) {
        const int ow = pi / pieces;
        int e = (pi - ow * pieces) * V;
        avec_t v;
        if (e >= Kcol) {
          v = avec_t{};  // zero pad columns
        } else {
          int r;
          int kh = fdiv_fix2(e, RUN, inv_RUN, r);
          const int sb0 = ow * g.sw * g.C;
#pragma unroll
          for (int j2 = 0; j2 < V; ++j2) {
            v[j2] = (e + j2 < Kcol) ? xs[kh * rsw + sb0 + r] : (T)0.0f;
            if (++r == RUN) { r = 0; ++kh; }
          }
        }
        *reinterpret_cast<avec_t*>(&out0[(int64_t)ow * ldcol + e]) = v;
      }
      __syncthreads();
      continue;
    }
    // emit: flat span [np0*Kcol, np0*Kcol + row_elems), aligned V pieces
    T* out = colT + bo * (int64_t)g.Wo * Kcol;
    const int64_t gbase = bo * (int64_t)g.Wo * Kcol;  // global elem index
    // first aligned element within the span
    const int head = (int)((V - (gbase & (V - 1))) & (V - 1));
    for (int e = head + (int)threadIdx.x * V; e < row_elems;
         e += blockDim.x * V) {
      int rem;
      int ow = fdiv_fix2(e, Kcol, inv_Kcol, rem);
      int r;
      int kh = fdiv_fix2(rem, RUN, inv_RUN, r);
      int sbase = ow * g.sw * g.C;
      avec_t v;
#pragma unroll
      for (int j2 = 0; j2 < V; ++j2) {
        v[j2] = xs[kh * rsw + sbase + r];
        if (++r == RUN) {
          r = 0;
          if (++kh == g.kh) {
            kh = 0;
            ++ow;
            sbase = ow * g.sw * g.C;
          }
        }
      }
      if (e + V <= row_elems)
        *reinterpret_cast<avec_t*>(&out[e]) = v;
      else
        for (int j2 = 0; e + j2 < row_elems; ++j2) out[e + j2] = v[j2];
    }
    // head elements (before the first aligned piece)
    for (int e = threadIdx.x; e < head && e < row_elems; e += blockDim.x) {
      int rem;
      int ow = fdiv_fix2(e, Kcol, inv_Kcol, rem);
      int r;
      int kh = fdiv_fix2(rem, RUN, inv_RUN, r);
      out[e] = xs[kh * rsw + ow * g.sw * g.C + r];
    }
    __syncthreads();
  }
}

// gather: one thread per (n, ih, iw, V-chunk of channels) -- channel chunks
// are contiguous in BOTH dcolT rows and NHWC dx, so loads/stores vectorize
template <typename T, int V>
__global__ void col2im_nhwc_k(const T* __restrict__ colT, T* __restrict__ dx,
                              ConvGeom g) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  typedef float facc_t __attribute__((ext_vector_type(V)));
  const int Cg = g.C / g.G;
  const int CV = Cg / V;
  const int KHW = g.kh * g.kw;
  const int Kcol = g.G * KHW * Cg;
  int64_t total = (int64_t)g.N * g.H * g.W * g.G * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t t = i / CV;
    int grp = t % g.G; t /= g.G;
    int iw = t % g.W; t /= g.W;
    int ih = t % g.H;
    int n = t / g.H;
    facc_t acc = {};
    for (int kkh = 0; kkh < g.kh; ++kkh) {
      int oh_num = ih + g.ph - kkh;
      if (oh_num < 0 || oh_num % g.sh) continue;
      int oh = oh_num / g.sh;
      if (oh >= g.Ho) continue;
      for (int kkw = 0; kkw < g.kw; ++kkw) {
        int ow_num = iw + g.pw - kkw;
        if (ow_num < 0 || ow_num % g.sw) continue;
        int ow = ow_num / g.sw;
        if (ow >= g.Wo) continue;
        int64_t np = ((int64_t)n * g.Ho + oh) * g.Wo + ow;
        int k = (grp * KHW + kkh * g.kw + kkw) * Cg + cv * V;
        vec_t v = *reinterpret_cast<const vec_t*>(&colT[np * Kcol + k]);
#pragma unroll
        for (int j = 0; j < V; ++j) acc[j] += to_f32(v[j]);
      }
    }
    vec_t out;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      T o;
      from_f32(acc[j], o);
      out[j] = o;
    }
    *reinterpret_cast<vec_t*>(
        &dx[(((int64_t)n * g.H + ih) * g.W + iw) * g.C + grp * Cg + cv * V]) = out;
  }
}

// NCHW [Co][Ci][kh][kw] -> khwc-per-group [Co][kh][kw][Cg] weight repack
// (and its inverse for the weight gradient). Co already encodes the group.
template <typename TI, typename TO>
__global__ void weight_to_khwc_k(const TI* src, TO* dst, int Co, int Cig,
                                 int KH, int KW) {
  int64_t total = (int64_t)Co * Cig * KH * KW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    // src index decomposition: [co][ci][kh][kw]
    int kkw = i % KW;
    int64_t t = i / KW;
    int kkh = t % KH; t /= KH;
    int ci = t % Cig;
    int co = t / Cig;
    int64_t dst_i = (((int64_t)co * KH + kkh) * KW + kkw) * Cig + ci;
    from_f32(to_f32(src[i]), dst[dst_i]);
  }
}

// NCHW [Co][Cig][kh][kw] -> TRANSPOSED khwc per group: [G][Kg][Cog] with
// Kg = kh*kw*Cig rows and the group's output channels contiguous -- the
// K-last B operand of the dgrad NT GEMM (dcol = dy @ W): avoids the
// K-major scatter staging path entirely.
template <typename TI, typename TO>
__global__ void weight_to_khwc_tr_k(const TI* src, TO* dst, int Co, int Cig,
                                    int KH, int KW, int G) {
  const int Cog = Co / G;
  const int Kg = KH * KW * Cig;
  int64_t total = (int64_t)Co * Kg;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    // src index decomposition: [co][ci][kh][kw]
    int kkw = i % KW;
    int64_t t = i / KW;
    int kkh = t % KH; t /= KH;
    int ci = t % Cig;
    int co = t / Cig;
    int grp = co / Cog, cog = co % Cog;
    int kg = (kkh * KW + kkw) * Cig + ci;
    from_f32(to_f32(src[i]),
             dst[((int64_t)grp * Kg + kg) * Cog + cog]);
  }
}

// Fused repack: one read of the fp32 master emits BOTH the khwc forward
// operand and its per-group transpose for dgrad (they were two kernels =
// two master reads + two launches per conv per iteration; GoogLeNet has 59
// convs).
template <typename TI, typename TO>
__global__ void weight_to_khwc_both_k(const TI* src, TO* dst, TO* dst_tr,
                                      int Co, int Cig, int KH, int KW, int G,
                                      int ldk) {
  const int Cog = Co / G;
  const int Kg = KH * KW * Cig;
  int64_t total = (int64_t)Co * Kg;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int kkw = i % KW;
    int64_t t = i / KW;
    int kkh = t % KH; t /= KH;
    int ci = t % Cig;
    int co = t / Cig;
    const float v = to_f32(src[i]);
    const int kg = (kkh * KW + kkw) * Cig + ci;
    from_f32(v, dst[(int64_t)co * ldk + kg]);
    from_f32(v, dst_tr[((int64_t)(co / Cog) * Kg + kg) * Cog + co % Cog]);
  }
}

// zero the pad columns [c_lo, ld) of a row-major matrix (colT pad strips)
template <typename T>
__global__ void zero_cols_k(T* m, int64_t rows, int ld, int c_lo) {
  const int pw2 = ld - c_lo;
  int64_t total = rows * pw2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x)
    m[(i / pw2) * (int64_t)ld + c_lo + i % pw2] = (T)0.0f;
}

// dgrad-as-forward weight repack: W[co][ci][kh][kw] (NCHW, per-group) ->
// wrot[(grp*Cg + ci)][ ((KH-1-kh)*KW + (KW-1-kw)) * Cog + cog ] -- the
// K-last B operand of the stride-1 dgrad-as-conv GEMM
// (dx = conv(dy, rot180(W), pad = K-1-p)).
template <typename TI, typename TO>
__global__ void weight_to_dgrad_k(const TI* src, TO* dst, int Co, int Cig,
                                  int KH, int KW, int G) {
  const int Cog = Co / G;
  const int K2 = KH * KW * Cog;
  int64_t total = (int64_t)Co * Cig * KH * KW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int kkw = i % KW;
    int64_t t = i / KW;
    int kkh = t % KH; t /= KH;
    int ci = t % Cig;
    int co = t / Cig;
    const int grp = co / Cog, cog = co % Cog;
    const int k2 = ((KH - 1 - kkh) * KW + (KW - 1 - kkw)) * Cog + cog;
    from_f32(to_f32(src[i]),
             dst[((int64_t)grp * Cig + ci) * K2 + k2]);
  }
}

template <typename TI, typename TO>
__global__ void weight_from_khwc_k(const TI* src, TO* dst, int Co, int Cig,
                                   int KH, int KW, int ld, float beta) {
  int64_t total = (int64_t)Co * Cig * KH * KW;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int kkw = i % KW;
    int64_t t = i / KW;
    int kkh = t % KH; t /= KH;
    int ci = t % Cig;
    int co = t / Cig;
    int64_t src_i = (int64_t)co * ld + ((int64_t)kkh * KW + kkw) * Cig + ci;
    float v = to_f32(src[src_i]);
    if (beta != 0.f) v += beta * to_f32(dst[i]);
    from_f32(v, dst[i]);
  }
}

// NCHW <-> NHWC activation converters (layout boundary with CPU/proto side)
template <typename TI, typename TO>
__global__ void nchw_to_nhwc_k(const TI* src, TO* dst, int N, int C, int H, int W) {
  int64_t total = (int64_t)N * C * H * W;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    // i indexes NHWC dst
    int c = i % C;
    int64_t t = i / C;
    int w = t % W; t /= W;
    int h = t % H;
    int n = t / H;
    from_f32(to_f32(src[(((int64_t)n * C + c) * H + h) * W + w]), dst[i]);
  }
}

// NHWC channel-block copy: out[:, c_off:c_off+C_in] = in (rows = N*H*W).
// Serves CONCAT forward (one call per bottom) and SLICE/CONCAT backward.
template <typename T, int V>
__global__ void chan_copy_k(const T* __restrict__ in, T* __restrict__ out,
                            int64_t rows, int C_in, int C_out, int c_off) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  const int CV = C_in / V;
  int64_t total = rows * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t r = i / CV;
    *reinterpret_cast<vec_t*>(&out[r * C_out + c_off + cv * V]) =
        *reinterpret_cast<const vec_t*>(&in[r * C_in + cv * V]);
  }
}

// 4-way fused channel concat/split: one launch per inception join instead
// of one per branch (GoogLeNet: 72 -> 18 concat/slice launches per step).
// Thread i owns one V-wide chunk of the WIDE tensor; an unrolled compare
// over the <=4 channel ranges picks the narrow tensor it pairs with.
struct Chan4 {
  void* p[4];    // narrow tensors (src for concat, dst for split)
  const void* m[4];  // optional relu masks (post-relu activations): split
                     // outputs are zeroed where the activation is zero --
                     // the consumer ReLU's backward fused into the scatter
  int c_end[4];  // exclusive channel end of each range in the wide tensor
  int c_begin;   // first channel this launch covers
  int n;         // live entries
};

template <typename T, int V, bool GATHER,  // GATHER: wide->narrow (split)
          bool MASKED = false>
__global__ void chan_concat4_k(T* __restrict__ wide, Chan4 t, int64_t rows,
                               int C_wide) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  const int CV = (t.c_end[t.n - 1] - t.c_begin) / V;  // slab chunks only
  int64_t total = rows * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = (int)(i % CV);
    int64_t r = i / CV;
    int c = t.c_begin + cv * V;
    int j = 0, c0 = t.c_begin;
#pragma unroll
    for (int q = 0; q < 3; ++q)
      if (q < t.n - 1 && c >= t.c_end[q]) { j = q + 1; c0 = t.c_end[q]; }
    const int Cn = t.c_end[j] - c0;
    T* nar = (T*)t.p[j] + r * Cn + (c - c0);
    vec_t* wp = reinterpret_cast<vec_t*>(&wide[r * C_wide + c]);
    if (GATHER) {
      vec_t v = *wp;
      if (MASKED && t.m[j]) {
        // post-relu activations are never negative: x == 0 <=> clamped
        const vec_t x = *reinterpret_cast<const vec_t*>(
            (const T*)t.m[j] + r * Cn + (c - c0));
#pragma unroll
        for (int e = 0; e < V; ++e)
          if (to_f32(x[e]) == 0.0f) v[e] = (T)0.0f;
      }
      *reinterpret_cast<vec_t*>(nar) = v;
    } else {
      *wp = *reinterpret_cast<const vec_t*>(nar);
    }
  }
}

// gather variant: out = in[:, c_off:c_off+C_out] (slice forward / concat bwd)
template <typename T, int V>
__global__ void chan_slice_k(const T* __restrict__ in, T* __restrict__ out,
                             int64_t rows, int C_in, int C_out, int c_off) {
  typedef T vec_t __attribute__((ext_vector_type(V)));
  const int CV = C_out / V;
  int64_t total = rows * CV;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int cv = i % CV;
    int64_t r = i / CV;
    *reinterpret_cast<vec_t*>(&out[r * C_out + cv * V]) =
        *reinterpret_cast<const vec_t*>(&in[r * C_in + c_off + cv * V]);
  }
}

extern "C" {

// n <= 4 narrow tensors <-> one wide tensor in a single launch.
// Requirements checked by caller: every range width and boundary divisible
// by V. gather=1 splits wide->narrow, 0 concats narrow->wide.
static inline Chan4 chan4_pack(void* const* ptrs, const int* c_end, int n,
                               int c_begin, const void* const* masks) {
  Chan4 t;
  t.n = n;
  t.c_begin = c_begin;
  for (int i = 0; i < 4; ++i) {
    t.p[i] = i < n ? const_cast<void*>(ptrs[i]) : nullptr;
    t.m[i] = (masks && i < n) ? masks[i] : nullptr;
    t.c_end[i] = i < n ? c_end[i] : (n ? c_end[n - 1] : 0);
  }
  return t;
}

void ps_chan_concat4_f32(float* wide, void* const* ptrs, const int* c_end,
                         int n, int c_begin, int64_t rows, int C_wide,
                         int gather, const void* const* masks,
                         hipStream_t s) {
  Chan4 t = chan4_pack(ptrs, c_end, n, c_begin, masks);
  int64_t work = rows * ((c_end[n - 1] - c_begin) / 4);
  if (gather && masks)
    chan_concat4_k<float, 4, true, true>
        <<<ew_grid(work), 256, 0, s>>>(wide, t, rows, C_wide);
  else if (gather)
    chan_concat4_k<float, 4, true>
        <<<ew_grid(work), 256, 0, s>>>(wide, t, rows, C_wide);
  else
    chan_concat4_k<float, 4, false>
        <<<ew_grid(work), 256, 0, s>>>(wide, t, rows, C_wide);
}
void ps_chan_concat4_bf16(void* wide, void* const* ptrs, const int* c_end,
                          int n, int c_begin, int64_t rows, int C_wide,
                          int gather, const void* const* masks,
                          hipStream_t s) {
  Chan4 t = chan4_pack(ptrs, c_end, n, c_begin, masks);
  int64_t work = rows * ((c_end[n - 1] - c_begin) / 8);
  if (gather && masks)
    chan_concat4_k<__bf16, 8, true, true>
        <<<ew_grid(work), 256, 0, s>>>((__bf16*)wide, t, rows, C_wide);
  else if (gather)
    chan_concat4_k<__bf16, 8, true>
        <<<ew_grid(work), 256, 0, s>>>((__bf16*)wide, t, rows, C_wide);
  else
    chan_concat4_k<__bf16, 8, false>
        <<<ew_grid(work), 256, 0, s>>>((__bf16*)wide, t, rows, C_wide);
}

void ps_chan_copy_f32(const float* in, float* out, int64_t rows, int C_in,
                      int C_out, int c_off, hipStream_t s) {
  if (C_in % 4 == 0 && c_off % 4 == 0)
    chan_copy_k<float, 4><<<ew_grid(rows * (C_in / 4)), 256, 0, s>>>(
        in, out, rows, C_in, C_out, c_off);
  else
    chan_copy_k<float, 1><<<ew_grid(rows * C_in), 256, 0, s>>>(
        in, out, rows, C_in, C_out, c_off);
}
void ps_chan_copy_bf16(const void* in, void* out, int64_t rows, int C_in,
                       int C_out, int c_off, hipStream_t s) {
  if (C_in % 8 == 0 && c_off % 8 == 0)
    chan_copy_k<__bf16, 8><<<ew_grid(rows * (C_in / 8)), 256, 0, s>>>(
        (const __bf16*)in, (__bf16*)out, rows, C_in, C_out, c_off);
  else
    chan_copy_k<__bf16, 1><<<ew_grid(rows * C_in), 256, 0, s>>>(
        (const __bf16*)in, (__bf16*)out, rows, C_in, C_out, c_off);
}
void ps_chan_slice_f32(const float* in, float* out, int64_t rows, int C_in,
                       int C_out, int c_off, hipStream_t s) {
  if (C_out % 4 == 0 && c_off % 4 == 0)
    chan_slice_k<float, 4><<<ew_grid(rows * (C_out / 4)), 256, 0, s>>>(
        in, out, rows, C_in, C_out, c_off);
  else
    chan_slice_k<float, 1><<<ew_grid(rows * C_out), 256, 0, s>>>(
        in, out, rows, C_in, C_out, c_off);
}
void ps_chan_slice_bf16(const void* in, void* out, int64_t rows, int C_in,
                        int C_out, int c_off, hipStream_t s) {
  if (C_out % 8 == 0 && c_off % 8 == 0)
    chan_slice_k<__bf16, 8><<<ew_grid(rows * (C_out / 8)), 256, 0, s>>>(
        (const __bf16*)in, (__bf16*)out, rows, C_in, C_out, c_off);
  else
    chan_slice_k<__bf16, 1><<<ew_grid(rows * C_out), 256, 0, s>>>(
        (const __bf16*)in, (__bf16*)out, rows, C_in, C_out, c_off);
}

void ps_im2col_nhwc_f32(const float* x, float* colT, const ConvGeom* g,
                        int ldcol, hipStream_t s) {
  int Cg = g->C / g->G;
  int64_t base = (int64_t)g->N * g->Ho * g->Wo * g->G * g->kh * g->kw;
  if (Cg % 4 == 0)
    im2col_nhwc_k<float, 4><<<ew_grid(base * (Cg / 4)), 256, 0, s>>>(x, colT, *g, ldcol);
  else if (g->G == 1) {
    int rsw = ((g->Wo - 1) * g->sw + g->kw) * g->C;
    int64_t lds = (int64_t)g->kh * rsw * sizeof(float);
    if (lds <= (48 << 10)) {
      int64_t blocks = (int64_t)g->N * g->Ho;
      if (blocks > (64 << 10)) blocks = 64 << 10;
      im2col_rowstage_k<float>
          <<<dim3((unsigned)blocks), 256, lds, s>>>(x, colT, *g, rsw, ldcol);
    } else {
      im2col_nhwc_rowrun_k<float, 16>
          <<<ew_grid((int64_t)g->N * g->Ho * g->Wo * g->kh), 256, 0, s>>>(
              x, colT, *g, ldcol);
    }
  }
  else
    im2col_nhwc_k<float, 1><<<ew_grid(base * Cg), 256, 0, s>>>(x, colT, *g, ldcol);
}
void ps_im2col_nhwc_bf16(const void* x, void* colT, const ConvGeom* g,
                         int ldcol, hipStream_t s) {
  int Cg = g->C / g->G;
  int64_t base = (int64_t)g->N * g->Ho * g->Wo * g->G * g->kh * g->kw;
  if (Cg % 8 == 0)
    im2col_nhwc_k<__bf16, 8><<<ew_grid(base * (Cg / 8)), 256, 0, s>>>(
        (const __bf16*)x, (__bf16*)colT, *g, ldcol);
  else if (g->G == 1) {
    int rsw = ((g->Wo - 1) * g->sw + g->kw) * g->C;
    int64_t lds = (int64_t)g->kh * rsw * sizeof(__bf16);
    if (lds <= (48 << 10)) {
      int64_t blocks = (int64_t)g->N * g->Ho;
      if (blocks > (64 << 10)) blocks = 64 << 10;
      im2col_rowstage_k<__bf16>
          <<<dim3((unsigned)blocks), 256, lds, s>>>((const __bf16*)x,
                                                    (__bf16*)colT, *g, rsw,
                                                    ldcol);
    } else {
      im2col_nhwc_rowrun_k<__bf16, 16>
          <<<ew_grid((int64_t)g->N * g->Ho * g->Wo * g->kh), 256, 0, s>>>(
              (const __bf16*)x, (__bf16*)colT, *g, ldcol);
    }
  }
  else
    im2col_nhwc_k<__bf16, 1><<<ew_grid(base * Cg), 256, 0, s>>>(
        (const __bf16*)x, (__bf16*)colT, *g, ldcol);
}
void ps_col2im_nhwc_f32(const float* colT, float* dx, const ConvGeom* g, hipStream_t s) {
  int Cg = g->C / g->G;
  int64_t base = (int64_t)g->N * g->H * g->W * g->G;
  if (Cg % 4 == 0)
    col2im_nhwc_k<float, 4><<<ew_grid(base * (Cg / 4)), 256, 0, s>>>(colT, dx, *g);
  else
    col2im_nhwc_k<float, 1><<<ew_grid(base * Cg), 256, 0, s>>>(colT, dx, *g);
}
void ps_col2im_nhwc_bf16(const void* colT, void* dx, const ConvGeom* g, hipStream_t s) {
  int Cg = g->C / g->G;
  int64_t base = (int64_t)g->N * g->H * g->W * g->G;
  if (Cg % 8 == 0)
    col2im_nhwc_k<__bf16, 8><<<ew_grid(base * (Cg / 8)), 256, 0, s>>>(
        (const __bf16*)colT, (__bf16*)dx, *g);
  else
    col2im_nhwc_k<__bf16, 1><<<ew_grid(base * Cg), 256, 0, s>>>(
        (const __bf16*)colT, (__bf16*)dx, *g);
}
void ps_weight_to_khwc_f32(const float* src, float* dst, int Co, int Cig,
                           int KH, int KW, hipStream_t s) {
  weight_to_khwc_k<float, float>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(src, dst, Co, Cig, KH, KW);
}
void ps_weight_to_khwc_f32_bf16(const float* src, void* dst, int Co, int Cig,
                                int KH, int KW, hipStream_t s) {
  weight_to_khwc_k<float, __bf16>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(src, (__bf16*)dst,
                                                            Co, Cig, KH, KW);
}
void ps_weight_to_khwc_tr_f32(const float* src, float* dst, int Co, int Cig,
                              int KH, int KW, int G, hipStream_t s) {
  weight_to_khwc_tr_k<float, float>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(src, dst, Co, Cig,
                                                            KH, KW, G);
}
void ps_weight_to_khwc_tr_f32_bf16(const float* src, void* dst, int Co, int Cig,
                                   int KH, int KW, int G, hipStream_t s) {
  weight_to_khwc_tr_k<float, __bf16>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(
          src, (__bf16*)dst, Co, Cig, KH, KW, G);
}
void ps_weight_to_khwc_both_f32(const float* src, float* dst, float* dst_tr,
                                int Co, int Cig, int KH, int KW, int G,
                                int ldk, hipStream_t s) {
  weight_to_khwc_both_k<float, float>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(
          src, dst, dst_tr, Co, Cig, KH, KW, G, ldk);
}
void ps_weight_to_khwc_both_f32_bf16(const float* src, void* dst,
                                     void* dst_tr, int Co, int Cig, int KH,
                                     int KW, int G, int ldk, hipStream_t s) {
  weight_to_khwc_both_k<float, __bf16>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(
          src, (__bf16*)dst, (__bf16*)dst_tr, Co, Cig, KH, KW, G, ldk);
}
void ps_zero_cols_f32(float* m, int64_t rows, int ld, int c_lo,
                      hipStream_t s) {
  zero_cols_k<float>
      <<<ew_grid(rows * (ld - c_lo)), 256, 0, s>>>(m, rows, ld, c_lo);
}
void ps_zero_cols_bf16(void* m, int64_t rows, int ld, int c_lo,
                       hipStream_t s) {
  zero_cols_k<__bf16>
      <<<ew_grid(rows * (ld - c_lo)), 256, 0, s>>>((__bf16*)m, rows, ld,
                                                   c_lo);
}
void ps_weight_to_dgrad_f32(const float* src, float* dst, int Co, int Cig,
                            int KH, int KW, int G, hipStream_t s) {
  weight_to_dgrad_k<float, float>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(src, dst, Co,
                                                            Cig, KH, KW, G);
}
void ps_weight_to_dgrad_f32_bf16(const float* src, void* dst, int Co,
                                 int Cig, int KH, int KW, int G,
                                 hipStream_t s) {
  weight_to_dgrad_k<float, __bf16>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(
          src, (__bf16*)dst, Co, Cig, KH, KW, G);
}
void ps_weight_from_khwc_f32(const float* src, float* dst, int Co, int Cig,
                             int KH, int KW, int ld, float beta,
                             hipStream_t s) {
  weight_from_khwc_k<float, float>
      <<<ew_grid((int64_t)Co * Cig * KH * KW), 256, 0, s>>>(src, dst, Co, Cig,
                                                            KH, KW, ld, beta);
}

}  // extern "C"

}  // namespace ps
