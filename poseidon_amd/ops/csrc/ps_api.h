// C API between the HIP kernel translation units and the torch bindings.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

// Implicit-GEMM gather descriptor: the GEMM's A operand (conv fwd) or
// K-major B operand (conv wgrad) is the im2col matrix, gathered on the fly
// from NHWC x instead of materialized. Out-of-bounds (padding) lanes load
// from a 16-byte zero page.
struct GatherDesc {
  const void* x;      // NHWC activations
  const void* zero;   // >=16B of zeros
  int C, H, W, Ho, Wo;
  int kh, kw, sh, sw, ph, pw;
  int Cg, c0;         // group channel count and channel offset
  int kg_max;         // k >= kg_max reads the zero page (padded N/K)
  // f32 reciprocals for division-free index decode (exact with the +-1
  // fixup in the kernel; operands < 2^24)
  float inv_Cg, inv_kw, inv_Wo, inv_Ho;
};

inline void ps_fill_gather_inv(GatherDesc* g) {
  g->inv_Cg = 1.0f / g->Cg;
  g->inv_kw = 1.0f / g->kw;
  g->inv_Wo = 1.0f / g->Wo;
  g->inv_Ho = 1.0f / g->Ho;
}

struct GemmArgs {
  const void* A;
  const void* B;
  void* C;
  const float* bias;  // may be null
  int M, N, K;
  int64_t lda, ldb, ldc;
  int64_t strideA, strideB, strideC;
  int batch;
  float alpha, beta;
  bool a_klast, b_klast;
  // split-K (for small-tile huge-K GEMMs, e.g. conv wgrad): when splitk > 1,
  // ws must hold splitk * M * N floats and batch must be 1.
  void* ws;
  int splitk;
  // implicit-GEMM: gather A (K-last rows = im2col rows) / gather B (K-major)
  const GatherDesc* gather_a;
  const GatherDesc* gather_b;
  // fused epilogue: y = max(y, 0) (conv/IP + in-place ReLU pairs)
  bool relu;
  // caller guarantees C is already zero (net-level zero_mt): atomic
  // split-K paths skip their per-launch memset
  bool c_prezeroed;
};

struct PoolGeom {
  int N, C, H, W, Ho, Wo;
  int kh, kw, sh, sw, ph, pw;
};

struct ConvGeom {
  int N, C, H, W;
  int Ho, Wo;
  int kh, kw, sh, sw, ph, pw;
  int G;
};

// Tile selection shared by the launcher (grid) and bindings (split-K
// workspace sizing). Cost model: MFMA time on the PADDED output plus LDS
// staging traffic (each A panel is re-staged once per N-tile and vice
// versa). ALPHA ~= MACs the matrix cores retire per element the memory
// system delivers (bf16: ~1.25e15 MAC/s vs ~3.2e12 elem/s -> ~400).
inline void ps_pick_gemm_tile(int M, int N, int* bm_out, int* bn_out) {
  const int cand[6][2] = {{128, 128}, {128, 32}, {32, 128}, {64, 64},
                          {128, 16},  {16, 128}};
  const double ALPHA = 400.0;
  double best = 0;
  for (int i = 0; i < 6; ++i) {
    int bm = cand[i][0], bn = cand[i][1];
    int64_t tm = (M + bm - 1) / bm, tn = (N + bn - 1) / bn;
    double flops = (double)(tm * bm) * (tn * bn) / ALPHA;  // (x K, common)
    double staging = (double)tm * bm * tn + (double)tn * bn * tm;
    double cost = flops + staging;
    if (i == 0 || cost < best) {
      best = cost;
      *bm_out = bm;
      *bn_out = bn;
    }
  }
}

// Materialized colT column stride: small-K first layers (G==1, channel
// count not vector-aligned, Kcol < 64) pad to 64 ZERO-FILLED columns so
// the forward GEMM's K-span is one full BK tile -- the K-edge otherwise
// forces the guarded (scalarized) staging path for every tile (VGG
// conv1_1 fwd, K=27: 2.57 ms -> glds path). Callers allocate colT/wk
// zero-initialized when the returned stride != Kcol.
inline int ps_colT_ld(int G, int C, int kh, int kw, int vec) {
  int Kcol = kh * kw * C;  // G*kh*kw*(C/G)
  (void)vec;
  // G==1: round every colT row up to a BK (64) multiple with zero columns
  // -- the fwd GEMM's K becomes glds-clean (no k-tail) and the wgrad's N
  // becomes tr16-clean (no edge strips)
  if (G == 1) return (Kcol + 63) & ~63;
  return Kcol;
}

extern "C" {
// gemm.hip
void ps_gemm_f32(const GemmArgs* g, hipStream_t s);
void ps_gemm_bf16_f32out(const GemmArgs* g, hipStream_t s);
void ps_gemm_bf16(const GemmArgs* g, hipStream_t s);
int64_t ps_gemm_tn_tr_ws_elems(const GemmArgs* g);

// elementwise.hip
void ps_relu_fwd_f32(const float*, float*, int64_t, float, hipStream_t);
void ps_relu_bwd_f32(const float*, const float*, float*, int64_t, float, hipStream_t);
void ps_sigmoid_fwd_f32(const float*, float*, int64_t, hipStream_t);
void ps_sigmoid_bwd_f32(const float*, const float*, float*, int64_t, hipStream_t);
void ps_tanh_fwd_f32(const float*, float*, int64_t, hipStream_t);
void ps_tanh_bwd_f32(const float*, const float*, float*, int64_t, hipStream_t);
void ps_bnll_fwd_f32(const float*, float*, int64_t, hipStream_t);
void ps_bnll_bwd_f32(const float*, const float*, float*, int64_t, hipStream_t);
void ps_dropout_fwd_f32(const float*, float*, uint8_t*, int64_t, float,
                        uint64_t, uint64_t, hipStream_t);
void ps_dropout_bwd_f32(const float*, const uint8_t*, float*, int64_t, float,
                        hipStream_t);
void ps_colsum_f32(const float*, float*, int64_t, int, hipStream_t);
void ps_relu_fwd_bf16(const void*, void*, int64_t, float, hipStream_t);
void ps_relu_bwd_bf16(const void*, const void*, void*, int64_t, float, hipStream_t);
void ps_sigmoid_fwd_bf16(const void*, void*, int64_t, hipStream_t);
void ps_sigmoid_bwd_bf16(const void*, const void*, void*, int64_t, hipStream_t);
void ps_tanh_fwd_bf16(const void*, void*, int64_t, hipStream_t);
void ps_tanh_bwd_bf16(const void*, const void*, void*, int64_t, hipStream_t);
void ps_bnll_fwd_bf16(const void*, void*, int64_t, hipStream_t);
void ps_bnll_bwd_bf16(const void*, const void*, void*, int64_t, hipStream_t);
void ps_dropout_fwd_bf16(const void*, void*, uint8_t*, int64_t, float,
                         uint64_t, uint64_t, hipStream_t);
void ps_dropout_bwd_bf16(const void*, const uint8_t*, void*, int64_t, float,
                         hipStream_t);
void ps_colsum_bf16(const void*, float*, int64_t, int, hipStream_t);
void ps_colsum_mt(const void*, const void*, int, int, int, hipStream_t);
void ps_threshold_fwd_f32(const float*, float*, int64_t, float, hipStream_t);
void ps_threshold_fwd_bf16(const void*, void*, int64_t, float, hipStream_t);
void ps_eltwise_max_fwd_f32(const float*, const float*, float*, uint8_t*,
                            int64_t, int, hipStream_t);
void ps_eltwise_max_fwd_bf16(const void*, const void*, void*, uint8_t*,
                             int64_t, int, hipStream_t);
void ps_eltwise_max_bwd_f32(const float*, const uint8_t*, float*, int64_t,
                            int, hipStream_t);
void ps_eltwise_max_bwd_bf16(const void*, const uint8_t*, void*, int64_t,
                             int, hipStream_t);
void ps_contrastive_fwd_f32(const float*, const float*, float*, int64_t,
                            float, int, hipStream_t);

// pool.hip
void ps_maxpool_fwd_f32(const float*, float*, uint8_t*, const PoolGeom*, hipStream_t);
void ps_maxpool_bwd_f32(const float*, const uint8_t*, float*, const PoolGeom*, hipStream_t);
void ps_avepool_fwd_f32(const float*, float*, const PoolGeom*, hipStream_t);
void ps_avepool_bwd_f32(const float*, float*, const PoolGeom*, hipStream_t);
void ps_stochpool_fwd_train_f32(const float*, float*, uint8_t*, const PoolGeom*,
                                uint64_t, hipStream_t);
void ps_stochpool_fwd_test_f32(const float*, float*, const PoolGeom*, hipStream_t);
void ps_maxpool_fwd_bf16(const void*, void*, uint8_t*, const PoolGeom*, hipStream_t);
void ps_maxpool_bwd_bf16(const void*, const uint8_t*, void*, const PoolGeom*, hipStream_t);
void ps_avepool_fwd_bf16(const void*, void*, const PoolGeom*, hipStream_t);
void ps_avepool_bwd_bf16(const void*, void*, const PoolGeom*, hipStream_t);

// lrn.hip
void ps_lrn_fwd_f32(const float*, float*, float*, int64_t, int, int, float,
                    float, hipStream_t);
void ps_lrn_bwd_f32(const float*, const float*, const float*, const float*,
                    float*, float*, int64_t, int, int, float, float, hipStream_t);
void ps_lrn_fwd_bf16(const void*, void*, float*, int64_t, int, int, float,
                     float, hipStream_t);
void ps_lrn_bwd_bf16(const void*, const void*, const float*, const void*,
                     void*, float*, int64_t, int, int, float, float, hipStream_t);

// softmax.hip
void ps_softmax_rows_f32(const float*, float*, int64_t, int, hipStream_t);
void ps_softmax_bwd_rows_f32(const float*, const float*, float*, int64_t, int,
                             hipStream_t);
void ps_softmax_loss_fwd_f32(const float*, const float*, float*, float*,
                             int64_t, int, hipStream_t);
void ps_softmax_loss_bwd_f32(const float*, const float*, float*, int64_t, int,
                             float, hipStream_t);
void ps_softmax_rows_bf16(const void*, void*, int64_t, int, hipStream_t);
void ps_softmax_bwd_rows_bf16(const void*, const void*, void*, int64_t, int,
                              hipStream_t);
void ps_softmax_loss_fwd_bf16(const void*, const float*, void*, float*,
                              int64_t, int, hipStream_t);
void ps_softmax_loss_bwd_bf16(const void*, const float*, void*, int64_t, int,
                              float, hipStream_t);

// sgd.hip
void ps_sgd_update(float*, const float*, float*, int64_t, float, float, float,
                   hipStream_t);
void ps_nesterov_update(float*, const float*, float*, int64_t, float, float,
                        float, hipStream_t);
void ps_adagrad_update(float*, const float*, float*, int64_t, float, float,
                       float, hipStream_t);
void ps_f32_to_bf16(const float*, void*, int64_t, hipStream_t);
void ps_sgd_update_lrdev(float*, const float*, float*, int64_t, float, float,
                         float, const float*, hipStream_t);
void ps_u64_inc(void*, hipStream_t);
int ps_lrn_v8_ok(int C, int size);
void ps_lrn_fwd_v8_f32(const float*, float*, int64_t, int, int, float, float,
                       hipStream_t);
void ps_lrn_fwd_v8_bf16(const void*, void*, int64_t, int, int, float, float,
                        hipStream_t);
void ps_lrn_bwd_v8_f32(const float*, const float*, const float*, float*,
                       int64_t, int, int, float, float, hipStream_t);
void ps_lrn_bwd_v8_bf16(const void*, const void*, const void*, void*, int64_t,
                        int, int, float, float, hipStream_t);
// multi-tensor apply (descriptor/chunk tables built by bindings; layouts in
// sgd.hip: MTDesc {w,g,h,n,lr_mult,wd}, MTZeroDesc {p,n}, MTChunk {t,off})
int ps_mt_chunk_elts(void);
void ps_sgd_mt(const void* descs, const void* chunks, int nchunks, float lr,
               float mom, const float* lr_dev, hipStream_t);
void ps_zero_mt(const void* descs, const void* chunks, int nchunks,
                hipStream_t);
void ps_repack_mt(const void* descs, const void* chunks, int nchunks,
                  hipStream_t);
void ps_unpack_mt(const void* descs, const void* chunks, int nchunks,
                  hipStream_t);
void ps_dropout_fwd_f32_offdev(const float*, float*, uint8_t*, int64_t, float,
                               uint64_t, const void*, hipStream_t);
void ps_dropout_fwd_bf16_offdev(const void*, void*, uint8_t*, int64_t, float,
                                uint64_t, const void*, hipStream_t);

// im2col.hip
void ps_chan_concat4_f32(float*, void* const*, const int*, int, int, int64_t,
                         int, int, const void* const*, hipStream_t);
void ps_chan_concat4_bf16(void*, void* const*, const int*, int, int, int64_t,
                          int, int, const void* const*, hipStream_t);
void ps_chan_copy_f32(const float*, float*, int64_t, int, int, int, hipStream_t);
void ps_chan_copy_bf16(const void*, void*, int64_t, int, int, int, hipStream_t);
void ps_chan_slice_f32(const float*, float*, int64_t, int, int, int, hipStream_t);
void ps_chan_slice_bf16(const void*, void*, int64_t, int, int, int, hipStream_t);
void ps_im2col_nhwc_f32(const float*, float*, const ConvGeom*, int ldcol,
                        hipStream_t);
void ps_col2im_nhwc_f32(const float*, float*, const ConvGeom*, hipStream_t);
void ps_im2col_nhwc_bf16(const void*, void*, const ConvGeom*, int ldcol,
                         hipStream_t);
void ps_col2im_nhwc_bf16(const void*, void*, const ConvGeom*, hipStream_t);
void ps_weight_to_khwc_f32(const float*, float*, int, int, int, int, hipStream_t);
void ps_weight_to_khwc_f32_bf16(const float*, void*, int, int, int, int, hipStream_t);
void ps_zero_cols_f32(float*, int64_t rows, int ld, int c_lo, hipStream_t);
void ps_zero_cols_bf16(void*, int64_t rows, int ld, int c_lo, hipStream_t);
void ps_weight_to_dgrad_f32(const float*, float*, int, int, int, int, int,
                            hipStream_t);
void ps_weight_to_dgrad_f32_bf16(const float*, void*, int, int, int, int,
                                 int, hipStream_t);
void ps_weight_from_khwc_f32(const float*, float*, int, int, int, int,
                             int ld, float beta, hipStream_t);
void ps_weight_to_khwc_tr_f32(const float*, float*, int, int, int, int, int,
                              hipStream_t);
void ps_weight_to_khwc_tr_f32_bf16(const float*, void*, int, int, int, int,
                                   int, hipStream_t);
void ps_weight_to_khwc_both_f32(const float*, float*, float*, int, int, int,
                                int, int, int ldk, hipStream_t);
void ps_weight_to_khwc_both_f32_bf16(const float*, void*, void*, int, int,
                                     int, int, int, int ldk, hipStream_t);
}
