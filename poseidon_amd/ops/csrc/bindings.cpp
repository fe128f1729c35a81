// Torch bindings for the poseidon_amd CDNA4 kernels. All GPU activations
// are channels-last (NHWC physical); bindings normalize layouts and compose
// the conv pipeline (im2col -> MFMA GEMM -> bias / col2im / weight repack).
//
// Dtypes: activations are fp32 or bf16 (the MFMA fast path); parameter
// master copies are always fp32 -- conv/linear cast weights to bf16 shadows
// per call when the activations are bf16, and weight/bias GRADIENTS are
// produced in fp32 (bf16 GEMM with fp32 accumulate and fp32 output).
//
// These entry points are the ONLY GPU compute path -- ops/functional.py has
// no eager-torch fallback on CUDA tensors, so a passing GPU test means these
// kernels ran.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <algorithm>
#include <atomic>
#include <cmath>
#include <cstring>

#include "ps_api.h"

namespace {

using at::Tensor;

hipStream_t stream() { return c10::hip::getCurrentHIPStream().stream(); }

bool is_bf16(const Tensor& t) { return t.scalar_type() == at::kBFloat16; }

void check_float_like(const Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == at::kFloat || t.scalar_type() == at::kBFloat16,
              name, ": expected float32 or bfloat16");
  TORCH_CHECK(t.is_cuda(), name, ": expected device tensor");
}

Tensor cl4(const Tensor& t) {  // channels-last contiguous view of a 4D tensor
  TORCH_CHECK(t.dim() == 4, "expected 4D tensor");
  return t.contiguous(at::MemoryFormat::ChannelsLast);
}

// 2D [N*H*W, C] view of a channels-last 4D tensor (physical NHWC rows).
Tensor rows2d(const Tensor& t_cl) {
  int64_t NP = t_cl.size(0) * t_cl.size(2) * t_cl.size(3);
  return t_cl.permute({0, 2, 3, 1}).reshape({NP, t_cl.size(1)});
}

// fp32 master weight -> compute-dtype shadow (bf16 cast via our kernel)
Tensor weight_shadow(const Tensor& w_f32, bool bf16) {
  if (!bf16) return w_f32.contiguous();
  auto wc = w_f32.contiguous();
  Tensor out = at::empty(wc.sizes(), wc.options().dtype(at::kBFloat16));
  ps_f32_to_bf16(wc.data_ptr<float>(), out.data_ptr(), wc.numel(), stream());
  return out;
}

// ---------------------------------------------------------------------------
// GEMM plumbing
// ---------------------------------------------------------------------------

// Implicit-GEMM mode: gather im2col inside the GEMM staging instead of
// materializing the column matrix. Correct for every Cg%VEC==0 conv and
// covered by tests in both modes; measured slightly behind the
// materialized+vectorized-im2col pipeline on VGG/GoogLeNet shapes (gather
// address ALU in the staging inner loop), so default off -- toggle with
// set_implicit_gemm for experiments.
std::atomic<bool> g_implicit_gemm{false};
// Per-layer auto-policy: a conv whose materialized colT would exceed this
// goes implicit regardless of the global flag (VGG's 224/112-resolution
// 3x3 convs build 1-2 GB column matrices; gathering from x reads KHW x
// less HBM). Measured: global implicit loses ~4-8% on AlexNet/GoogLeNet
// (gather decode overhead on small colT), wins on the giant-colT layers.
std::atomic<int64_t> g_implicit_thresh{192LL << 20};

void set_implicit_gemm(bool on) { g_implicit_gemm.store(on); }
void set_implicit_threshold(int64_t bytes) { g_implicit_thresh.store(bytes); }

// 64B zero page for implicit-GEMM padding loads (per device, persistent)
const void* zero_page(const Tensor& like) {
  static Tensor z;
  if (!z.defined())
    z = at::zeros({64}, like.options().dtype(at::kByte));
  return z.data_ptr();
}

// in_bf16: A/B are bf16; out_f32: C is fp32 (else C matches input dtype)
void run_gemm(const Tensor& A, const Tensor& B, Tensor& C,
              const float* bias, int M, int N, int K,
              int64_t lda, int64_t ldb, int64_t ldc,
              int64_t a_off, int64_t b_off, int64_t c_off,
              bool a_klast, bool b_klast, float alpha, float beta,
              const GatherDesc* gather_a = nullptr,
              const GatherDesc* gather_b = nullptr, bool relu = false,
              bool c_prezeroed = false) {
  const bool in_bf16 = is_bf16(A);
  TORCH_CHECK(is_bf16(B) == in_bf16, "gemm: A/B dtype mismatch");
  const bool out_f32 = !is_bf16(C);
  GemmArgs g;
  g.A = in_bf16 ? (const void*)((const at::BFloat16*)A.data_ptr() + a_off)
                : (const void*)(A.data_ptr<float>() + a_off);
  g.B = in_bf16 ? (const void*)((const at::BFloat16*)B.data_ptr() + b_off)
                : (const void*)(B.data_ptr<float>() + b_off);
  g.C = out_f32 ? (void*)(C.data_ptr<float>() + c_off)
                : (void*)((at::BFloat16*)C.data_ptr() + c_off);
  g.bias = bias;
  g.M = M; g.N = N; g.K = K;
  g.lda = lda; g.ldb = ldb; g.ldc = ldc;
  g.strideA = 0; g.strideB = 0; g.strideC = 0;
  g.batch = 1;
  g.alpha = alpha; g.beta = beta;
  g.a_klast = a_klast; g.b_klast = b_klast;
  g.ws = nullptr;
  g.splitk = 1;
  g.gather_a = gather_a;
  g.gather_b = gather_b;
  g.relu = relu;
  g.c_prezeroed = c_prezeroed;
  // tr16 TN path with split-K: hand it a workspace so the splits store
  // plain partials + one reduce instead of memset + per-element atomicAdd
  // chains (VGG wgrad: ~2M contended atomics per GEMM removed)
  Tensor trws;
  // workspace split-K measured a net LOSS end-to-end vs the atomic form
  // (GoogLeNet -4%, VGG -2%: the reduce launch + ws traffic outweigh the
  // atomic contention); PS_TR_WS=1 re-enables it for experiments
  static const bool tr_ws_on = [] {
    const char* e = getenv("PS_TR_WS");
    return e != nullptr && strcmp(e, "1") == 0;
  }();
  if (tr_ws_on && in_bf16 && out_f32) {
    int64_t trw = ps_gemm_tn_tr_ws_elems(&g);
    if (trw > 0 && trw * 4 <= (256LL << 20)) {
      trws = at::empty({trw}, C.options().dtype(at::kFloat));
      g.ws = trws.data_ptr<float>();
      ps_gemm_bf16_f32out(&g, stream());
      return;
    }
  }
  // Split-K when the output tile grid cannot fill 256 CUs but K is deep
  // (conv wgrad: M=Cout, N=Kcol, K=N*OH*OW up to ~800k): target ~512
  // workgroups, cap the f32 workspace at 256 MB.
  int tbm, tbn;
  ps_pick_gemm_tile(M, N, &tbm, &tbn);
  int64_t tiles = (int64_t)((M + tbm - 1) / tbm) * ((N + tbn - 1) / tbn);
  Tensor ws;  // keep alive until launch returns
  // 2 blocks/CU resident -> ~512 workgroups fill the chip; split K until
  // the grid gets there (wgrad at 512x4608 is 144 tiles = 28% occupancy
  // without this)
  const bool atomic_ok = out_f32 && bias == nullptr && beta == 0.0f &&
                         !relu && ldc == N;
  if (atomic_ok && tiles < 384 && K >= 512) {
    // atomic split-K: no workspace, so split much deeper (chunk >= 256)
    // -- the un-split grid leaves most of the 256 CUs idle
    int sk = (int)std::min<int64_t>((512 + tiles - 1) / tiles,
                                    (K + 255) / 256);
    if (sk > 1) {
      g.splitk = sk;
      g.ws = nullptr;
      // callers owning a buffer that is zeroed once per iteration by the
      // net-level zero_mt launch skip this per-GEMM memset (GoogLeNet:
      // ~57 fillBuffer launches/step eliminated)
      if (!c_prezeroed)
        hipMemsetAsync(g.C, 0, (size_t)M * N * sizeof(float), stream());
    }
  } else if (tiles < 384 && K >= 4096) {
    int sk = (int)std::min<int64_t>((512 + tiles - 1) / tiles,
                                    (K + 2047) / 2048);
    int64_t ws_elems = (int64_t)sk * M * N;
    if (sk > 1 && ws_elems * 4 <= (256LL << 20)) {
      g.splitk = sk;
      ws = at::empty({ws_elems}, C.options().dtype(at::kFloat));
      g.ws = ws.data_ptr<float>();
    }
  }
  if (!in_bf16) {
    TORCH_CHECK(out_f32, "f32 gemm must have f32 out");
    ps_gemm_f32(&g, stream());
  } else if (out_f32) {
    ps_gemm_bf16_f32out(&g, stream());
  } else {
    ps_gemm_bf16(&g, stream());
  }
}

// Generic exposed GEMM (tests): C[M,N] = op(A)@op(B); op via *_klast flags.
Tensor gemm(const Tensor& A, const Tensor& B, int M, int N, int K,
            bool a_klast, bool b_klast) {
  check_float_like(A, "A");
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  Tensor C = at::empty({M, N}, A.options().dtype(at::kFloat));
  int64_t lda = a_klast ? K : M;
  int64_t ldb = b_klast ? K : N;
  run_gemm(Ac, Bc, C, nullptr, M, N, K, lda, ldb, N, 0, 0, 0,
           a_klast, b_klast, 1.0f, 0.0f);
  return C;
}

// ---------------------------------------------------------------------------
// InnerProduct
// ---------------------------------------------------------------------------

Tensor linear_forward(const Tensor& x, const Tensor& w,
                      const c10::optional<Tensor>& bias, bool fuse_relu,
                      const c10::optional<Tensor>& w_shadow) {
  check_float_like(x, "x");
  auto xc = x.contiguous();
  // per-step bf16 shadow from the net-level repack table (skips the
  // per-call f32->bf16 cast of the fc masters: VGG casts 138M params
  // twice per step otherwise)
  auto wc = (w_shadow.has_value() && is_bf16(x) &&
             w_shadow->numel() == w.numel())
                ? w_shadow->view(w.sizes())
                : weight_shadow(w, is_bf16(x));
  int M = xc.size(0), K = xc.size(1), N = wc.size(0);
  TORCH_CHECK(wc.size(1) == K, "linear: K mismatch");
  Tensor y = at::empty({M, N}, x.options());
  const float* bp = nullptr;
  Tensor bc;
  if (bias.has_value()) {
    bc = bias->contiguous();
    bp = bc.data_ptr<float>();
  }
  run_gemm(xc, wc, y, bp, M, N, K, K, K, N, 0, 0, 0, true, true, 1.0f, 0.0f,
           nullptr, nullptr, fuse_relu);
  return y;
}

std::vector<c10::optional<Tensor>> linear_backward(
    const Tensor& x, const Tensor& w, const Tensor& dy,
    bool need_dx, bool need_dw, bool has_bias,
    const c10::optional<Tensor>& w_shadow,
    const c10::optional<Tensor>& dw_acc) {
  check_float_like(dy, "dy");
  auto xc = x.contiguous();
  auto wc = (w_shadow.has_value() && is_bf16(dy) &&
             w_shadow->numel() == w.numel())
                ? w_shadow->view(w.sizes())
                : weight_shadow(w, is_bf16(dy));
  auto dyc = dy.contiguous();
  int M = xc.size(0), K = xc.size(1), N = wc.size(0);
  c10::optional<Tensor> dx, dw, db;
  if (need_dx) {
    Tensor t = at::empty({M, K}, x.options());
    // dx[M,K] = dy[M,N] @ W[N,K]: contraction N; A=dy K-last, B=W K-major
    run_gemm(dyc, wc, t, nullptr, M, K, N, N, K, K, 0, 0, 0, true, false,
             1.0f, 0.0f);
    dx = t;
  }
  if (need_dw) {
    if (dw_acc.has_value()) {
      // accumulate straight into the param diff (beta=1): saves the
      // separate diff.add_(dw) pass over the fc weights (VGG: a 138M
      // element read-modify-write per step)
      Tensor t = *dw_acc;
      TORCH_CHECK(t.scalar_type() == at::kFloat && t.is_contiguous());
      run_gemm(dyc, xc, t, nullptr, N, K, M, N, K, K, 0, 0, 0, false, false,
               1.0f, 1.0f);
    } else {
      Tensor t = at::empty({N, K}, x.options().dtype(at::kFloat));
      // dW[N,K] = dy^T[N,M] @ x[M,K]: contraction M; both K-major; fp32 out
      run_gemm(dyc, xc, t, nullptr, N, K, M, N, K, K, 0, 0, 0, false, false,
               1.0f, 0.0f);
      dw = t;
    }
  }
  if (has_bias) {
    Tensor t = at::zeros({N}, x.options().dtype(at::kFloat));
    if (is_bf16(dyc))
      ps_colsum_bf16(dyc.data_ptr(), t.data_ptr<float>(), M, N, stream());
    else
      ps_colsum_f32(dyc.data_ptr<float>(), t.data_ptr<float>(), M, N, stream());
    db = t;
  }
  return {dx, dw, db};
}

// SFB reconstruction: dW[N,K] = a[M,N]^T @ b[M,K] -> fp32
Tensor gemm_at_b(const Tensor& a, const Tensor& b) {
  check_float_like(a, "a");
  auto ac = a.contiguous();
  auto bc = b.contiguous();
  int M = ac.size(0), N = ac.size(1), K = bc.size(1);
  Tensor t = at::empty({N, K}, a.options().dtype(at::kFloat));
  run_gemm(ac, bc, t, nullptr, N, K, M, N, K, K, 0, 0, 0, false, false,
           1.0f, 0.0f);
  return t;
}

// ---------------------------------------------------------------------------
// Convolution (im2col + grouped MFMA GEMM, NHWC)
// ---------------------------------------------------------------------------

ConvGeom conv_geom(const Tensor& x_cl, int kh, int kw, int sh, int sw,
                   int ph, int pw, int G) {
  ConvGeom g;
  g.N = x_cl.size(0); g.C = x_cl.size(1);
  g.H = x_cl.size(2); g.W = x_cl.size(3);
  g.kh = kh; g.kw = kw; g.sh = sh; g.sw = sw; g.ph = ph; g.pw = pw; g.G = G;
  g.Ho = (g.H + 2 * ph - kh) / sh + 1;
  g.Wo = (g.W + 2 * pw - kw) / sw + 1;
  return g;
}

// weights [Co,Cig,kh,kw] fp32 -> khwc [Co, kh*kw*Cig] in compute dtype
Tensor weight_khwc_tr(const Tensor& w, int G, bool bf16) {
  auto wc = w.contiguous();
  int Co = wc.size(0), Cig = wc.size(1), kh = wc.size(2), kw = wc.size(3);
  int Kg = kh * kw * Cig;
  Tensor wkT = at::empty({(int64_t)G * Kg, (int64_t)(Co / G)},
                         w.options().dtype(bf16 ? at::kBFloat16 : at::kFloat));
  if (bf16)
    ps_weight_to_khwc_tr_f32_bf16(wc.data_ptr<float>(), wkT.data_ptr(), Co,
                                  Cig, kh, kw, G, stream());
  else
    ps_weight_to_khwc_tr_f32(wc.data_ptr<float>(), wkT.data_ptr<float>(), Co,
                             Cig, kh, kw, G, stream());
  return wkT;
}

std::vector<Tensor> conv2d_forward_ex(const Tensor& x, const Tensor& w,
                                      const c10::optional<Tensor>& bias,
                                      int sh, int sw, int ph, int pw, int G,
                                      bool fuse_relu,
                                      const c10::optional<Tensor>& wk_cached,
                                      const c10::optional<Tensor>& wkT_cached) {
  check_float_like(x, "x");
  const bool bf16 = is_bf16(x);
  auto x_cl = cl4(x);
  int Co = w.size(0), Cig = w.size(1), kh = w.size(2), kw = w.size(3);
  ConvGeom g = conv_geom(x_cl, kh, kw, sh, sw, ph, pw, G);
  TORCH_CHECK(Cig * G == g.C, "conv channel/group mismatch");
  int Cg = g.C / G;
  int Kg = kh * kw * Cg;
  int Kcol = G * Kg;
  int64_t NP = (int64_t)g.N * g.Ho * g.Wo;
  int Cog = Co / G;
  const int VEC = bf16 ? 8 : 4;
  // small-K pad (ps_api.h ps_colT_ld): zero columns up to one full BK so
  // the fwd GEMM is a single interior k-tile (glds path)
  // 1x1 convs alias x rows as colT (physical stride = C): never pad those
  const bool pre_1x1 = (kh == 1 && kw == 1 && sh == 1 && sw == 1 &&
                        ph == 0 && pw == 0);
  const int ldc_col = pre_1x1 ? Kcol : ps_colT_ld(G, g.C, kh, kw, VEC);
  const bool kpad = ldc_col != Kcol;

  // fused repack: one kernel writes both the khwc fwd operand and the
  // per-group transpose the dgrad GEMM wants (cached by the layer). When
  // the layer provides PRE-REPACKED buffers (the net-level multi-tensor
  // repack, repack_mt_run), skip the per-layer kernel entirely.
  Tensor wk, wkT;
  if (wk_cached.has_value() && wkT_cached.has_value() &&
      wk_cached->size(1) == ldc_col / G &&
      wk_cached->scalar_type() == (bf16 ? at::kBFloat16 : at::kFloat)) {
    wk = *wk_cached;
    wkT = *wkT_cached;
  } else {
    auto wc = w.contiguous();
    auto wkopts = w.options().dtype(bf16 ? at::kBFloat16 : at::kFloat);
    wk = kpad ? at::zeros({Co, (int64_t)ldc_col}, wkopts)
              : at::empty({Co, (int64_t)Kg}, wkopts);
    wkT = at::empty({(int64_t)G * kh * kw * Cig, (int64_t)(Co / G)}, wkopts);
    if (bf16)
      ps_weight_to_khwc_both_f32_bf16(wc.data_ptr<float>(), wk.data_ptr(),
                                      wkT.data_ptr(), Co, Cig, kh, kw, G,
                                      ldc_col / G, stream());
    else
      ps_weight_to_khwc_both_f32(wc.data_ptr<float>(), wk.data_ptr<float>(),
                                 wkT.data_ptr<float>(), Co, Cig, kh, kw, G,
                                 ldc_col / G, stream());
  }

  bool is_1x1 = (kh == 1 && kw == 1 && sh == 1 && sw == 1 && ph == 0 && pw == 0);
  // implicit GEMM: gather im2col rows inside the GEMM staging whenever the
  // group channel count keeps 16B runs contiguous -- no column matrix at
  // all (the wgrad GEMM gathers too); conv1-style small-C layers still
  // materialize (and 1x1 convs alias x directly)
  int64_t colT_bytes = NP * Kcol * (bf16 ? 2 : 4);
  // big-colT layers go implicit: with the gather-staged tr16 wgrad
  // (gemm.hip stage_kmaj_tr<GATHER>) the implicit path now beats
  // materialization wherever the column matrix is large (VGG/AlexNet
  // early-mid layers: no im2col write, wgrad gathers x directly);
  // small-colT inception-style layers still measure faster materialized
  bool implicit = (g_implicit_gemm.load() ||
                   colT_bytes > g_implicit_thresh.load())
                  && !is_1x1 && (Cg % VEC == 0);
  Tensor colT;
  if (is_1x1) {
    colT = rows2d(x_cl);  // alias: x rows ARE the col rows
  } else if (!implicit) {
    colT = at::empty({NP, (int64_t)ldc_col}, x.options());
    if (bf16)
      ps_im2col_nhwc_bf16(x_cl.data_ptr(), colT.data_ptr(), &g, ldc_col,
                          stream());
    else
      ps_im2col_nhwc_f32(x_cl.data_ptr<float>(), colT.data_ptr<float>(), &g,
                         ldc_col, stream());
    // pad columns (ldc_col > Kcol) must be zero for the GEMMs; only the
    // rowstage kernel writes them itself -- zero unconditionally (cheap,
    // <= 63 columns)
    if (kpad) {
      if (bf16)
        ps_zero_cols_bf16(colT.data_ptr(), NP, ldc_col, Kcol, stream());
      else
        ps_zero_cols_f32(colT.data_ptr<float>(), NP, ldc_col, Kcol,
                         stream());
    }
  }

  Tensor y = at::empty({g.N, Co, g.Ho, g.Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const float* bp = nullptr;
  Tensor bc;
  if (bias.has_value()) {
    bc = bias->contiguous();
    bp = bc.data_ptr<float>();
  }
  for (int grp = 0; grp < G; ++grp) {
    if (implicit) {
      GatherDesc ga;
      ga.x = x_cl.data_ptr();
      ga.zero = zero_page(x_cl);
      ga.C = g.C; ga.H = g.H; ga.W = g.W; ga.Ho = g.Ho; ga.Wo = g.Wo;
      ga.kh = kh; ga.kw = kw; ga.sh = sh; ga.sw = sw; ga.ph = ph; ga.pw = pw;
      ga.Cg = Cg; ga.c0 = grp * Cg;
      ga.kg_max = Kg;
      ps_fill_gather_inv(&ga);
      // wk rows are PADDED to ldc_col/G; run the GEMM over the padded K
      // (gather k >= kg_max reads the zero page, wk pad cols are zero)
      const int Kgi = ldc_col / G;
      run_gemm(x_cl, wk, y, bp ? bp + grp * Cog : nullptr,
               (int)NP, Cog, Kgi,
               /*lda=*/Kgi, /*ldb=*/Kgi, /*ldc=*/Co,
               /*a_off=*/0, /*b_off=*/(int64_t)grp * Cog * Kgi,
               /*c_off=*/(int64_t)grp * Cog,
               true, true, 1.0f, 0.0f, &ga, nullptr, fuse_relu);
    } else {
      const int Kgl = ldc_col / G;  // per-group K incl. pad
      run_gemm(colT, wk, y, bp ? bp + grp * Cog : nullptr,
               (int)NP, Cog, Kgl,
               /*lda=*/ldc_col, /*ldb=*/Kgl, /*ldc=*/Co,
               /*a_off=*/(int64_t)grp * Kgl,
               /*b_off=*/(int64_t)grp * Cog * Kgl,
               /*c_off=*/(int64_t)grp * Cog,
               true, true, 1.0f, 0.0f, nullptr, nullptr, fuse_relu);
    }
  }
  if (!colT.defined())
    colT = at::empty({0}, x.options());  // implicit mode: wgrad gathers too
  return {y, colT, wkT};
}

Tensor conv2d_backward_input(const Tensor& w, const Tensor& dy,
                             std::vector<int64_t> x_shape, int sh, int sw,
                             int ph, int pw, int G,
                             const c10::optional<Tensor>& wkT_cache,
                             const c10::optional<Tensor>& dx_out) {
  check_float_like(dy, "dy");
  const bool bf16 = is_bf16(dy);
  auto dy_cl = cl4(dy);
  int Co = w.size(0), kh = w.size(2), kw = w.size(3);
  // persistent dx (layer cache): keeps the consumer's dy identity stable
  // across iterations so net-level multi-tensor batching (bias colsum)
  // can key on it
  Tensor dx = (dx_out.has_value() && dx_out->sizes() ==
               at::IntArrayRef(x_shape) && dx_out->scalar_type() ==
               dy.scalar_type())
      ? *dx_out
      : at::empty({x_shape[0], x_shape[1], x_shape[2], x_shape[3]},
                  dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  ConvGeom g;
  g.N = x_shape[0]; g.C = x_shape[1]; g.H = x_shape[2]; g.W = x_shape[3];
  g.kh = kh; g.kw = kw; g.sh = sh; g.sw = sw; g.ph = ph; g.pw = pw; g.G = G;
  g.Ho = dy_cl.size(2); g.Wo = dy_cl.size(3);
  int Cg = g.C / G, Cog = Co / G;
  int Kg = kh * kw * Cg;
  int Kcol = G * Kg;
  int64_t NP = (int64_t)g.N * g.Ho * g.Wo;

  // stride-1 dgrad runs as a FORWARD conv of dy with rotated weights
  // (dx = conv(dy, rot180 W, pad = K-1-p)) through the implicit-gather
  // GEMM: no dcolT round-trip (up to 2 GB for VGG conv1_2) and no col2im.
  const int VEC = bf16 ? 8 : 4;
  bool is_1x1e = (kh == 1 && kw == 1 && sh == 1 && sw == 1 && ph == 0 &&
                  pw == 0);
  // worth it only when the dcolT it eliminates is substantial (GoogLeNet's
  // small inception dgrads measured faster on the materialized NT+col2im)
  int64_t dcolT_bytes = NP * Kcol * (bf16 ? 2 : 4);
  if (sh == 1 && sw == 1 && !is_1x1e && (Cog % VEC) == 0 && ph <= kh - 1 &&
      pw <= kw - 1 && dcolT_bytes > (128LL << 20)) {
    const int K2 = kh * kw * Cog;
    auto wc = w.contiguous();
    Tensor wr = at::empty({(int64_t)G * Cg, (int64_t)K2}, dy.options());
    if (bf16)
      ps_weight_to_dgrad_f32_bf16(wc.data_ptr<float>(), wr.data_ptr(), Co,
                                  Cg, kh, kw, G, stream());
    else
      ps_weight_to_dgrad_f32(wc.data_ptr<float>(), wr.data_ptr<float>(), Co,
                             Cg, kh, kw, G, stream());
    int64_t NP2 = (int64_t)g.N * g.H * g.W;
    for (int grp = 0; grp < G; ++grp) {
      GatherDesc ga;
      ga.x = dy_cl.data_ptr();
      ga.zero = zero_page(dy_cl);
      ga.C = Co; ga.H = g.Ho; ga.W = g.Wo;
      ga.Ho = g.H; ga.Wo = g.W;
      ga.kh = kh; ga.kw = kw; ga.sh = 1; ga.sw = 1;
      ga.ph = kh - 1 - ph; ga.pw = kw - 1 - pw;
      ga.Cg = Cog; ga.c0 = grp * Cog;
      ga.kg_max = K2;
      ps_fill_gather_inv(&ga);
      run_gemm(dy_cl, wr, dx, nullptr,
               (int)NP2, Cg, K2,
               /*lda=*/K2, /*ldb=*/K2, /*ldc=*/g.C,
               /*a_off=*/0, /*b_off=*/(int64_t)grp * Cg * K2,
               /*c_off=*/(int64_t)grp * Cg,
               true, true, 1.0f, 0.0f, &ga);
    }
    return dx;
  }

  // transposed khwc repack [G][Kg][Cog]: the dgrad GEMM becomes pure NT
  // (both operands K-last) instead of a K-major-staged NN
  Tensor wkT = wkT_cache.has_value() && wkT_cache->scalar_type() ==
                       (bf16 ? at::kBFloat16 : at::kFloat)
                   ? *wkT_cache
                   : weight_khwc_tr(w, G, bf16);

  bool is_1x1 = (kh == 1 && kw == 1 && sh == 1 && sw == 1 && ph == 0 && pw == 0);
  Tensor dcolT = is_1x1 ? rows2d(dx)
                        : at::empty({NP, (int64_t)Kcol}, dy.options());
  Tensor dy2 = rows2d(dy_cl);
  for (int grp = 0; grp < G; ++grp) {
    // dcolT_g[NP, Kg] = dy_g[NP, Cog] @ wkT_g[Kg, Cog]^T: NT, contraction Cog
    run_gemm(dy2, wkT, dcolT, nullptr,
             (int)NP, Kg, Cog,
             /*lda=*/Co, /*ldb=*/Cog, /*ldc=*/Kcol,
             /*a_off=*/(int64_t)grp * Cog, /*b_off=*/(int64_t)grp * Kg * Cog,
             /*c_off=*/(int64_t)grp * Kg,
             true, true, 1.0f, 0.0f);
  }
  if (!is_1x1) {
    if (bf16)
      ps_col2im_nhwc_bf16(dcolT.data_ptr(), dx.data_ptr(), &g, stream());
    else
      ps_col2im_nhwc_f32(dcolT.data_ptr<float>(), dx.data_ptr<float>(), &g,
                         stream());
  }
  return dx;
}

// dW accumulated into dw_out (fp32 NCHW [Co,Cg,kh,kw]); db into db_out.
// Returns the khwc scratch dwk it used: callers that keep it alive, hand it
// back as dwk_buf AND guarantee it is zeroed each iteration (net-level
// zero_mt) save both the allocation and the atomic-split-K memset.
Tensor conv2d_backward_weight_acc(const Tensor& x, const Tensor& colT,
                                  const Tensor& dy, Tensor dw_out,
                                  c10::optional<Tensor> db_out,
                                  int sh, int sw, int ph, int pw, int G,
                                  const c10::optional<Tensor>& dwk_buf,
                                  bool skip_unpack, bool skip_db) {
  check_float_like(dy, "dy");
  auto dy_cl = cl4(dy);
  auto x_cl = cl4(x);
  int Co = dw_out.size(0), Cig = dw_out.size(1);
  int kh = dw_out.size(2), kw = dw_out.size(3);
  int Cog = Co / G;
  int Kg = kh * kw * Cig;
  int64_t NP = (int64_t)dy_cl.size(0) * dy_cl.size(2) * dy_cl.size(3);
  const bool implicit = colT.numel() == 0 && colT.dim() == 1;
  int64_t Kcol = implicit ? 0 : colT.size(1);
  // colT may carry zero pad columns (ps_colT_ld): run the GEMM over the
  // padded width too (zero B columns -> zero dwk columns, skipped below).
  // Implicit wgrads pad the same way via the gather's kg_max bound, which
  // makes them tr16-eligible (N % 64 == 0, no edge strips) -- grouped
  // convs included (per-group dwk slices stay contiguous at c_off =
  // grp*Cog*Kgw; AlexNet conv2 Kg=1200 was stuck on the register-staged
  // gather TN path at 403 us/group without this).
  const int Kgw = implicit ? (Kg + 63) & ~63 : (int)(Kcol / G);

  // fp32 gradient accumulation regardless of activation dtype
  const bool prez = dwk_buf.has_value() && dwk_buf->numel() == (int64_t)Co * Kgw;
  Tensor dwk = prez ? *dwk_buf
             : at::empty({Co, (int64_t)Kgw}, dy.options().dtype(at::kFloat));
  Tensor dy2 = rows2d(dy_cl);
  for (int grp = 0; grp < G; ++grp) {
    // dwk_g[Cog, Kg] = dy_g^T[Cog, NP] @ colT_g[NP, Kg]: contraction NP
    if (implicit) {
      GatherDesc gb;
      gb.x = x_cl.data_ptr();
      gb.zero = zero_page(x_cl);
      gb.C = x_cl.size(1); gb.H = x_cl.size(2); gb.W = x_cl.size(3);
      gb.Ho = dy_cl.size(2); gb.Wo = dy_cl.size(3);
      gb.kh = kh; gb.kw = kw; gb.sh = sh; gb.sw = sw; gb.ph = ph; gb.pw = pw;
      gb.Cg = Cig; gb.c0 = grp * Cig;
      gb.kg_max = Kg;
      ps_fill_gather_inv(&gb);
      run_gemm(dy2, x_cl, dwk, nullptr,
               Cog, Kgw, (int)NP,
               /*lda=*/Co, /*ldb=*/Kgw, /*ldc=*/Kgw,
               /*a_off=*/(int64_t)grp * Cog, /*b_off=*/0,
               /*c_off=*/(int64_t)grp * Cog * Kgw,
               false, false, 1.0f, 0.0f, nullptr, &gb, false, prez);
    } else {
      run_gemm(dy2, colT, dwk, nullptr,
               Cog, Kgw, (int)NP,
               /*lda=*/Co, /*ldb=*/Kcol, /*ldc=*/Kgw,
               /*a_off=*/(int64_t)grp * Cog, /*b_off=*/(int64_t)grp * Kgw,
               /*c_off=*/(int64_t)grp * Cog * Kgw,
               false, false, 1.0f, 0.0f, nullptr, nullptr, false, prez);
    }
  }
  if (!skip_unpack)
    ps_weight_from_khwc_f32(dwk.data_ptr<float>(), dw_out.data_ptr<float>(),
                            Co, Cig, kh, kw, /*ld=*/Kgw, /*beta=*/1.0f,
                            stream());
  if (db_out.has_value() && !skip_db) {
    if (is_bf16(dy_cl))
      ps_colsum_bf16(dy_cl.data_ptr(), db_out->data_ptr<float>(), NP, Co,
                     stream());
    else
      ps_colsum_f32(dy_cl.data_ptr<float>(), db_out->data_ptr<float>(), NP,
                    Co, stream());
  }
  return dwk;
}

// ---------------------------------------------------------------------------
// NHWC channel concat / slice (CONCAT + SLICE layers, GoogLeNet inception)
// ---------------------------------------------------------------------------

Tensor concat_channels(std::vector<Tensor> inputs) {
  TORCH_CHECK(!inputs.empty());
  auto x0 = cl4(inputs[0]);
  int64_t N = x0.size(0), H = x0.size(2), W = x0.size(3);
  int64_t rows = N * H * W;
  int C_out = 0;
  for (auto& t : inputs) C_out += t.size(1);
  Tensor y = at::empty({N, (int64_t)C_out, H, W},
                       x0.options().memory_format(at::MemoryFormat::ChannelsLast));
  const bool bf16 = is_bf16(x0);
  const int V = bf16 ? 8 : 4;
  bool aligned = true;
  for (auto& t : inputs) aligned = aligned && (t.size(1) % V == 0);
  std::vector<Tensor> cls;
  cls.reserve(inputs.size());
  for (auto& t : inputs) cls.push_back(cl4(t));
  if (aligned) {
    // up to 4 branches per launch (GoogLeNet inception: one launch/join)
    int off = 0;
    for (size_t i0 = 0; i0 < cls.size(); i0 += 4) {
      void* ptrs[4];
      int c_end[4];
      int n = (int)std::min<size_t>(4, cls.size() - i0);
      int c_begin = off;
      for (int j = 0; j < n; ++j) {
        ptrs[j] = cls[i0 + j].data_ptr();
        off += (int)cls[i0 + j].size(1);
        c_end[j] = off;
      }
      if (bf16)
        ps_chan_concat4_bf16(y.data_ptr(), ptrs, c_end, n, c_begin, rows,
                             C_out, 0, nullptr, stream());
      else
        ps_chan_concat4_f32(y.data_ptr<float>(), ptrs, c_end, n, c_begin,
                            rows, C_out, 0, nullptr, stream());
    }
    return y;
  }
  int off = 0;
  for (auto& tc : cls) {
    int Ci = tc.size(1);
    if (is_bf16(tc))
      ps_chan_copy_bf16(tc.data_ptr(), y.data_ptr(), rows, Ci, C_out, off,
                        stream());
    else
      ps_chan_copy_f32(tc.data_ptr<float>(), y.data_ptr<float>(), rows, Ci,
                       C_out, off, stream());
    off += Ci;
  }
  return y;
}

// split the wide NHWC tensor into per-range narrow tensors (concat backward
// / slice forward), up to 4 ranges per launch
std::vector<Tensor> split_channels(const Tensor& x,
                                   std::vector<int64_t> sizes,
                                   const c10::optional<std::vector<Tensor>>&
                                       outs_cache,
                                   const c10::optional<std::vector<Tensor>>&
                                       relu_masks) {
  auto xc = cl4(x);
  int64_t N = xc.size(0), H = xc.size(2), W = xc.size(3);
  int64_t rows = N * H * W;
  int C_in = xc.size(1);
  const bool bf16 = is_bf16(xc);
  const int V = bf16 ? 8 : 4;
  std::vector<Tensor> outs;
  outs.reserve(sizes.size());
  const bool reuse = outs_cache.has_value() &&
      outs_cache->size() == sizes.size() &&
      std::all_of(outs_cache->begin(), outs_cache->end(),
                  [&](const Tensor& t) {
                    return t.scalar_type() == x.scalar_type();
                  });
  for (size_t j = 0; j < sizes.size(); ++j) {
    int64_t c = sizes[j];
    if (reuse && (*outs_cache)[j].sizes() ==
        at::IntArrayRef({N, c, H, W})) {
      outs.push_back((*outs_cache)[j]);
      continue;
    }
    outs.push_back(at::empty({N, c, H, W},
        x.options().memory_format(at::MemoryFormat::ChannelsLast)));
  }
  bool aligned = true;
  for (int64_t c : sizes) aligned = aligned && (c % V == 0);
  if (aligned) {
    int off = 0;
    for (size_t i0 = 0; i0 < outs.size(); i0 += 4) {
      void* ptrs[4];
      const void* masks[4];
      bool any_mask = false;
      int c_end[4];
      int n = (int)std::min<size_t>(4, outs.size() - i0);
      int c_begin = off;
      for (int j = 0; j < n; ++j) {
        ptrs[j] = outs[i0 + j].data_ptr();
        masks[j] = nullptr;
        if (relu_masks.has_value() && i0 + j < relu_masks->size() &&
            (*relu_masks)[i0 + j].defined()) {
          const Tensor& mk = (*relu_masks)[i0 + j];
          TORCH_CHECK(mk.scalar_type() == x.scalar_type() &&
                      mk.numel() == outs[i0 + j].numel(),
                      "split mask dtype/shape mismatch");
          masks[j] = mk.data_ptr();
          any_mask = true;
        }
        off += (int)sizes[i0 + j];
        c_end[j] = off;
      }
      if (bf16)
        ps_chan_concat4_bf16(xc.data_ptr(), ptrs, c_end, n, c_begin, rows,
                             C_in, 1, any_mask ? masks : nullptr, stream());
      else
        ps_chan_concat4_f32(xc.data_ptr<float>(), ptrs, c_end, n, c_begin,
                            rows, C_in, 1, any_mask ? masks : nullptr,
                            stream());
    }
  } else {
    int off = 0;
    for (size_t j = 0; j < outs.size(); ++j) {
      int Ci = (int)sizes[j];
      if (bf16)
        ps_chan_slice_bf16(xc.data_ptr(), outs[j].data_ptr(), rows, C_in, Ci,
                           off, stream());
      else
        ps_chan_slice_f32(xc.data_ptr<float>(), outs[j].data_ptr<float>(),
                          rows, C_in, Ci, off, stream());
      off += Ci;
    }
  }
  return outs;
}

Tensor slice_channels(const Tensor& x, int64_t c_off, int64_t c_len) {
  auto xc = cl4(x);
  int64_t N = xc.size(0), H = xc.size(2), W = xc.size(3);
  int64_t rows = N * H * W;
  Tensor y = at::empty({N, c_len, H, W},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(xc))
    ps_chan_slice_bf16(xc.data_ptr(), y.data_ptr(), rows, xc.size(1),
                       (int)c_len, (int)c_off, stream());
  else
    ps_chan_slice_f32(xc.data_ptr<float>(), y.data_ptr<float>(), rows,
                      xc.size(1), (int)c_len, (int)c_off, stream());
  return y;
}

// ---------------------------------------------------------------------------
// Pooling
// ---------------------------------------------------------------------------

int pool_out(int h, int k, int p, int s) {
  int out = (int)std::ceil((double)(h + 2 * p - k) / s) + 1;
  if (p > 0 && (out - 1) * s >= h + p) --out;
  return out;
}

PoolGeom pool_geom(const Tensor& x_cl, int kh, int kw, int sh, int sw,
                   int ph, int pw) {
  PoolGeom g;
  g.N = x_cl.size(0); g.C = x_cl.size(1); g.H = x_cl.size(2); g.W = x_cl.size(3);
  g.kh = kh; g.kw = kw; g.sh = sh; g.sw = sw; g.ph = ph; g.pw = pw;
  g.Ho = pool_out(g.H, kh, ph, sh);
  g.Wo = pool_out(g.W, kw, pw, sw);
  return g;
}

std::vector<Tensor> pool_max_forward(const Tensor& x, int kh, int kw, int sh,
                                     int sw, int ph, int pw) {
  check_float_like(x, "x");
  auto x_cl = cl4(x);
  PoolGeom g = pool_geom(x_cl, kh, kw, sh, sw, ph, pw);
  auto opts_cl = x.options().memory_format(at::MemoryFormat::ChannelsLast);
  Tensor y = at::empty({g.N, g.C, g.Ho, g.Wo}, opts_cl);
  Tensor mask = at::empty({g.N, g.C, g.Ho, g.Wo},
                          x.options().dtype(at::kByte)
                              .memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(x))
    ps_maxpool_fwd_bf16(x_cl.data_ptr(), y.data_ptr(),
                        mask.data_ptr<uint8_t>(), &g, stream());
  else
    ps_maxpool_fwd_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(),
                       mask.data_ptr<uint8_t>(), &g, stream());
  return {y, mask};
}

Tensor pool_max_backward(const Tensor& dy, const Tensor& mask,
                         std::vector<int64_t> x_shape, int kh, int kw, int sh,
                         int sw, int ph, int pw,
                         const c10::optional<Tensor>& dx_out) {
  auto dy_cl = cl4(dy);
  auto mask_cl = cl4(mask);
  PoolGeom g;
  g.N = x_shape[0]; g.C = x_shape[1]; g.H = x_shape[2]; g.W = x_shape[3];
  g.kh = kh; g.kw = kw; g.sh = sh; g.sw = sw; g.ph = ph; g.pw = pw;
  g.Ho = dy_cl.size(2); g.Wo = dy_cl.size(3);
  Tensor dx = (dx_out.has_value() &&
               dx_out->sizes() == at::IntArrayRef(x_shape) &&
               dx_out->scalar_type() == dy.scalar_type())
      ? *dx_out
      : at::empty({g.N, g.C, g.H, g.W},
                  dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(dy))
    ps_maxpool_bwd_bf16(dy_cl.data_ptr(), mask_cl.data_ptr<uint8_t>(),
                        dx.data_ptr(), &g, stream());
  else
    ps_maxpool_bwd_f32(dy_cl.data_ptr<float>(), mask_cl.data_ptr<uint8_t>(),
                       dx.data_ptr<float>(), &g, stream());
  return dx;
}

Tensor pool_ave_forward(const Tensor& x, int kh, int kw, int sh, int sw,
                        int ph, int pw) {
  check_float_like(x, "x");
  auto x_cl = cl4(x);
  PoolGeom g = pool_geom(x_cl, kh, kw, sh, sw, ph, pw);
  Tensor y = at::empty({g.N, g.C, g.Ho, g.Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(x))
    ps_avepool_fwd_bf16(x_cl.data_ptr(), y.data_ptr(), &g, stream());
  else
    ps_avepool_fwd_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(), &g, stream());
  return y;
}

Tensor pool_ave_backward(const Tensor& dy, std::vector<int64_t> x_shape,
                         int kh, int kw, int sh, int sw, int ph, int pw) {
  auto dy_cl = cl4(dy);
  PoolGeom g;
  g.N = x_shape[0]; g.C = x_shape[1]; g.H = x_shape[2]; g.W = x_shape[3];
  g.kh = kh; g.kw = kw; g.sh = sh; g.sw = sw; g.ph = ph; g.pw = pw;
  g.Ho = dy_cl.size(2); g.Wo = dy_cl.size(3);
  Tensor dx = at::empty({g.N, g.C, g.H, g.W},
                        dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(dy))
    ps_avepool_bwd_bf16(dy_cl.data_ptr(), dx.data_ptr(), &g, stream());
  else
    ps_avepool_bwd_f32(dy_cl.data_ptr<float>(), dx.data_ptr<float>(), &g,
                       stream());
  return dx;
}

std::vector<Tensor> pool_stoch_forward_train(const Tensor& x, int kh, int kw,
                                             int sh, int sw, int ph, int pw,
                                             int64_t seed) {
  check_float_like(x, "x");
  auto x_cl = cl4(x).to(at::kFloat);  // stochastic pooling runs fp32
  PoolGeom g = pool_geom(x_cl, kh, kw, sh, sw, ph, pw);
  Tensor y = at::empty({g.N, g.C, g.Ho, g.Wo},
                       x_cl.options().memory_format(at::MemoryFormat::ChannelsLast));
  Tensor mask = at::empty({g.N, g.C, g.Ho, g.Wo},
                          x_cl.options().dtype(at::kByte)
                              .memory_format(at::MemoryFormat::ChannelsLast));
  ps_stochpool_fwd_train_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(),
                             mask.data_ptr<uint8_t>(), &g, (uint64_t)seed,
                             stream());
  return {y.to(x.scalar_type()), mask};
}

Tensor pool_stoch_forward_test(const Tensor& x, int kh, int kw, int sh,
                               int sw, int ph, int pw) {
  check_float_like(x, "x");
  auto x_cl = cl4(x).to(at::kFloat);
  PoolGeom g = pool_geom(x_cl, kh, kw, sh, sw, ph, pw);
  Tensor y = at::empty({g.N, g.C, g.Ho, g.Wo},
                       x_cl.options().memory_format(at::MemoryFormat::ChannelsLast));
  ps_stochpool_fwd_test_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(), &g,
                            stream());
  return y.to(x.scalar_type());
}

// ---------------------------------------------------------------------------
// LRN (scale is fp32 in both paths)
// ---------------------------------------------------------------------------

std::vector<Tensor> lrn_forward(const Tensor& x, int size, double alpha,
                                double beta) {
  check_float_like(x, "x");
  auto x_cl = cl4(x);
  int64_t rows = (int64_t)x_cl.size(0) * x_cl.size(2) * x_cl.size(3);
  int C = x_cl.size(1);
  auto opts_cl = x.options().memory_format(at::MemoryFormat::ChannelsLast);
  Tensor y = at::empty_like(x_cl, opts_cl);
  if (ps_lrn_v8_ok(C, size)) {
    // halo-register path: no scale tensor at all (backward recomputes it)
    if (is_bf16(x))
      ps_lrn_fwd_v8_bf16(x_cl.data_ptr(), y.data_ptr(), rows, C, size,
                         (float)alpha, (float)beta, stream());
    else
      ps_lrn_fwd_v8_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(), rows, C,
                        size, (float)alpha, (float)beta, stream());
    return {y, at::empty({0}, x.options().dtype(at::kFloat))};
  }
  Tensor scale = at::empty_like(x_cl,
      x.options().dtype(at::kFloat).memory_format(at::MemoryFormat::ChannelsLast));
  if (is_bf16(x))
    ps_lrn_fwd_bf16(x_cl.data_ptr(), y.data_ptr(), scale.data_ptr<float>(),
                    rows, C, size, (float)alpha, (float)beta, stream());
  else
    ps_lrn_fwd_f32(x_cl.data_ptr<float>(), y.data_ptr<float>(),
                   scale.data_ptr<float>(), rows, C, size, (float)alpha,
                   (float)beta, stream());
  return {y, scale};
}

Tensor lrn_backward(const Tensor& x, const Tensor& y, const Tensor& scale,
                    const Tensor& dy, int size, double alpha, double beta) {
  auto x_cl = cl4(x);
  auto y_cl = cl4(y);
  auto dy_cl = cl4(dy);
  int64_t rows = (int64_t)x_cl.size(0) * x_cl.size(2) * x_cl.size(3);
  int C = x_cl.size(1);
  Tensor dx = at::empty_like(x_cl, x.options().memory_format(at::MemoryFormat::ChannelsLast));
  if (scale.numel() == 0) {  // forward used the halo-register path
    TORCH_CHECK(ps_lrn_v8_ok(C, size), "lrn: empty scale but v8 ineligible");
    if (is_bf16(x))
      ps_lrn_bwd_v8_bf16(x_cl.data_ptr(), y_cl.data_ptr(), dy_cl.data_ptr(),
                         dx.data_ptr(), rows, C, size, (float)alpha,
                         (float)beta, stream());
    else
      ps_lrn_bwd_v8_f32(x_cl.data_ptr<float>(), y_cl.data_ptr<float>(),
                        dy_cl.data_ptr<float>(), dx.data_ptr<float>(), rows,
                        C, size, (float)alpha, (float)beta, stream());
    return dx;
  }
  auto sc_cl = cl4(scale);
  // ratio workspace only feeds the C>256 fallback path; the row-block
  // kernels keep the ratio in LDS
  Tensor ratio = at::empty({C > 256 ? rows * C : 0},
                           x.options().dtype(at::kFloat));
  if (is_bf16(x))
    ps_lrn_bwd_bf16(x_cl.data_ptr(), y_cl.data_ptr(), sc_cl.data_ptr<float>(),
                    dy_cl.data_ptr(), dx.data_ptr(), ratio.data_ptr<float>(),
                    rows, C, size, (float)alpha, (float)beta, stream());
  else
    ps_lrn_bwd_f32(x_cl.data_ptr<float>(), y_cl.data_ptr<float>(),
                   sc_cl.data_ptr<float>(), dy_cl.data_ptr<float>(),
                   dx.data_ptr<float>(), ratio.data_ptr<float>(), rows, C,
                   size, (float)alpha, (float)beta, stream());
  return dx;
}

// ---------------------------------------------------------------------------
// Softmax family
// ---------------------------------------------------------------------------

Tensor softmax_forward(const Tensor& x) {
  check_float_like(x, "x");
  Tensor xc, y;
  int64_t rows;
  int C;
  if (x.dim() == 4) {
    xc = cl4(x);
    rows = (int64_t)xc.size(0) * xc.size(2) * xc.size(3);
    C = xc.size(1);
    y = at::empty_like(xc, x.options().memory_format(at::MemoryFormat::ChannelsLast));
  } else {
    xc = x.contiguous();
    rows = xc.numel() / xc.size(-1);
    C = xc.size(-1);
    y = at::empty_like(xc);
  }
  if (is_bf16(x))
    ps_softmax_rows_bf16(xc.data_ptr(), y.data_ptr(), rows, C, stream());
  else
    ps_softmax_rows_f32(xc.data_ptr<float>(), y.data_ptr<float>(), rows, C,
                        stream());
  return y;
}

Tensor softmax_backward(const Tensor& y, const Tensor& dy) {
  Tensor yc, dyc, dx;
  int64_t rows;
  int C;
  if (y.dim() == 4) {
    yc = cl4(y);
    dyc = cl4(dy);
    rows = (int64_t)yc.size(0) * yc.size(2) * yc.size(3);
    C = yc.size(1);
    dx = at::empty_like(yc, y.options().memory_format(at::MemoryFormat::ChannelsLast));
  } else {
    yc = y.contiguous();
    dyc = dy.contiguous();
    rows = yc.numel() / yc.size(-1);
    C = yc.size(-1);
    dx = at::empty_like(yc);
  }
  if (is_bf16(y))
    ps_softmax_bwd_rows_bf16(yc.data_ptr(), dyc.data_ptr(), dx.data_ptr(),
                             rows, C, stream());
  else
    ps_softmax_bwd_rows_f32(yc.data_ptr<float>(), dyc.data_ptr<float>(),
                            dx.data_ptr<float>(), rows, C, stream());
  return dx;
}

std::vector<Tensor> softmax_loss_forward(const Tensor& logits,
                                         const Tensor& labels) {
  check_float_like(logits, "logits");
  auto xc = logits.contiguous();
  auto lc = labels.contiguous().to(at::kFloat);
  int64_t rows = xc.size(0);
  int C = xc.size(1);
  Tensor prob = at::empty_like(xc);
  Tensor loss = at::zeros({}, xc.options().dtype(at::kFloat));
  if (is_bf16(xc))
    ps_softmax_loss_fwd_bf16(xc.data_ptr(), lc.data_ptr<float>(),
                             prob.data_ptr(), loss.data_ptr<float>(), rows, C,
                             stream());
  else
    ps_softmax_loss_fwd_f32(xc.data_ptr<float>(), lc.data_ptr<float>(),
                            prob.data_ptr<float>(), loss.data_ptr<float>(),
                            rows, C, stream());
  loss.div_((double)rows);
  return {loss, prob};
}

Tensor softmax_loss_backward(const Tensor& prob, const Tensor& labels,
                             double loss_weight) {
  auto pc = prob.contiguous();
  auto lc = labels.contiguous().to(at::kFloat);
  int64_t rows = pc.size(0);
  int C = pc.size(1);
  Tensor dx = at::empty_like(pc);
  if (is_bf16(pc))
    ps_softmax_loss_bwd_bf16(pc.data_ptr(), lc.data_ptr<float>(),
                             dx.data_ptr(), rows, C,
                             (float)(loss_weight / rows), stream());
  else
    ps_softmax_loss_bwd_f32(pc.data_ptr<float>(), lc.data_ptr<float>(),
                            dx.data_ptr<float>(), rows, C,
                            (float)(loss_weight / rows), stream());
  return dx;
}

// ---------------------------------------------------------------------------
// neuron ops
// ---------------------------------------------------------------------------

Tensor any_contig(const Tensor& t) {
  return (t.dim() == 4 && t.is_contiguous(at::MemoryFormat::ChannelsLast))
             ? t : t.contiguous();
}

Tensor relu_forward(const Tensor& x, double slope) {
  check_float_like(x, "x");
  auto xc = any_contig(x);
  Tensor y = at::empty_like(xc);
  if (is_bf16(x))
    ps_relu_fwd_bf16(xc.data_ptr(), y.data_ptr(), xc.numel(), (float)slope,
                     stream());
  else
    ps_relu_fwd_f32(xc.data_ptr<float>(), y.data_ptr<float>(), xc.numel(),
                    (float)slope, stream());
  return y;
}

Tensor relu_backward(const Tensor& x, const Tensor& dy, double slope,
                     bool in_place) {
  auto xc = any_contig(x);
  auto dyc = xc.dim() == 4 && xc.is_contiguous(at::MemoryFormat::ChannelsLast)
                 ? cl4(dy) : dy.contiguous();
  // in-place (dx == dy): elementwise same-index, safe; keeps the grad
  // buffer identity stable for the net-level colsum batch (a fresh dx
  // every iteration churned every fused-relu conv's dy pointer)
  Tensor dx = (in_place && dyc.is_alias_of(dy) &&
               dyc.data_ptr() == dy.data_ptr())
                  ? dyc : at::empty_like(xc);
  if (is_bf16(x))
    ps_relu_bwd_bf16(xc.data_ptr(), dyc.data_ptr(), dx.data_ptr(), xc.numel(),
                     (float)slope, stream());
  else
    ps_relu_bwd_f32(xc.data_ptr<float>(), dyc.data_ptr<float>(),
                    dx.data_ptr<float>(), xc.numel(), (float)slope, stream());
  return dx;
}

#define PS_BIND_UNARY(pyname, fn32, fn16)                                   \
  Tensor pyname(const Tensor& x) {                                          \
    check_float_like(x, #pyname);                                           \
    auto xc = any_contig(x);                                                \
    Tensor y = at::empty_like(xc);                                          \
    if (is_bf16(x))                                                         \
      fn16(xc.data_ptr(), y.data_ptr(), xc.numel(), stream());              \
    else                                                                    \
      fn32(xc.data_ptr<float>(), y.data_ptr<float>(), xc.numel(), stream());\
    return y;                                                               \
  }

#define PS_BIND_BINARY(pyname, fn32, fn16)                                  \
  Tensor pyname(const Tensor& a, const Tensor& b) {                         \
    auto ac = any_contig(a);                                                \
    auto bc = any_contig(b);                                                \
    Tensor y = at::empty_like(ac);                                          \
    if (is_bf16(a))                                                         \
      fn16(ac.data_ptr(), bc.data_ptr(), y.data_ptr(), ac.numel(), stream()); \
    else                                                                    \
      fn32(ac.data_ptr<float>(), bc.data_ptr<float>(), y.data_ptr<float>(), \
           ac.numel(), stream());                                           \
    return y;                                                               \
  }

PS_BIND_UNARY(sigmoid_forward, ps_sigmoid_fwd_f32, ps_sigmoid_fwd_bf16)
PS_BIND_BINARY(sigmoid_backward, ps_sigmoid_bwd_f32, ps_sigmoid_bwd_bf16)
PS_BIND_UNARY(tanh_forward, ps_tanh_fwd_f32, ps_tanh_fwd_bf16)
PS_BIND_BINARY(tanh_backward, ps_tanh_bwd_f32, ps_tanh_bwd_bf16)
PS_BIND_UNARY(bnll_forward, ps_bnll_fwd_f32, ps_bnll_fwd_bf16)
PS_BIND_BINARY(bnll_backward, ps_bnll_bwd_f32, ps_bnll_bwd_bf16)

std::vector<Tensor> dropout_forward(const Tensor& x, double ratio,
                                    int64_t seed, int64_t offset) {
  check_float_like(x, "x");
  auto xc = any_contig(x);
  Tensor y = at::empty_like(xc);
  Tensor mask = at::empty(xc.sizes(), xc.options().dtype(at::kByte));
  if (is_bf16(x))
    ps_dropout_fwd_bf16(xc.data_ptr(), y.data_ptr(), mask.data_ptr<uint8_t>(),
                        xc.numel(), (float)ratio, (uint64_t)seed,
                        (uint64_t)offset, stream());
  else
    ps_dropout_fwd_f32(xc.data_ptr<float>(), y.data_ptr<float>(),
                       mask.data_ptr<uint8_t>(), xc.numel(), (float)ratio,
                       (uint64_t)seed, (uint64_t)offset, stream());
  return {y, mask};
}

// graph-replayable dropout: philox offset lives in device memory and is
// incremented in-stream, so each replay draws a fresh mask
std::vector<Tensor> dropout_forward_offdev(const Tensor& x, double ratio,
                                           int64_t seed, Tensor offset_dev) {
  check_float_like(x, "x");
  auto xc = any_contig(x);
  Tensor y = at::empty_like(xc);
  Tensor mask = at::empty(xc.sizes(), xc.options().dtype(at::kByte));
  if (is_bf16(x))
    ps_dropout_fwd_bf16_offdev(xc.data_ptr(), y.data_ptr(),
                               mask.data_ptr<uint8_t>(), xc.numel(),
                               (float)ratio, (uint64_t)seed,
                               offset_dev.data_ptr(), stream());
  else
    ps_dropout_fwd_f32_offdev(xc.data_ptr<float>(), y.data_ptr<float>(),
                              mask.data_ptr<uint8_t>(), xc.numel(),
                              (float)ratio, (uint64_t)seed,
                              offset_dev.data_ptr(), stream());
  ps_u64_inc(offset_dev.data_ptr(), stream());
  return {y, mask};
}

Tensor dropout_backward(const Tensor& dy, const Tensor& mask, double ratio) {
  auto dyc = any_contig(dy);
  Tensor dx = at::empty_like(dyc);
  if (is_bf16(dy))
    ps_dropout_bwd_bf16(dyc.data_ptr(), mask.data_ptr<uint8_t>(),
                        dx.data_ptr(), dyc.numel(), (float)ratio, stream());
  else
    ps_dropout_bwd_f32(dyc.data_ptr<float>(), mask.data_ptr<uint8_t>(),
                       dx.data_ptr<float>(), dyc.numel(), (float)ratio,
                       stream());
  return dx;
}

// ---------------------------------------------------------------------------
// optimizer updates (in-place on fp32 master params)
// ---------------------------------------------------------------------------

Tensor threshold_forward(const Tensor& x, double thr) {
  auto xc = any_contig(x);
  Tensor y = at::empty_like(xc);
  if (is_bf16(x))
    ps_threshold_fwd_bf16(xc.data_ptr(), y.data_ptr(), xc.numel(),
                          (float)thr, stream());
  else
    ps_threshold_fwd_f32(xc.data_ptr<float>(), y.data_ptr<float>(),
                         xc.numel(), (float)thr, stream());
  return y;
}

// pairwise running max: y/mask updated in place; mask[i] = blob index of
// the current max (start by calling with mask filled 0 and y = blob 0)
void eltwise_max_step(Tensor y, const Tensor& b, Tensor mask, int64_t idx_b) {
  auto bc = any_contig(b);
  if (is_bf16(y))
    ps_eltwise_max_fwd_bf16(y.data_ptr(), bc.data_ptr(), y.data_ptr(),
                            mask.data_ptr<uint8_t>(), y.numel(), (int)idx_b,
                            stream());
  else
    ps_eltwise_max_fwd_f32(y.data_ptr<float>(), bc.data_ptr<float>(),
                           y.data_ptr<float>(), mask.data_ptr<uint8_t>(),
                           y.numel(), (int)idx_b, stream());
}

Tensor eltwise_max_backward(const Tensor& dy, const Tensor& mask,
                            int64_t idx) {
  auto dyc = any_contig(dy);
  Tensor dx = at::empty_like(dyc);
  if (is_bf16(dy))
    ps_eltwise_max_bwd_bf16(dyc.data_ptr(), mask.data_ptr<uint8_t>(),
                            dx.data_ptr(), dyc.numel(), (int)idx, stream());
  else
    ps_eltwise_max_bwd_f32(dyc.data_ptr<float>(), mask.data_ptr<uint8_t>(),
                           dx.data_ptr<float>(), dyc.numel(), (int)idx,
                           stream());
  return dx;
}

Tensor contrastive_forward(const Tensor& dist_sq, const Tensor& sim,
                           double margin, bool legacy) {
  auto d = dist_sq.contiguous().to(at::kFloat);
  auto sm = sim.contiguous().to(at::kFloat);
  Tensor loss = at::empty_like(d);
  ps_contrastive_fwd_f32(d.data_ptr<float>(), sm.data_ptr<float>(),
                         loss.data_ptr<float>(), d.numel(), (float)margin,
                         legacy ? 1 : 0, stream());
  return loss;
}

void sgd_update(Tensor w, const Tensor& g, Tensor h, double lr, double mom,
                double wd) {
  ps_sgd_update(w.data_ptr<float>(), g.data_ptr<float>(), h.data_ptr<float>(),
                w.numel(), (float)lr, (float)mom, (float)wd, stream());
}
// graph-replayable: lr read from lr_dev[0] * lr_mult at kernel time
void sgd_update_lrdev(Tensor w, const Tensor& g, Tensor h, double lr_mult,
                      double mom, double wd, const Tensor& lr_dev) {
  ps_sgd_update_lrdev(w.data_ptr<float>(), g.data_ptr<float>(),
                      h.data_ptr<float>(), w.numel(), (float)lr_mult,
                      (float)mom, (float)wd, lr_dev.data_ptr<float>(),
                      stream());
}
void nesterov_update(Tensor w, const Tensor& g, Tensor h, double lr,
                     double mom, double wd) {
  ps_nesterov_update(w.data_ptr<float>(), g.data_ptr<float>(),
                     h.data_ptr<float>(), w.numel(), (float)lr, (float)mom,
                     (float)wd, stream());
}
void adagrad_update(Tensor w, const Tensor& g, Tensor h, double lr,
                    double delta, double wd) {
  ps_adagrad_update(w.data_ptr<float>(), g.data_ptr<float>(),
                    h.data_ptr<float>(), w.numel(), (float)lr, (float)delta,
                    (float)wd, stream());
}

// ---------------------------------------------------------------------------
// Multi-tensor apply tables (sgd.hip MTDesc/MTZeroDesc/MTChunk). Host-side
// mirror structs -- layouts must match sgd.hip exactly (same compiler, both
// trivially copyable, so they do).
// ---------------------------------------------------------------------------
struct MTDescHost {
  float* w;
  const float* g;
  float* h;
  int64_t n;
  float lr_mult;
  float wd;
};
struct MTZeroDescHost {
  float* p;
  int64_t n;
};
struct MTChunkHost {
  int t;
  int64_t off;
};

Tensor blob_to_dev(const void* src, int64_t bytes, const Tensor& like) {
  Tensor host = at::from_blob(const_cast<void*>(src), {bytes},
                              at::TensorOptions().dtype(at::kByte));
  return host.to(like.device());
}

// returns {desc_dev(u8), chunk_dev(u8), nchunks} -- build once, reuse forever
std::vector<Tensor> sgd_mt_prepare(std::vector<Tensor> ws,
                                   std::vector<Tensor> gs,
                                   std::vector<Tensor> hs,
                                   std::vector<double> lr_mults,
                                   std::vector<double> wds) {
  const int CHUNK = ps_mt_chunk_elts();
  const size_t nt = ws.size();
  TORCH_CHECK(gs.size() == nt && hs.size() == nt && lr_mults.size() == nt &&
              wds.size() == nt, "sgd_mt_prepare: length mismatch");
  std::vector<MTDescHost> descs(nt);
  std::vector<MTChunkHost> chunks;
  for (size_t t = 0; t < nt; ++t) {
    TORCH_CHECK(ws[t].is_cuda() && ws[t].is_contiguous() &&
                ws[t].scalar_type() == at::kFloat &&
                gs[t].is_contiguous() && hs[t].is_contiguous(),
                "sgd_mt_prepare: params must be contiguous f32 CUDA");
    int64_t n = ws[t].numel();
    TORCH_CHECK(gs[t].numel() == n && hs[t].numel() == n);
    descs[t] = {ws[t].data_ptr<float>(), gs[t].data_ptr<float>(),
                hs[t].data_ptr<float>(), n, (float)lr_mults[t],
                (float)wds[t]};
    for (int64_t off = 0; off < n; off += CHUNK)
      chunks.push_back({(int)t, off});
  }
  return {blob_to_dev(descs.data(), nt * sizeof(MTDescHost), ws[0]),
          blob_to_dev(chunks.data(), chunks.size() * sizeof(MTChunkHost),
                      ws[0]),
          at::scalar_tensor((int64_t)chunks.size())};
}

void sgd_mt_run(const Tensor& desc_dev, const Tensor& chunk_dev,
                int64_t nchunks, double lr, double mom,
                const c10::optional<Tensor>& lr_dev) {
  ps_sgd_mt(desc_dev.data_ptr(), chunk_dev.data_ptr(), (int)nchunks,
            (float)lr, (float)mom,
            lr_dev.has_value() ? lr_dev->data_ptr<float>() : nullptr,
            stream());
}

std::vector<Tensor> zero_mt_prepare(std::vector<Tensor> ts) {
  const int CHUNK = ps_mt_chunk_elts();
  std::vector<MTZeroDescHost> descs(ts.size());
  std::vector<MTChunkHost> chunks;
  for (size_t t = 0; t < ts.size(); ++t) {
    TORCH_CHECK(ts[t].is_cuda() && ts[t].is_contiguous() &&
                ts[t].scalar_type() == at::kFloat,
                "zero_mt_prepare: tensors must be contiguous f32 CUDA");
    descs[t] = {ts[t].data_ptr<float>(), ts[t].numel()};
    for (int64_t off = 0; off < ts[t].numel(); off += CHUNK)
      chunks.push_back({(int)t, off});
  }
  return {blob_to_dev(descs.data(), descs.size() * sizeof(MTZeroDescHost),
                      ts[0]),
          blob_to_dev(chunks.data(), chunks.size() * sizeof(MTChunkHost),
                      ts[0]),
          at::scalar_tensor((int64_t)chunks.size())};
}

void zero_mt_run(const Tensor& desc_dev, const Tensor& chunk_dev,
                 int64_t nchunks) {
  ps_zero_mt(desc_dev.data_ptr(), chunk_dev.data_ptr(), (int)nchunks,
             stream());
}

// layout mirror of ps::ColsumDesc (elementwise.hip)
struct ColsumDescHost {
  const void* dy;
  float* db;
  int64_t R;
  int64_t rows_per;  // rows per chunk (block)
  int C;
};

// one launch sums every deferred conv bias gradient: descs keyed on the
// (now identity-stable) activation-grad buffers. Chunk.off = start row.
std::vector<Tensor> colsum_mt_prepare(std::vector<Tensor> dys,
                                      std::vector<Tensor> dbs) {
  std::vector<ColsumDescHost> descs(dys.size());
  std::vector<MTChunkHost> chunks;
  for (size_t t = 0; t < dys.size(); ++t) {
    const Tensor& d = dys[t];
    // NO cl4() here: a layout copy would leave the desc pointing at a
    // temporary. The conv top diffs are channels-last by construction.
    const int vec = 16 / (int)d.element_size();
    TORCH_CHECK(d.is_cuda() && d.dim() == 4 &&
                d.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                d.size(1) % vec == 0 &&
                dbs[t].scalar_type() == at::kFloat,
                "colsum_mt: channels-last CUDA dy with C % 16B required");
    int C = (int)d.size(1);
    int64_t R = d.numel() / C;
    // ~256K elements per block: coarse enough that terminal atomics on
    // the C-wide bias vectors stay rare
    int64_t rows_per = std::max<int64_t>(4, (262144 / C + 3) & ~3LL);
    descs[t] = {d.data_ptr(), dbs[t].data_ptr<float>(), R, rows_per, C};
    for (int64_t r = 0; r < R; r += rows_per)
      chunks.push_back({(int)t, r});
  }
  int max_c = 0;
  for (auto& de : descs) max_c = std::max(max_c, de.C);
  TORCH_CHECK(max_c <= 8192, "colsum_mt: C exceeds the LDS budget");
  return {blob_to_dev(descs.data(), descs.size() * sizeof(ColsumDescHost),
                      dys[0]),
          blob_to_dev(chunks.data(), chunks.size() * sizeof(MTChunkHost),
                      dys[0]),
          at::scalar_tensor((int64_t)chunks.size()),
          at::scalar_tensor((int64_t)max_c)};
}

void colsum_mt_run(const Tensor& desc_dev, const Tensor& chunk_dev,
                   int64_t nchunks, bool bf16, int64_t max_c) {
  ps_colsum_mt(desc_dev.data_ptr(), chunk_dev.data_ptr(), (int)nchunks,
               bf16 ? 1 : 0, (int)max_c, stream());
}

struct MTUnpackDescHost {
  const float* dwk;
  float* dw;
  int64_t n;
  int Co, Cig, kh, kw, ldk;
};

// one table for every deferred conv-wgrad unpack (dwk khwc -> dw NCHW,
// accumulate); identities are the persistent dwk buffers + param diffs
std::vector<Tensor> unpack_mt_prepare(std::vector<Tensor> dwks,
                                      std::vector<Tensor> dws,
                                      std::vector<int64_t> Cigs,
                                      std::vector<int64_t> khs,
                                      std::vector<int64_t> kws) {
  const int CHUNK = ps_mt_chunk_elts();
  const size_t nt = dwks.size();
  std::vector<MTUnpackDescHost> descs(nt);
  std::vector<MTChunkHost> chunks;
  for (size_t t = 0; t < nt; ++t) {
    TORCH_CHECK(dwks[t].is_cuda() && dwks[t].scalar_type() == at::kFloat &&
                dws[t].is_cuda() && dws[t].is_contiguous() &&
                dws[t].scalar_type() == at::kFloat,
                "unpack_mt_prepare: f32 CUDA tensors required");
    int Co = (int)dwks[t].size(0);
    int ldk = (int)dwks[t].size(1);
    int64_t n = dws[t].numel();
    TORCH_CHECK(n == (int64_t)Co * Cigs[t] * khs[t] * kws[t]);
    descs[t] = {dwks[t].data_ptr<float>(), dws[t].data_ptr<float>(), n,
                Co, (int)Cigs[t], (int)khs[t], (int)kws[t], ldk};
    for (int64_t off = 0; off < n; off += CHUNK)
      chunks.push_back({(int)t, off});
  }
  return {blob_to_dev(descs.data(), nt * sizeof(MTUnpackDescHost), dwks[0]),
          blob_to_dev(chunks.data(), chunks.size() * sizeof(MTChunkHost),
                      dwks[0]),
          at::scalar_tensor((int64_t)chunks.size())};
}

void unpack_mt_run(const Tensor& desc_dev, const Tensor& chunk_dev,
                   int64_t nchunks) {
  ps_unpack_mt(desc_dev.data_ptr(), chunk_dev.data_ptr(), (int)nchunks,
               stream());
}

struct MTRepackDescHost {
  const float* src;
  void* wk;
  void* wkT;
  int64_t n;
  int Co, Cig, kh, kw, G, ldk;
};

std::vector<Tensor> repack_mt_prepare(std::vector<Tensor> masters,
                                      std::vector<Tensor> wks,
                                      std::vector<Tensor> wkTs,
                                      std::vector<int64_t> Gs) {
  const int CHUNK = ps_mt_chunk_elts();
  const size_t nt = masters.size();
  std::vector<MTRepackDescHost> descs(nt);
  std::vector<MTChunkHost> chunks;
  for (size_t t = 0; t < nt; ++t) {
    const Tensor& m = masters[t];
    TORCH_CHECK(m.is_cuda() && m.is_contiguous() &&
                m.scalar_type() == at::kFloat && m.dim() == 4 &&
                wks[t].scalar_type() == at::kBFloat16,
                "repack_mt_prepare: f32 NCHW masters + bf16 outputs");
    const bool has_t = wkTs[t].numel() > 0;
    TORCH_CHECK(!has_t || wkTs[t].scalar_type() == at::kBFloat16);
    descs[t] = {m.data_ptr<float>(), wks[t].data_ptr(),
                has_t ? wkTs[t].data_ptr() : nullptr,
                m.numel(), (int)m.size(0), (int)m.size(1), (int)m.size(2),
                (int)m.size(3), (int)Gs[t], (int)wks[t].size(1)};
    for (int64_t off = 0; off < m.numel(); off += CHUNK)
      chunks.push_back({(int)t, off});
  }
  return {blob_to_dev(descs.data(), nt * sizeof(MTRepackDescHost),
                      masters[0]),
          blob_to_dev(chunks.data(), chunks.size() * sizeof(MTChunkHost),
                      masters[0]),
          at::scalar_tensor((int64_t)chunks.size())};
}

void repack_mt_run(const Tensor& desc_dev, const Tensor& chunk_dev,
                   int64_t nchunks) {
  ps_repack_mt(desc_dev.data_ptr(), chunk_dev.data_ptr(), (int)nchunks,
               stream());
}

// standalone bias colsum (eager fallback for the net-level batch)
void colsum_acc(const Tensor& dy, Tensor db) {
  auto d = dy.dim() == 4 ? cl4(dy) : dy.contiguous();
  int C = (int)(dy.dim() == 4 ? d.size(1) : d.size(-1));
  int64_t R = d.numel() / C;
  if (is_bf16(d))
    ps_colsum_bf16(d.data_ptr(), db.data_ptr<float>(), R, C, stream());
  else
    ps_colsum_f32(d.data_ptr<float>(), db.data_ptr<float>(), R, C,
                  stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gemm", &gemm);
  m.def("linear_forward", &linear_forward);
  m.def("linear_backward", &linear_backward);
  m.def("gemm_at_b", &gemm_at_b);
  m.def("conv2d_forward_ex", &conv2d_forward_ex);
  m.def("conv2d_backward_input", &conv2d_backward_input);
  m.def("conv2d_backward_weight_acc", &conv2d_backward_weight_acc);
  m.def("set_implicit_gemm", &set_implicit_gemm);
  m.def("set_implicit_threshold", &set_implicit_threshold);
  m.def("concat_channels", &concat_channels);
  m.def("slice_channels", &slice_channels);
  m.def("split_channels", &split_channels);
  m.def("pool_max_forward", &pool_max_forward);
  m.def("pool_max_backward", &pool_max_backward);
  m.def("pool_ave_forward", &pool_ave_forward);
  m.def("pool_ave_backward", &pool_ave_backward);
  m.def("pool_stoch_forward_train", &pool_stoch_forward_train);
  m.def("pool_stoch_forward_test", &pool_stoch_forward_test);
  m.def("lrn_forward", &lrn_forward);
  m.def("lrn_backward", &lrn_backward);
  m.def("softmax_forward", &softmax_forward);
  m.def("softmax_backward", &softmax_backward);
  m.def("softmax_loss_forward", &softmax_loss_forward);
  m.def("softmax_loss_backward", &softmax_loss_backward);
  m.def("relu_forward", &relu_forward);
  m.def("relu_backward", &relu_backward);
  m.def("sigmoid_forward", &sigmoid_forward);
  m.def("sigmoid_backward", &sigmoid_backward);
  m.def("tanh_forward", &tanh_forward);
  m.def("tanh_backward", &tanh_backward);
  m.def("bnll_forward", &bnll_forward);
  m.def("bnll_backward", &bnll_backward);
  m.def("dropout_forward", &dropout_forward);
  m.def("dropout_backward", &dropout_backward);
  m.def("sgd_update", &sgd_update);
  m.def("sgd_update_lrdev", &sgd_update_lrdev);
  m.def("threshold_forward", &threshold_forward);
  m.def("eltwise_max_step", &eltwise_max_step);
  m.def("eltwise_max_backward", &eltwise_max_backward);
  m.def("contrastive_forward", &contrastive_forward);
  m.def("sgd_mt_prepare", &sgd_mt_prepare);
  m.def("sgd_mt_run", &sgd_mt_run);
  m.def("zero_mt_prepare", &zero_mt_prepare);
  m.def("zero_mt_run", &zero_mt_run);
  m.def("repack_mt_prepare", &repack_mt_prepare);
  m.def("colsum_mt_prepare", &colsum_mt_prepare);
  m.def("colsum_mt_run", &colsum_mt_run);
  m.def("colsum_acc", &colsum_acc);
  m.def("unpack_mt_prepare", &unpack_mt_prepare);
  m.def("unpack_mt_run", &unpack_mt_run);
  m.def("repack_mt_run", &repack_mt_run);
  m.def("colT_ld", [](int64_t G, int64_t C, int64_t kh, int64_t kw,
                      int64_t vec) {
    return (int64_t)ps_colT_ld((int)G, (int)C, (int)kh, (int)kw, (int)vec);
  });
  m.def("dropout_forward_offdev", &dropout_forward_offdev);
  m.def("nesterov_update", &nesterov_update);
  m.def("adagrad_update", &adagrad_update);
  m.attr("compute_dtypes") = std::vector<std::string>{"float32", "bfloat16"};
}
