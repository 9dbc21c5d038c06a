// Torch bindings for the tnn_amd gfx950 kernels.
//
// Tensor-level checks live here; the .hip translation units only see raw
// pointers + shapes + the stream. Every entry point CHECKs device/layout —
// a CPU tensor reaching these is a dispatch bug in the python layer.

#include <array>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "kernels.h"

namespace tnn {

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}

// 16-byte zero page for glds source redirection (invalid lanes load
// zeros instead of being exec-masked, which would leave stale LDS)
static void* zero_page(const at::Tensor& like) {
  static at::Tensor z;
  if (!z.defined() || z.device() != like.device())
    z = at::zeros({16}, like.options().dtype(at::kFloat));
  return z.data_ptr();
}

static DT dt_of(const at::Tensor& t) {
  if (t.scalar_type() == at::kFloat) return DT::F32;
  if (t.scalar_type() == at::kBFloat16) return DT::BF16;
  TORCH_CHECK(false, "tnn_amd kernels support fp32/bf16, got ", t.scalar_type());
}

#define CHECK_IN(t)                                             \
  TORCH_CHECK((t).is_cuda(), #t " must be on GPU");             \
  TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// ---- elementwise -----------------------------------------------------------
at::Tensor act_fwd(const at::Tensor& x, int64_t kind) {
  CHECK_IN(x);
  auto y = at::empty_like(x);
  act_fwd_launch(dt_of(x), x.data_ptr(), y.data_ptr(), x.numel(), (int)kind,
                 cur_stream());
  return y;
}

at::Tensor act_bwd(const at::Tensor& dy, const at::Tensor& x,
                   const at::Tensor& y, int64_t kind) {
  CHECK_IN(dy);
  auto dx = at::empty_like(dy);
  act_bwd_launch(dt_of(dy), dy.data_ptr(), x.data_ptr(), y.data_ptr(),
                 dx.data_ptr(), dy.numel(), (int)kind, cur_stream());
  return dx;
}

at::Tensor add_act_fwd(const at::Tensor& a, const at::Tensor& b,
                       int64_t kind) {
  CHECK_IN(a);
  CHECK_IN(b);
  TORCH_CHECK(a.sizes() == b.sizes() && a.scalar_type() == b.scalar_type());
  auto y = at::empty_like(a);
  add_act_fwd_launch(dt_of(a), a.data_ptr(), b.data_ptr(), y.data_ptr(),
                     a.numel(), (int)kind, cur_stream());
  return y;
}

at::Tensor add_act_bwd(const at::Tensor& dy, const at::Tensor& y,
                       int64_t kind) {
  CHECK_IN(dy);
  auto g = at::empty_like(dy);
  add_act_bwd_launch(dt_of(dy), dy.data_ptr(), y.data_ptr(), g.data_ptr(),
                     dy.numel(), (int)kind, cur_stream());
  return g;
}

at::Tensor relu_bwd_mask(const at::Tensor& dy, const at::Tensor& y) {
  CHECK_IN(dy);
  CHECK_IN(y);
  auto dx = at::empty_like(dy);
  relu_bwd_mask_launch(dt_of(dy), dy.data_ptr(), y.data_ptr(), dx.data_ptr(),
                       dy.numel(), cur_stream());
  return dx;
}

std::vector<at::Tensor> dropout_fwd(const at::Tensor& x, double p,
                                    int64_t seed,
                                    const c10::optional<at::Tensor>& ctr) {
  CHECK_IN(x);
  auto y = at::empty_like(x);
  auto mask = at::empty(x.sizes(), x.options().dtype(at::kByte));
  const int64_t* ctrp = ctr.has_value() ? ctr->data_ptr<int64_t>() : nullptr;
  dropout_fwd_launch(dt_of(x), x.data_ptr(), y.data_ptr(),
                     mask.data_ptr<uint8_t>(), x.numel(), (float)p,
                     (uint64_t)seed, ctrp, cur_stream());
  return {y, mask};
}

at::Tensor dropout_bwd(const at::Tensor& dy, const at::Tensor& mask, double p) {
  CHECK_IN(dy);
  auto dx = at::empty_like(dy);
  dropout_bwd_launch(dt_of(dy), dy.data_ptr(), mask.data_ptr<uint8_t>(),
                     dx.data_ptr(), dy.numel(), (float)p, cur_stream());
  return dx;
}

at::Tensor colsum(const at::Tensor& x) {
  CHECK_IN(x);
  TORCH_CHECK(x.dim() == 2, "colsum wants [rows, cols]");
  const int slices = colsum_ws_slices(dt_of(x), x.data_ptr(), x.size(0),
                                      x.size(1));
  // vec path writes per-slice slabs + a finalize (no zero-init needed);
  // scalar fallback atomically accumulates into a zeroed output
  auto out = slices >= 1
                 ? at::empty({x.size(1)}, x.options().dtype(at::kFloat))
                 : at::zeros({x.size(1)}, x.options().dtype(at::kFloat));
  at::Tensor ws;
  float* wp = nullptr;
  if (slices > 1) {
    ws = at::empty({(int64_t)slices * x.size(1)},
                   x.options().dtype(at::kFloat));
    wp = ws.data_ptr<float>();
  }
  colsum_launch(dt_of(x), x.data_ptr(), out.data_ptr(), wp, slices, x.size(0),
                x.size(1), cur_stream());
  if (x.scalar_type() != at::kFloat) {
    auto out_t = at::empty({x.size(1)}, x.options());
    cast_f32_launch(dt_of(x), out.data_ptr<float>(), out_t.data_ptr(),
                    out.numel(), cur_stream());
    return out_t;
  }
  return out;
}

// ---- gemm ------------------------------------------------------------------
at::Tensor gemm(const at::Tensor& a, const at::Tensor& b,
                const c10::optional<at::Tensor>& bias, int64_t act_kind,
                const c10::optional<at::Tensor>& residual = c10::nullopt,
                const c10::optional<at::Tensor>& ln_gamma = c10::nullopt,
                const c10::optional<at::Tensor>& ln_beta = c10::nullopt,
                double ln_eps = 1e-5) {
  CHECK_IN(a);
  CHECK_IN(b);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(0),
              "gemm shapes ", a.sizes(), " @ ", b.sizes());
  TORCH_CHECK(a.scalar_type() == b.scalar_type(), "gemm dtype mismatch ",
              a.scalar_type(), " vs ", b.scalar_type());
  int M = a.size(0), K = a.size(1), N = b.size(1);
  auto c = at::empty({M, N}, a.options());
  const void* bp = bias.has_value() ? bias->data_ptr() : nullptr;
  const void* rp = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous() &&
                residual->scalar_type() == a.scalar_type() &&
                residual->numel() == (int64_t)M * N,
                "gemm residual mismatch");
    rp = residual->data_ptr();
  }
  const float* lng = nullptr;
  const float* lnb = nullptr;
  if (ln_gamma.has_value()) {
    TORCH_CHECK(M == 1 && K <= GEMV1_MAX_K_DECL && ln_beta.has_value() &&
                ln_gamma->scalar_type() == at::kFloat &&
                ln_beta->scalar_type() == at::kFloat,
                "fused ln: decode GEMV only, fp32 gamma/beta");
    lng = ln_gamma->data_ptr<float>();
    lnb = ln_beta->data_ptr<float>();
  }
  if (M == 1) {
    if (K <= GEMV1_MAX_K_DECL) {
      // LDS-cached matvec; K-split across blocks to fill the chip (fp32
      // partials + a parallel finalize) when N alone is too narrow
      int ns = gemv_nn1_nsplit(N, K);
      at::Tensor part;
      float* pp = nullptr;
      if (ns > 1) {
        part = at::empty({(int64_t)ns * N}, a.options().dtype(at::kFloat));
        pp = part.data_ptr<float>();
      }
      gemv_nn1_launch(dt_of(a), a.data_ptr(), b.data_ptr(), bp, rp,
                      c.data_ptr(), pp, ns, lng, lnb, (float)ln_eps, N, K,
                      (int)act_kind, cur_stream());
      return c;
    }
    // huge-K fallback: K-split partials + finalize (no residual fusion)
    TORCH_CHECK(rp == nullptr, "gemm residual unsupported for K > 8192 M=1");
    int ks = gemv_nn_ksplits(N, K);
    auto ws = at::empty({(int64_t)ks * N}, a.options().dtype(at::kFloat));
    gemv_nn_launch(dt_of(a), a.data_ptr(), b.data_ptr(), bp, c.data_ptr(),
                   ws.data_ptr<float>(), ks, N, K, (int)act_kind,
                   cur_stream());
    return c;
  }
  bool vec_ok = (dt_of(a) == DT::F32 ? K % 4 == 0 : K % 8 == 0) &&
                (((uintptr_t)a.data_ptr() & 15) == 0);
  if (M >= 64 && vec_ok) {
    // transpose the (weight-sized) B once and take the NT glds path: B rows
    // become k-contiguous LDS-DMA targets instead of a VGPR scatter
    auto bt = at::empty({(int64_t)N, (int64_t)K}, b.options());
    transpose_w_launch(dt_of(b), b.data_ptr(), bt.data_ptr(), 1, 1, K, N,
                       cur_stream());
    static const bool use32 = [] {
      const char* e = getenv("TNN_GEMM32");
      return !(e && e[0] == '0');
    }();
    if (use32 && dt_of(a) == DT::BF16) {
      // 32x32x16-MFMA tile core (peak issue at this kernel's 2
      // waves/SIMD residency; the 16x16x32 core needs 3-4)
      int z = N % 4 == 0 ? gemm_nt32_zsplits(M, N, K) : 1;
      at::Tensor ws;
      float* wp = nullptr;
      if (z > 1) {
        ws = at::empty({(int64_t)z * M * N}, a.options().dtype(at::kFloat));
        wp = ws.data_ptr<float>();
      }
      gemm_nt32_launch(a.data_ptr(), bt.data_ptr(), wp, nullptr, bp, rp,
                       c.data_ptr(), z, zero_page(a), M, N, K, (int)act_kind,
                       cur_stream());
      return c;
    }
    const int zf = N % 4 == 0 ? gemm_nt_fsplits(dt_of(a), M, N, K) : 1;
    if (zf > 1 && K <= 3072 && ((uintptr_t)bt.data_ptr() & 15) == 0) {
      auto ws = at::empty({(int64_t)zf * M * N},
                          a.options().dtype(at::kFloat));
      gemm_nt_zf_launch(dt_of(a), a.data_ptr(), bt.data_ptr(),
                        ws.data_ptr<float>(), bp, rp, c.data_ptr(), zf,
                        zero_page(a), M, N, K, (int)act_kind, cur_stream());
      return c;
    }
    gemm_launch(dt_of(a), a.data_ptr(), bt.data_ptr(), bp, c.data_ptr(),
                zero_page(a), rp, M, N, K, /*trans_b=*/true, (int)act_kind,
                cur_stream());
    return c;
  }
  gemm_launch(dt_of(a), a.data_ptr(), b.data_ptr(), bp, c.data_ptr(),
              zero_page(a), rp, M, N, K, /*trans_b=*/false, (int)act_kind,
              cur_stream());
  return c;
}

at::Tensor gemm_nt(const at::Tensor& a, const at::Tensor& b) {
  // c[m, j] = sum_k a[m, k] * b[j, k]
  CHECK_IN(a);
  CHECK_IN(b);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(1) == b.size(1),
              "gemm_nt shapes ", a.sizes(), " @ ", b.sizes(), "^T");
  int M = a.size(0), K = a.size(1), N = b.size(0);
  auto c = at::empty({M, N}, a.options());
  static const bool use32nt = [] {
    const char* e = getenv("TNN_GEMM32");
    return !(e && e[0] == '0');
  }();
  if (use32nt && dt_of(a) == DT::BF16 && K % 8 == 0 &&
      ((uintptr_t)a.data_ptr() & 15) == 0 &&
      ((uintptr_t)b.data_ptr() & 15) == 0) {
    int z32 = N % 4 == 0 ? gemm_nt32_zsplits(M, N, K) : 1;
    at::Tensor ws;
    float* wp = nullptr;
    if (z32 > 1) {
      ws = at::empty({(int64_t)z32 * M * N}, a.options().dtype(at::kFloat));
      wp = ws.data_ptr<float>();
    }
    gemm_nt32_launch(a.data_ptr(), b.data_ptr(), wp, nullptr, nullptr,
                     nullptr, c.data_ptr(), z32, zero_page(a), M, N, K, 0,
                     cur_stream());
    return c;
  }
  int z = gemm_nt_zsplits(dt_of(a), M, N, K);
  if (z > 1 && ((uintptr_t)a.data_ptr() & 15) == 0 &&
      ((uintptr_t)b.data_ptr() & 15) == 0) {
    // huge-K underfilled shape (vocab-head dgrad): K-split fp32 slabs
    auto ws = at::empty({(int64_t)z * M * N}, a.options().dtype(at::kFloat));
    gemm_nt_z_launch(dt_of(a), a.data_ptr(), b.data_ptr(),
                     ws.data_ptr<float>(), c.data_ptr(), dt_of(a), z,
                     zero_page(a), M, N, K, cur_stream());
    return c;
  }
  gemm_launch(dt_of(a), a.data_ptr(), b.data_ptr(), nullptr, c.data_ptr(),
              zero_page(a), /*resid=*/nullptr, M, N, K, /*trans_b=*/true, 0,
              cur_stream());
  return c;
}

at::Tensor gemm_tn(const at::Tensor& a, const at::Tensor& b, bool out_in_dt) {
  // c[k, n] = sum_m a[m, k] * b[m, n]  (fp32 accumulate; output fp32, or
  // a's dtype when out_in_dt -- the cast rides the split-K reduce)
  CHECK_IN(a);
  CHECK_IN(b);
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2 && a.size(0) == b.size(0),
              "gemm_tn shapes ", a.sizes(), "^T @ ", b.sizes());
  int M = a.size(0), K = a.size(1), N = b.size(1);
  DT out_dt = out_in_dt ? dt_of(a) : DT::F32;
  auto c = at::empty({K, N}, out_dt == DT::F32
                                 ? a.options().dtype(at::kFloat)
                                 : a.options());
  int z = gemm_tn_zsplits(M, N, K);
  at::Tensor ws;
  float* wsp = nullptr;
  if (z > 1 || out_dt != DT::F32) {
    ws = at::empty({(int64_t)z * K * N}, a.options().dtype(at::kFloat));
    wsp = ws.data_ptr<float>();
  }
  gemm_tn_launch(dt_of(a), a.data_ptr(), b.data_ptr(), c.data_ptr(), out_dt,
                 wsp, z, zero_page(a), M, N, K, cur_stream());
  return c;
}

at::Tensor mfma_selftest(const at::Tensor& a, const at::Tensor& b) {
  // a: [16, 32] bf16, b: [32, 16] bf16 -> d [16, 16] f32 (layout check)
  CHECK_IN(a);
  auto d = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  mfma_selftest_launch(a.data_ptr(), b.data_ptr(), d.data_ptr<float>(),
                       cur_stream());
  return d;
}

at::Tensor mfma_selftest_f32(const at::Tensor& a, const at::Tensor& b) {
  // a: [16, 4] f32, b: [4, 16] f32 -> d [16, 16] f32
  CHECK_IN(a);
  auto d = at::zeros({16, 16}, a.options().dtype(at::kFloat));
  mfma_selftest_f32_launch(a.data_ptr<float>(), b.data_ptr<float>(),
                           d.data_ptr<float>(), cur_stream());
  return d;
}

// ---- conv2d ----------------------------------------------------------------
static ConvShape conv_shape(const at::Tensor& x_like, int Cin, int Cout, int KH,
                            int KW, int SH, int SW, int PH, int PW) {
  ConvShape cs;
  cs.N = x_like.size(0);
  cs.H = x_like.size(1);
  cs.W = x_like.size(2);
  cs.Cin = Cin;
  cs.Cout = Cout;
  cs.KH = KH;
  cs.KW = KW;
  cs.SH = SH;
  cs.SW = SW;
  cs.PH = PH;
  cs.PW = PW;
  cs.OH = (cs.H + 2 * PH - KH) / SH + 1;
  cs.OW = (cs.W + 2 * PW - KW) / SW + 1;
  cs.init_fdiv();
  return cs;
}

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, int64_t sh,
                      int64_t sw, int64_t ph, int64_t pw, bool relu) {
  CHECK_IN(x);
  CHECK_IN(w);
  TORCH_CHECK(x.dim() == 4 && w.dim() == 4 && x.size(3) == w.size(2),
              "conv2d_fwd x ", x.sizes(), " w ", w.sizes());
  TORCH_CHECK(x.scalar_type() == w.scalar_type(), "conv2d dtype mismatch ",
              x.scalar_type(), " vs ", w.scalar_type());
  TORCH_CHECK(!bias.has_value() || bias->scalar_type() == x.scalar_type(),
              "conv2d bias dtype mismatch");
  auto cs = conv_shape(x, w.size(2), w.size(3), w.size(0), w.size(1), sh, sw,
                       ph, pw);
  auto y = at::empty({cs.N, cs.OH, cs.OW, cs.Cout}, x.options());
  const void* bp = bias.has_value() ? bias->data_ptr() : nullptr;
  at::Tensor wt2;
  const void* wt2p = nullptr;
  if (conv2d_fwd_wants_db(dt_of(x), x.data_ptr(), cs)) {
    wt2 = at::empty({(int64_t)cs.Cout, (int64_t)cs.KH * cs.KW * cs.Cin},
                    w.options());
    transpose_w_fwd_launch(dt_of(x), w.data_ptr(), wt2.data_ptr(),
                           cs.KH * cs.KW, cs.Cin, cs.Cout, cur_stream());
    wt2p = wt2.data_ptr();
  }
  conv2d_fwd_launch(dt_of(x), x.data_ptr(), w.data_ptr(), wt2p, bp,
                    y.data_ptr(), zero_page(x), nullptr, cs, relu,
                    cur_stream());
  return y;
}

// conv fwd + fused per-channel sum/sumsq for a following train-mode BN
// (linear act). Returns {y, stats[2, Cout]}; stats is EMPTY when the
// double-buffered kernel (the only one carrying the stats epilogue) is not
// eligible -- callers fall back to the separate BN stats pass.
std::vector<at::Tensor> conv2d_fwd_stats(const at::Tensor& x,
                                         const at::Tensor& w,
                                         const c10::optional<at::Tensor>& bias,
                                         int64_t sh, int64_t sw, int64_t ph,
                                         int64_t pw) {
  CHECK_IN(x);
  CHECK_IN(w);
  auto cs = conv_shape(x, w.size(2), w.size(3), w.size(0), w.size(1), sh, sw,
                       ph, pw);
  auto y = at::empty({cs.N, cs.OH, cs.OW, cs.Cout}, x.options());
  const void* bp = bias.has_value() ? bias->data_ptr() : nullptr;
  if (!conv2d_fwd_wants_db(dt_of(x), x.data_ptr(), cs)) {
    // stats fusion needs the db kernel; ineligible shapes (e.g. K > 2304)
    // still must COMPUTE y via the plain path. (Round-1 latent bug: this
    // early return handed back uninitialized y — garbage activations for
    // every 512-channel conv under BN training.)
    conv2d_fwd_launch(dt_of(x), x.data_ptr(), w.data_ptr(), nullptr, bp,
                      y.data_ptr(), zero_page(x), nullptr, cs,
                      /*relu=*/false, cur_stream());
    return {y, at::empty({0}, x.options().dtype(at::kFloat))};
  }
  auto wt2 = at::empty({(int64_t)cs.Cout, (int64_t)cs.KH * cs.KW * cs.Cin},
                       w.options());
  transpose_w_fwd_launch(dt_of(x), w.data_ptr(), wt2.data_ptr(),
                         cs.KH * cs.KW, cs.Cin, cs.Cout, cur_stream());
  auto stats = at::zeros({2, (int64_t)cs.Cout},
                         x.options().dtype(at::kFloat));
  conv2d_fwd_launch(dt_of(x), x.data_ptr(), w.data_ptr(), wt2.data_ptr(), bp,
                    y.data_ptr(), zero_page(x), stats.data_ptr<float>(), cs,
                    /*relu=*/false, cur_stream());
  return {y, stats};
}

at::Tensor conv2d_dgrad(const at::Tensor& dy, const at::Tensor& w, int64_t H,
                        int64_t W, int64_t sh, int64_t sw, int64_t ph,
                        int64_t pw) {
  CHECK_IN(dy);
  CHECK_IN(w);
  int KH = w.size(0), KW = w.size(1), Cin = w.size(2), Cout = w.size(3);
  ConvShape cs;
  cs.N = dy.size(0);
  cs.H = H;
  cs.W = W;
  cs.Cin = Cin;
  cs.Cout = Cout;
  cs.KH = KH;
  cs.KW = KW;
  cs.SH = sh;
  cs.SW = sw;
  cs.PH = ph;
  cs.PW = pw;
  cs.OH = dy.size(1);
  cs.OW = dy.size(2);
  cs.init_fdiv();
  auto dx = at::empty({cs.N, H, W, Cin}, dy.options());
  if (conv2d_dgrad_wants_db(dt_of(dy), dy.data_ptr(), cs)) {
    // double-buffered path: weight as [Cin][(kh,kw,co)] k-contiguous rows
    auto w_t2d = at::empty({Cin, (int64_t)KH * KW * Cout}, w.options());
    transpose_w_dgrad_launch(dt_of(w), w.data_ptr(), w_t2d.data_ptr(), KH * KW,
                             Cin, Cout, cur_stream());
    conv2d_dgrad_launch(dt_of(dy), dy.data_ptr(), nullptr, w_t2d.data_ptr(),
                        dx.data_ptr(), zero_page(dy), cs, cur_stream());
    return dx;
  }
  // dgrad consumes the weight as B[(kh,kw,co)][ci]: transpose once per call
  auto w_t = at::empty({KH, KW, Cout, Cin}, w.options());
  transpose_w_launch(dt_of(w), w.data_ptr(), w_t.data_ptr(), KH, KW, Cin, Cout,
                     cur_stream());
  conv2d_dgrad_launch(dt_of(dy), dy.data_ptr(), w_t.data_ptr(), nullptr,
                      dx.data_ptr(), zero_page(dy), cs, cur_stream());
  return dx;
}

at::Tensor conv2d_wgrad(const at::Tensor& x, const at::Tensor& dy, int64_t KH,
                        int64_t KW, int64_t sh, int64_t sw, int64_t ph,
                        int64_t pw, bool out_in_dt) {
  CHECK_IN(x);
  CHECK_IN(dy);
  auto cs = conv_shape(x, x.size(3), dy.size(3), KH, KW, sh, sw, ph, pw);
  TORCH_CHECK(dy.size(1) == cs.OH && dy.size(2) == cs.OW, "wgrad shape");
  DT out_dt = out_in_dt ? dt_of(x) : DT::F32;
  auto dw = at::empty({KH, KW, cs.Cin, cs.Cout},
                      out_dt == DT::F32 ? x.options().dtype(at::kFloat)
                                        : x.options());
  int z = conv2d_wgrad_zsplits(cs);
  at::Tensor ws;
  float* wsp = nullptr;
  if (z > 1 || out_dt != DT::F32) {
    ws = at::empty({(int64_t)z * KH * KW * cs.Cin * cs.Cout},
                   x.options().dtype(at::kFloat));
    wsp = ws.data_ptr<float>();
  }
  conv2d_wgrad_launch(dt_of(x), x.data_ptr(), dy.data_ptr(), dw.data_ptr(),
                      out_dt, wsp, z, zero_page(x), cs, cur_stream());
  return dw;
}

// ---- batch norm ------------------------------------------------------------
std::vector<at::Tensor> bn_fwd_train(const at::Tensor& x,
                                     const at::Tensor& gamma,
                                     const at::Tensor& beta,
                                     const c10::optional<at::Tensor>& rmean,
                                     const c10::optional<at::Tensor>& rvar,
                                     double momentum, double eps, bool relu,
                                     double dropout_p, int64_t seed,
                                     const c10::optional<at::Tensor>& precomp,
                                     const c10::optional<at::Tensor>& ctr =
                                         c10::nullopt) {
  CHECK_IN(x);
  TORCH_CHECK(dropout_p == 0.0 || relu, "fused BN dropout requires relu");
  int C = x.size(-1);
  int64_t rows = x.numel() / C;
  auto mean = at::empty({C}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({C}, x.options().dtype(at::kFloat));
  auto y = at::empty_like(x);
  float* rm = rmean.has_value() ? rmean->data_ptr<float>() : nullptr;
  float* rv = rvar.has_value() ? rvar->data_ptr<float>() : nullptr;
  if (precomp.has_value()) {
    // sums already produced by the conv epilogue ({2, C}: sum, sumsq)
    TORCH_CHECK(precomp->numel() == 2 * C, "bad precomputed stats");
    const float* ps = precomp->data_ptr<float>();
    bn_finalize_launch(ps, ps + C, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), rm, rv, (float)momentum,
                       rows, C, (float)eps, cur_stream());
  } else {
    const int64_t wsn = bn_stats_ws_floats(dt_of(x), x.data_ptr(), rows, C);
    at::Tensor ws;
    float* wp = nullptr;
    if (wsn) {
      ws = at::empty({wsn}, x.options().dtype(at::kFloat));
      wp = ws.data_ptr<float>();
    }
    bn_stats_launch(dt_of(x), x.data_ptr(), mean.data_ptr<float>(),
                    invstd.data_ptr<float>(), rm, rv, (float)momentum, wp,
                    rows, C, (float)eps, cur_stream());
  }
  if (dropout_p > 0.0)
    bn_apply_drop_launch(dt_of(x), x.data_ptr(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                         beta.data_ptr<float>(), y.data_ptr(), rows, C,
                         (float)dropout_p, (uint64_t)seed,
                         ctr.has_value() ? ctr->data_ptr<int64_t>() : nullptr,
                         cur_stream());
  else
    bn_apply_launch(dt_of(x), x.data_ptr(), mean.data_ptr<float>(),
                    invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                    beta.data_ptr<float>(), y.data_ptr(), rows, C, relu,
                    cur_stream());
  return {y, mean, invstd};
}

at::Tensor bn_fwd_infer(const at::Tensor& x, const at::Tensor& gamma,
                        const at::Tensor& beta, const at::Tensor& rmean,
                        const at::Tensor& rvar, double eps, bool relu) {
  CHECK_IN(x);
  int C = x.size(-1);
  int64_t rows = x.numel() / C;
  auto y = at::empty_like(x);
  bn_infer_launch(dt_of(x), x.data_ptr(), rmean.data_ptr<float>(),
                  rvar.data_ptr<float>(), gamma.data_ptr<float>(),
                  beta.data_ptr<float>(), y.data_ptr(), rows, C, (float)eps,
                  relu, cur_stream());
  return y;
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& x, const at::Tensor& dy,
                               const at::Tensor& gamma, const at::Tensor& mean,
                               const at::Tensor& invstd,
                               const c10::optional<at::Tensor>& y_relu,
                               double dy_scale,
                               const c10::optional<at::Tensor>& resid =
                                   c10::nullopt) {
  CHECK_IN(x);
  CHECK_IN(dy);
  const void* resp = nullptr;
  if (resid.has_value()) {
    TORCH_CHECK(resid->is_contiguous() &&
                resid->scalar_type() == x.scalar_type() &&
                resid->numel() == x.numel(), "bn_bwd resid mismatch");
    resp = resid->data_ptr();
  }
  int C = x.size(-1);
  int64_t rows = x.numel() / C;
  const int64_t wsn = bn_bwd_ws_floats(dt_of(x), x.data_ptr(),
                                       dy.data_ptr(), rows, C);
  // slab path (vec) needs no zero-init; scalar fallback atomically
  // accumulates into zeroed sums
  auto sums = wsn ? at::empty({2, C}, x.options().dtype(at::kFloat))
                  : at::zeros({2, C}, x.options().dtype(at::kFloat));
  auto sum_dy = sums[0];
  auto sum_dy_xhat = sums[1];
  at::Tensor ws;
  float* wp = nullptr;
  if (wsn) {
    ws = at::empty({wsn}, x.options().dtype(at::kFloat));
    wp = ws.data_ptr<float>();
  }
  auto dx = at::empty_like(x);
  const void* yr = y_relu.has_value() ? y_relu->data_ptr() : nullptr;
  bn_bwd_reduce_launch(dt_of(x), x.data_ptr(), dy.data_ptr(), yr,
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       sum_dy.data_ptr<float>(), sum_dy_xhat.data_ptr<float>(),
                       wp, rows, C, (float)dy_scale, cur_stream());
  bn_bwd_apply_launch(dt_of(x), x.data_ptr(), dy.data_ptr(), yr,
                      mean.data_ptr<float>(), invstd.data_ptr<float>(),
                      gamma.data_ptr<float>(), sum_dy.data_ptr<float>(),
                      sum_dy_xhat.data_ptr<float>(), dx.data_ptr(), resp,
                      rows, C, (float)dy_scale, cur_stream());
  // dgamma = sum_dy_xhat, dbeta = sum_dy (fp32, matching fp32 gamma/beta)
  return {dx, sum_dy_xhat, sum_dy};
}

// ---- pooling ---------------------------------------------------------------
static PoolShape pool_shape(const at::Tensor& x, int KH, int KW, int SH, int SW,
                            int PH, int PW) {
  PoolShape ps;
  ps.N = x.size(0);
  ps.H = x.size(1);
  ps.W = x.size(2);
  ps.C = x.size(3);
  ps.KH = KH;
  ps.KW = KW;
  ps.SH = SH;
  ps.SW = SW;
  ps.PH = PH;
  ps.PW = PW;
  ps.OH = (ps.H + 2 * PH - KH) / SH + 1;
  ps.OW = (ps.W + 2 * PW - KW) / SW + 1;
  return ps;
}

std::vector<at::Tensor> maxpool_fwd(const at::Tensor& x, int64_t kh, int64_t kw,
                                    int64_t sh, int64_t sw, int64_t ph,
                                    int64_t pw) {
  CHECK_IN(x);
  auto ps = pool_shape(x, kh, kw, sh, sw, ph, pw);
  auto y = at::empty({ps.N, ps.OH, ps.OW, ps.C}, x.options());
  auto idx = at::empty({ps.N, ps.OH, ps.OW, ps.C}, x.options().dtype(at::kInt));
  maxpool_fwd_launch(dt_of(x), x.data_ptr(), y.data_ptr(), idx.data_ptr<int>(),
                     ps, cur_stream());
  return {y, idx};
}

at::Tensor maxpool_bwd(const at::Tensor& dy, const at::Tensor& idx, int64_t H,
                       int64_t W) {
  CHECK_IN(dy);
  PoolShape ps;
  ps.N = dy.size(0);
  ps.OH = dy.size(1);
  ps.OW = dy.size(2);
  ps.C = dy.size(3);
  ps.H = H;
  ps.W = W;
  auto dxf = at::zeros({ps.N, H, W, ps.C}, dy.options().dtype(at::kFloat));
  maxpool_bwd_launch(dt_of(dy), dy.data_ptr(), idx.data_ptr<int>(),
                     dxf.data_ptr<float>(), ps, cur_stream());
  if (dy.scalar_type() == at::kFloat) return dxf;
  auto dx = at::empty({ps.N, H, W, ps.C}, dy.options());
  cast_f32_launch(dt_of(dy), dxf.data_ptr<float>(), dx.data_ptr(), dx.numel(),
                  cur_stream());
  return dx;
}

at::Tensor avgpool_fwd(const at::Tensor& x, int64_t kh, int64_t kw, int64_t sh,
                       int64_t sw, int64_t ph, int64_t pw) {
  CHECK_IN(x);
  auto ps = pool_shape(x, kh, kw, sh, sw, ph, pw);
  auto y = at::empty({ps.N, ps.OH, ps.OW, ps.C}, x.options());
  avgpool_fwd_launch(dt_of(x), x.data_ptr(), y.data_ptr(), ps, cur_stream());
  return y;
}

at::Tensor avgpool_bwd(const at::Tensor& dy, int64_t H, int64_t W, int64_t kh,
                       int64_t kw, int64_t sh, int64_t sw, int64_t ph,
                       int64_t pw) {
  CHECK_IN(dy);
  PoolShape ps;
  ps.N = dy.size(0);
  ps.OH = dy.size(1);
  ps.OW = dy.size(2);
  ps.C = dy.size(3);
  ps.H = H;
  ps.W = W;
  ps.KH = kh;
  ps.KW = kw;
  ps.SH = sh;
  ps.SW = sw;
  ps.PH = ph;
  ps.PW = pw;
  auto dx = at::empty({ps.N, H, W, ps.C}, dy.options());
  avgpool_bwd_launch(dt_of(dy), dy.data_ptr(), dx.data_ptr(), ps, cur_stream());
  return dx;
}

// ---- loss ------------------------------------------------------------------
std::vector<at::Tensor> ce_fwd(const at::Tensor& logits,
                               const at::Tensor& targets) {
  CHECK_IN(logits);
  CHECK_IN(targets);
  int64_t rows = logits.size(0);
  int cols = logits.size(1);
  auto loss = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  ce_fwd_launch(dt_of(logits), logits.data_ptr(), targets.data_ptr<int64_t>(),
                loss.data_ptr<float>(), lse.data_ptr<float>(), rows, cols,
                cur_stream());
  return {loss, lse};
}

at::Tensor ce_bwd(const at::Tensor& logits, const at::Tensor& targets,
                  const at::Tensor& lse, const at::Tensor& dloss) {
  CHECK_IN(logits);
  auto dl = at::empty_like(logits);
  ce_bwd_launch(dt_of(logits), logits.data_ptr(), targets.data_ptr<int64_t>(),
                lse.data_ptr<float>(), dloss.contiguous().data_ptr<float>(),
                dl.data_ptr(), logits.size(0), logits.size(1), cur_stream());
  return dl;
}

at::Tensor ptloss_fwd(const at::Tensor& pred, const at::Tensor& tgt,
                      int64_t kind, double delta) {
  CHECK_IN(pred);
  CHECK_IN(tgt);
  TORCH_CHECK(pred.sizes() == tgt.sizes(), "pred/target shape mismatch");
  auto out = at::zeros({}, pred.options().dtype(at::kFloat));
  ptloss_fwd_launch(dt_of(pred), pred.data_ptr(), tgt.data_ptr(),
                    out.data_ptr<float>(), pred.numel(), (int)kind,
                    (float)delta, cur_stream());
  return out;
}

at::Tensor ptloss_bwd(const at::Tensor& pred, const at::Tensor& tgt,
                      const at::Tensor& dloss, int64_t kind, double delta) {
  CHECK_IN(pred);
  auto dpred = at::empty_like(pred);
  ptloss_bwd_launch(dt_of(pred), pred.data_ptr(), tgt.data_ptr(),
                    dloss.contiguous().data_ptr<float>(), dpred.data_ptr(),
                    pred.numel(), (int)kind, (float)delta, cur_stream());
  return dpred;
}

// ---- group norm ------------------------------------------------------------
// x [N, ..., C] NHWC-contiguous; gamma/beta fp32 [C]
std::vector<at::Tensor> gn_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta, int64_t groups,
                               double eps) {
  CHECK_IN(x);
  const int C = x.size(-1);
  const int64_t N = x.size(0);
  const int64_t HW = x.numel() / (N * C);
  TORCH_CHECK(C % groups == 0, "C % groups != 0");
  auto y = at::empty_like(x);
  auto mean = at::empty({N * groups}, x.options().dtype(at::kFloat));
  auto invstd = at::empty_like(mean);
  gn_fwd_launch(dt_of(x), x.data_ptr(), gamma.data_ptr<float>(),
                beta.data_ptr<float>(), y.data_ptr(), mean.data_ptr<float>(),
                invstd.data_ptr<float>(), N, HW, C, (int)groups, (float)eps,
                cur_stream());
  return {y, mean, invstd};
}

std::vector<at::Tensor> gn_bwd(const at::Tensor& x, const at::Tensor& dy,
                               const at::Tensor& mean, const at::Tensor& invstd,
                               const at::Tensor& gamma, int64_t groups) {
  CHECK_IN(x);
  CHECK_IN(dy);
  const int C = x.size(-1);
  const int64_t N = x.size(0);
  const int64_t HW = x.numel() / (N * C);
  auto dx = at::empty_like(x);
  auto s1 = at::empty({N * groups}, x.options().dtype(at::kFloat));
  auto s2 = at::empty_like(s1);
  auto dgamma = at::zeros({C}, x.options().dtype(at::kFloat));
  auto dbeta = at::zeros({C}, x.options().dtype(at::kFloat));
  gn_bwd_launch(dt_of(x), x.data_ptr(), dy.data_ptr(), mean.data_ptr<float>(),
                invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                s1.data_ptr<float>(), s2.data_ptr<float>(), dx.data_ptr(),
                dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), N, HW, C,
                (int)groups, cur_stream());
  return {dx, dgamma, dbeta};
}

// ---- layer norm ------------------------------------------------------------
std::vector<at::Tensor> ln_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta, double eps) {
  CHECK_IN(x);
  int cols = x.size(-1);
  int64_t rows = x.numel() / cols;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({rows}, x.options().dtype(at::kFloat));
  ln_fwd_launch(dt_of(x), x.data_ptr(), gamma.data_ptr<float>(),
                beta.data_ptr<float>(), y.data_ptr(), mean.data_ptr<float>(),
                invstd.data_ptr<float>(), rows, cols, (float)eps, cur_stream());
  return {y, mean, invstd};
}

std::vector<at::Tensor> ln_bwd(const at::Tensor& x, const at::Tensor& dy,
                               const at::Tensor& gamma, const at::Tensor& mean,
                               const at::Tensor& invstd,
                               const c10::optional<at::Tensor>& resid =
                                   c10::nullopt) {
  CHECK_IN(x);
  CHECK_IN(dy);
  const void* rp = nullptr;
  if (resid.has_value()) {
    TORCH_CHECK(resid->is_contiguous() &&
                resid->scalar_type() == x.scalar_type() &&
                resid->numel() == x.numel(), "ln_bwd resid mismatch");
    rp = resid->data_ptr();
  }
  int cols = x.size(-1);
  int64_t rows = x.numel() / cols;
  auto dx = at::empty_like(x);
  const int64_t wsn = ln_bwd_ws_floats(dt_of(x), x.data_ptr(), dy.data_ptr(),
                                       rows, cols);
  const bool vec = (dt_of(x) == DT::F32 ? cols % 4 == 0 : cols % 8 == 0) &&
                   ((uintptr_t)x.data_ptr() & 15) == 0 &&
                   ((uintptr_t)dy.data_ptr() & 15) == 0;
  // vec path single-writes (or slab-folds) the [2, cols] buffer; the
  // scalar fallback atomically accumulates into zeros
  auto sums = vec ? at::empty({2, cols}, x.options().dtype(at::kFloat))
                  : at::zeros({2, cols}, x.options().dtype(at::kFloat));
  at::Tensor ws;
  float* wp = nullptr;
  if (wsn) {
    ws = at::empty({wsn}, x.options().dtype(at::kFloat));
    wp = ws.data_ptr<float>();
  }
  ln_bwd_launch(dt_of(x), x.data_ptr(), dy.data_ptr(), gamma.data_ptr<float>(),
                mean.data_ptr<float>(), invstd.data_ptr<float>(), dx.data_ptr(),
                rp, sums.data_ptr<float>(), wp, rows, cols, cur_stream());
  return {dx, sums[0], sums[1]};
}

// ---- embedding -------------------------------------------------------------
at::Tensor embedding_fwd(const at::Tensor& ids, const at::Tensor& table) {
  CHECK_IN(ids);
  CHECK_IN(table);
  auto sizes = ids.sizes().vec();
  sizes.push_back(table.size(1));
  auto y = at::empty(sizes, table.options());
  embedding_fwd_launch(dt_of(table), ids.data_ptr<int64_t>(), table.data_ptr(),
                       y.data_ptr(), ids.numel(), table.size(1), cur_stream());
  return y;
}

at::Tensor embedding_bwd(const at::Tensor& ids, const at::Tensor& dy,
                         int64_t rows) {
  CHECK_IN(ids);
  CHECK_IN(dy);
  int dim = dy.size(-1);
  auto dt = at::zeros({rows, dim}, dy.options().dtype(at::kFloat));
  embedding_bwd_launch(dt_of(dy), ids.data_ptr<int64_t>(), dy.data_ptr(),
                       dt.data_ptr<float>(), ids.numel(), dim, cur_stream());
  return dt;
}

// ---- flash attention -------------------------------------------------------
// q/k/v/o/dout are [B,H,S,D] views that may be non-contiguous as long as
// d stays contiguous: BHSD-dense tensors, transposed views of BSHD
// buffers, or head-slices of one merged [B,S,3*H*D] QKV buffer. The
// kernels read rows through {row, head, batch} stride tuples, so no
// transpose/contiguous copies happen anywhere in the attention path.
static bool attn_view_ok(const at::Tensor& t, bool need_align) {
  if (t.stride(3) != 1) return false;
  if (!need_align) return true;  // scalar reads/writes: any row alignment
  // glds stages 16-B vectors: rows must stay 16-B aligned across s and h
  return t.stride(2) % 8 == 0 && t.stride(1) % 8 == 0 &&
         ((uintptr_t)t.data_ptr() & 15) == 0;
}

static std::array<int64_t, 3> attn_strides(const at::Tensor& t) {
  return {t.stride(2), t.stride(1), t.stride(0)};
}

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 bool causal) {
  TORCH_CHECK(q.is_cuda(), "q must be on GPU");
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "flash attention is bf16");
  TORCH_CHECK(q.dim() == 4, "q must be [B,H,S,D]");
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "head dim must be 64 or 128, got ", D);
  const bool strided_ok = attn_view_ok(q, true) && attn_view_ok(k, true) &&
                          attn_view_ok(v, true) &&
                          q.strides() == k.strides() &&
                          k.strides() == v.strides();
  if (!strided_ok) {
    q = q.contiguous(); k = k.contiguous(); v = v.contiguous();
  }
  // o lands BSHD-contiguous so the caller's transpose+reshape to
  // [B,S,H*D] is a free view
  auto o_buf = at::empty({B, S, (int64_t)H * D}, q.options());
  auto o = o_buf.view({B, S, H, D}).permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  float scale = 1.0f / std::sqrt((float)D);
  attn_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o_buf.data_ptr(),
                  lse.data_ptr<float>(), zero_page(q), B, H, S, D,
                  attn_strides(q).data(), attn_strides(o).data(), causal,
                  scale, cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 const at::Tensor& o, at::Tensor dout,
                                 const at::Tensor& lse, bool causal,
                                 c10::optional<at::Tensor> dq_out,
                                 c10::optional<at::Tensor> dk_out,
                                 c10::optional<at::Tensor> dv_out) {
  TORCH_CHECK(q.is_cuda() && dout.is_cuda(), "attn_bwd wants GPU tensors");
  int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const bool in_ok = attn_view_ok(q, true) && attn_view_ok(k, true) &&
                     attn_view_ok(v, true) && q.strides() == k.strides() &&
                     k.strides() == v.strides();
  if (!in_ok) { q = q.contiguous(); k = k.contiguous(); v = v.contiguous(); }
  if (!attn_view_ok(dout, true)) dout = dout.contiguous();
  TORCH_CHECK(attn_view_ok(o, false), "o must be d-contiguous");
  at::Tensor dq, dk, dv;
  if (dq_out.has_value()) {
    TORCH_CHECK(dk_out.has_value() && dv_out.has_value(),
                "provide all three grad views or none");
    // caller-provided grad views (e.g. three slices of one merged dQKV
    // buffer: the fused-QKV backward writes it in place, no cat/pad)
    dq = *dq_out; dk = *dk_out; dv = *dv_out;
    TORCH_CHECK(dq.strides() == dk.strides() && dk.strides() == dv.strides()
                    && dq.stride(3) == 1,
                "grad views must share a d-contiguous stride tuple");
  } else {
    auto mk = [&]() {
      return at::empty({B, S, (int64_t)H * D}, q.options())
          .view({B, S, H, D}).permute({0, 2, 1, 3});
    };
    dq = mk(); dk = mk(); dv = mk();
  }
  auto dqw = at::zeros({B, H, S, D}, q.options().dtype(at::kFloat));
  auto di = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  float scale = 1.0f / std::sqrt((float)D);
  attn_bwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                  dout.data_ptr(), lse.data_ptr<float>(), di.data_ptr<float>(),
                  dqw.data_ptr<float>(), dq.data_ptr(), dk.data_ptr(),
                  dv.data_ptr(), zero_page(q), B, H, S, D,
                  attn_strides(q).data(), attn_strides(o).data(),
                  attn_strides(dout).data(), attn_strides(dq).data(), causal,
                  scale, cur_stream());
  return {dq, dk, dv};
}

// ---- optimizers ------------------------------------------------------------
void sgd_step(at::Tensor param, at::Tensor master, const at::Tensor& grad,
              const c10::optional<at::Tensor>& momentum_buf, double lr,
              double momentum, double weight_decay, bool nesterov) {
  CHECK_IN(param);
  bool has_master = master.data_ptr() != param.data_ptr();
  float* mb = momentum_buf.has_value() ? momentum_buf->data_ptr<float>()
                                       : nullptr;
  sgd_step_launch(dt_of(param), grad.data_ptr(), dt_of(grad), param.data_ptr(),
                  master.data_ptr<float>(), mb, param.numel(), (float)lr,
                  (float)momentum, (float)weight_decay, nesterov, has_master,
                  cur_stream());
}

void adam_step(at::Tensor param, at::Tensor master, const at::Tensor& grad,
               at::Tensor m, at::Tensor v, int64_t step, double lr,
               double beta1, double beta2, double eps, double weight_decay,
               bool adamw,
               const c10::optional<at::Tensor>& step_dev = c10::nullopt) {
  CHECK_IN(param);
  bool has_master = master.data_ptr() != param.data_ptr();
  const int64_t* sd =
      step_dev.has_value() ? step_dev->data_ptr<int64_t>() : nullptr;
  adam_step_launch(dt_of(param), grad.data_ptr(), dt_of(grad), param.data_ptr(),
                   master.data_ptr<float>(), m.data_ptr<float>(),
                   v.data_ptr<float>(), param.numel(), (int)step, sd,
                   (float)lr, (float)beta1, (float)beta2, (float)eps,
                   (float)weight_decay, adamw, has_master, cur_stream());
}

void adam_step_mt(const at::Tensor& desc, const at::Tensor& chunks,
                  int64_t dt_p, int64_t dt_g, bool has_master, int64_t step,
                  double lr, double beta1, double beta2, double eps,
                  double weight_decay, bool adamw,
                  const c10::optional<at::Tensor>& step_dev = c10::nullopt) {
  CHECK_IN(desc);
  CHECK_IN(chunks);
  const int64_t* sd =
      step_dev.has_value() ? step_dev->data_ptr<int64_t>() : nullptr;
  adam_mt_launch((DT)dt_p, (DT)dt_g, has_master,
                 desc.data_ptr<int64_t>(), chunks.data_ptr<int64_t>(),
                 (int)chunks.numel(), (int)step, sd, (float)lr, (float)beta1,
                 (float)beta2, (float)eps, (float)weight_decay, adamw,
                 cur_stream());
}

// ---- batched MFMA gemm -----------------------------------------------------
// a: logical [batch, M, K]; b: logical [batch, K, N]. Either operand may
// be a transpose VIEW (unit stride in dim -2 instead of -1) — the layout
// is read off the strides, the tensor-level analog of
// cublasGemmStridedBatchedEx's opA/opB (reference src/math/cuda/gemm.cu:84).
at::Tensor bmm(const at::Tensor& a_in, const at::Tensor& b_in) {
  TORCH_CHECK(a_in.dim() == 3 && b_in.dim() == 3, "bmm expects 3-D tensors");
  TORCH_CHECK(a_in.is_cuda() && b_in.is_cuda(), "bmm expects GPU tensors");
  TORCH_CHECK(a_in.scalar_type() == b_in.scalar_type(), "bmm dtype mismatch");
  auto norm = [](const at::Tensor& t) {
    return (t.stride(2) == 1 || t.stride(1) == 1) ? t : t.contiguous();
  };
  at::Tensor a = norm(a_in), b = norm(b_in);
  const int64_t batch = a.size(0), M = a.size(1), K = a.size(2), N = b.size(2);
  TORCH_CHECK(b.size(0) == batch && b.size(1) == K, "bmm shape mismatch");
  TORCH_CHECK(batch <= 65535, "bmm batch > 65535");
  bool ta, tb;
  int64_t lda, ldb;
  if (a.stride(2) == 1) { ta = false; lda = a.stride(1); }
  else { ta = true; lda = a.stride(2); }
  if (b.stride(2) == 1) { tb = false; ldb = b.stride(1); }
  else { tb = true; ldb = b.stride(2); }
  TORCH_CHECK(lda <= INT32_MAX && ldb <= INT32_MAX && M <= INT32_MAX &&
              N <= INT32_MAX && K <= INT32_MAX, "bmm dims exceed int32");
  auto c = at::empty({batch, M, N}, a.options());
  bmm_launch(dt_of(a), a.data_ptr(), b.data_ptr(), c.data_ptr(), zero_page(a),
             (int)batch, (int)M, (int)N, (int)K, (int)lda, (int)ldb, (int)N,
             a.stride(0), b.stride(0), M * N, ta, tb, cur_stream());
  return c;
}

// ---- scaled / causal softmax ----------------------------------------------
at::Tensor smax_fwd(const at::Tensor& x, int64_t mrows, int64_t qoff,
                    double scale, bool causal) {
  CHECK_IN(x);
  const int cols = x.size(-1);
  const int64_t rows = x.numel() / cols;
  auto y = at::empty_like(x);
  smax_fwd_launch(dt_of(x), x.data_ptr(), y.data_ptr(), rows, cols,
                  (int)mrows, (int)qoff, (float)scale, causal, cur_stream());
  return y;
}

at::Tensor smax_bwd(const at::Tensor& p, const at::Tensor& dy, double scale) {
  CHECK_IN(p);
  CHECK_IN(dy);
  const int cols = p.size(-1);
  const int64_t rows = p.numel() / cols;
  auto dx = at::empty_like(p);
  smax_bwd_launch(dt_of(p), p.data_ptr(), dy.data_ptr(), dx.data_ptr(), rows,
                  cols, (float)scale, cur_stream());
  return dx;
}

// ---- fused decode attention ------------------------------------------------
// q [BH, D] bf16; k/v [BH, cap, D] bf16 full cache buffers. Either pos
// (device int64 [1], len = pos+1 — graph-capturable) or len gives the
// live KV length.
at::Tensor attn_decode(const at::Tensor& q, const at::Tensor& k,
                       const at::Tensor& v,
                       const c10::optional<at::Tensor>& pos, int64_t len,
                       double scale) {
  CHECK_IN(q);
  CHECK_IN(k);
  CHECK_IN(v);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn_decode is bf16-only");
  const int BH = q.size(0), D = q.size(1), cap = k.size(1);
  TORCH_CHECK(D == 64 || D == 128, "attn_decode supports D in {64,128}");
  TORCH_CHECK(k.size(0) == BH && k.size(2) == D && v.sizes() == k.sizes());
  // split KV so the grid approaches the CU count (decode is latency-
  // bound; 12 heads x 1 split measured 22us/call): static from (BH, cap)
  // for hipGraph capture
  int splits = std::max<int>(1, std::min<int>(16, 256 / BH));
  auto po = at::empty({(int64_t)BH * splits, D},
                      q.options().dtype(at::kFloat));
  auto ml = at::empty({(int64_t)BH * splits, 2},
                      q.options().dtype(at::kFloat));
  auto out = at::empty_like(q);
  const int64_t* pos_ptr = nullptr;
  if (pos.has_value()) {
    TORCH_CHECK(pos->scalar_type() == at::kLong && pos->is_cuda());
    pos_ptr = pos->data_ptr<int64_t>();
  }
  attn_decode_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                     po.data_ptr<float>(), ml.data_ptr<float>(),
                     out.data_ptr(), pos_ptr, (int)len, BH, cap, D, splits,
                     (float)scale, cur_stream());
  return out;
}

namespace imgcodec {
at::Tensor decode_image(const py::bytes& data);  // imagecodec.cpp (host-only)
}

}  // namespace tnn

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("act_fwd", &tnn::act_fwd);
  m.def("act_bwd", &tnn::act_bwd);
  m.def("relu_bwd_mask", &tnn::relu_bwd_mask);
  m.def("add_act_fwd", &tnn::add_act_fwd);
  m.def("add_act_bwd", &tnn::add_act_bwd);
  m.def("dropout_fwd", &tnn::dropout_fwd, py::arg("x"), py::arg("p"),
        py::arg("seed"), py::arg("ctr") = c10::nullopt);
  m.def("dropout_bwd", &tnn::dropout_bwd);
  m.def("colsum", &tnn::colsum);
  m.def("gemm", &tnn::gemm, py::arg("a"), py::arg("b"), py::arg("bias"),
        py::arg("act_kind"), py::arg("residual") = c10::nullopt,
        py::arg("ln_gamma") = c10::nullopt, py::arg("ln_beta") = c10::nullopt,
        py::arg("ln_eps") = 1e-5);
  m.def("gemm_nt", &tnn::gemm_nt);
  m.def("gemm_tn", &tnn::gemm_tn);
  m.def("mfma_selftest", &tnn::mfma_selftest);
  m.def("mfma_selftest_f32", &tnn::mfma_selftest_f32);
  m.def("conv2d_fwd", &tnn::conv2d_fwd);
  m.def("conv2d_fwd_stats", &tnn::conv2d_fwd_stats);
  m.def("conv2d_dgrad", &tnn::conv2d_dgrad);
  m.def("conv2d_wgrad", &tnn::conv2d_wgrad);
  m.def("bn_fwd_train", &tnn::bn_fwd_train, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("rmean"), py::arg("rvar"),
        py::arg("momentum"), py::arg("eps"), py::arg("relu"),
        py::arg("dropout_p"), py::arg("seed"), py::arg("precomp"),
        py::arg("ctr") = c10::nullopt);
  m.def("bn_fwd_infer", &tnn::bn_fwd_infer);
  m.def("bn_bwd", &tnn::bn_bwd, py::arg("x"), py::arg("dy"),
        py::arg("gamma"), py::arg("mean"), py::arg("invstd"),
        py::arg("y_relu"), py::arg("dy_scale"),
        py::arg("resid") = py::none());
  m.def("maxpool_fwd", &tnn::maxpool_fwd);
  m.def("maxpool_bwd", &tnn::maxpool_bwd);
  m.def("avgpool_fwd", &tnn::avgpool_fwd);
  m.def("avgpool_bwd", &tnn::avgpool_bwd);
  m.def("ce_fwd", &tnn::ce_fwd);
  m.def("ce_bwd", &tnn::ce_bwd);
  m.def("ptloss_fwd", &tnn::ptloss_fwd);
  m.def("ptloss_bwd", &tnn::ptloss_bwd);
  m.def("gn_fwd", &tnn::gn_fwd);
  m.def("gn_bwd", &tnn::gn_bwd);
  m.def("ln_fwd", &tnn::ln_fwd);
  m.def("ln_bwd", &tnn::ln_bwd, py::arg("x"), py::arg("dy"),
        py::arg("gamma"), py::arg("mean"), py::arg("invstd"),
        py::arg("resid") = py::none());
  m.def("embedding_fwd", &tnn::embedding_fwd);
  m.def("embedding_bwd", &tnn::embedding_bwd);
  m.def("attn_fwd", &tnn::attn_fwd);
  m.def("attn_bwd", &tnn::attn_bwd, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("o"), py::arg("dout"), py::arg("lse"),
        py::arg("causal"), py::arg("dq_out") = py::none(),
        py::arg("dk_out") = py::none(), py::arg("dv_out") = py::none());
  m.def("bmm", &tnn::bmm);
  m.def("smax_fwd", &tnn::smax_fwd);
  m.def("smax_bwd", &tnn::smax_bwd);
  m.def("attn_decode", &tnn::attn_decode);
  m.def("sgd_step", &tnn::sgd_step);
  m.def("adam_step", &tnn::adam_step, py::arg("param"), py::arg("master"),
        py::arg("grad"), py::arg("m"), py::arg("v"), py::arg("step"),
        py::arg("lr"), py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("weight_decay"), py::arg("adamw"),
        py::arg("step_dev") = c10::nullopt);
  m.def("adam_step_mt", &tnn::adam_step_mt, py::arg("desc"),
        py::arg("chunks"), py::arg("dt_p"), py::arg("dt_g"),
        py::arg("has_master"), py::arg("step"), py::arg("lr"),
        py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("weight_decay"), py::arg("adamw"),
        py::arg("step_dev") = c10::nullopt);
  m.def("decode_image", &tnn::imgcodec::decode_image);
}
