// Launcher declarations shared between .hip translation units and the
// torch bindings. All functions are synchronous-on-stream (enqueue only).
#pragma once

#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdint>

namespace tnn {

enum class DT { F32, BF16 };

// ---- elementwise.hip -------------------------------------------------------
void act_fwd_launch(DT dt, const void* x, void* y, int64_t n, int kind,
                    hipStream_t s);
void act_bwd_launch(DT dt, const void* dy, const void* x, const void* y,
                    void* dx, int64_t n, int kind, hipStream_t s);
void add_act_fwd_launch(DT dt, const void* a, const void* b, void* y,
                        int64_t n, int kind, hipStream_t s);
void add_act_bwd_launch(DT dt, const void* dy, const void* y, void* g,
                        int64_t n, int kind, hipStream_t s);
void relu_bwd_mask_launch(DT dt, const void* dy, const void* y, void* dx,
                          int64_t n, hipStream_t s);
void dropout_fwd_launch(DT dt, const void* x, void* y, uint8_t* mask,
                        int64_t n, float p, uint64_t seed,
                        const int64_t* ctr, hipStream_t s);
void dropout_bwd_launch(DT dt, const void* dy, const uint8_t* mask, void* dx,
                        int64_t n, float p, hipStream_t s);
int colsum_ws_slices(DT dt, const void* x, int64_t rows, int64_t cols);
void colsum_launch(DT dt, const void* x, void* out_f32, float* ws,
                   int rslices, int64_t rows, int64_t cols, hipStream_t s);
void slab_fin_launch(const float* ws, float* out, int slices, int c2,
                     hipStream_t s);
void cast_f32_launch(DT dt_out, const float* x, void* y, int64_t n,
                     hipStream_t s);

// ---- gemm.hip --------------------------------------------------------------
void gemm_launch(DT dt, const void* a, const void* b, const void* bias,
                 void* c, const void* zero16, const void* resid, int M, int N,
                 int K, bool trans_b, int act_kind, hipStream_t s);
int gemv_nn_ksplits(int N, int K);
int gemv_nn1_nsplit(int N, int K);
void gemv_nn1_launch(DT dt, const void* x, const void* b, const void* bias,
                     const void* res, void* y, float* part, int nsplit,
                     const float* ln_g, const float* ln_b, float ln_eps, int N,
                     int K, int act_kind, hipStream_t s);
constexpr int GEMV1_MAX_K_DECL = 8192;
void gemv_nn_launch(DT dt, const void* x, const void* b, const void* bias,
                    void* y, float* ws, int ks, int N, int K, int act_kind,
                    hipStream_t s);
int gemm_nt_zsplits(DT dt, int M, int N, int K);
int gemm_nt_fsplits(DT dt, int M, int N, int K);
int gemm_nt32_zsplits(int M, int N, int K);
void gemm_nt32_launch(const void* a, const void* b, float* ws,
                      const void* bias_f32, const void* bias_t,
                      const void* resid, void* c, int z, const void* zero16,
                      int M, int N, int K, int act_kind, hipStream_t s);
void gemm_nt_zf_launch(DT dt, const void* a, const void* b, float* ws,
                       const void* bias_f32_or_t, const void* resid, void* c,
                       int z, const void* zero16, int M, int N, int K,
                       int act_kind, hipStream_t s);
void gemm_nt_z_launch(DT dt, const void* a, const void* b, float* ws,
                      void* c_out, DT out_dt, int z, const void* zero16,
                      int M, int N, int K, hipStream_t s);
int gemm_tn_zsplits(int M, int N, int K);
void gemm_tn_launch(DT dt, const void* a, const void* b, void* c_out,
                    DT out_dt, float* ws, int z, const void* zero16, int M,
                    int N, int K, hipStream_t s);
void splitk_reduce_launch(const float* ws, void* out, DT out_dt, int z,
                          int64_t n, hipStream_t s);
void mfma_selftest_launch(const void* a_bf16, const void* b_bf16, float* d,
                          hipStream_t s);
void mfma_selftest_f32_launch(const float* a, const float* b, float* d,
                              hipStream_t s);

// ---- conv2d.hip ------------------------------------------------------------
// shift-based fast division for the im2col address math (every hot divisor
// — OW, OH*OW, Cin, Cout — is a power of two in the CNN zoo; others fall
// back to hardware division)
struct IDiv {
  int d = 1;
  int lg = 0;        // log2(d) if power of two, else -1
  int sh = 0;        // magic shift (non-pow2), 0 = use plain division
  unsigned long long magic = 0;  // ceil(2^sh / d) + exactness headroom
  void set(int dd) {
    d = dd;
    lg = (dd & (dd - 1)) == 0 ? __builtin_ctz(dd) : -1;
    magic = 0;
    sh = 0;
    // mul-shift reciprocal, exact for 0 <= x < 2^26 and 1 < d < 2^16
    // (error term x/2^sh < 1/d strictly; covers every zoo shape --
    // e.g. 224x224 ImageNet at batch 1024). Larger divisors fall back
    // to hardware division.
    if (lg < 0 && dd > 1 && dd < 65536) {
      sh = dd < 1024 ? 36 : 42;
      magic = ((1ull << sh) + dd - 1) / dd;
    }
  }
};

struct ConvShape {
  int N, H, W, Cin, Cout, KH, KW, SH, SW, PH, PW, OH, OW;
  IDiv d_ohow, d_ow, d_cin, d_hw, d_w, d_cout;
  int mkw16;  // ceil-reciprocal for kidx / KW: exact while kidx * KW < 2^16
  void init_fdiv() {
    d_ohow.set(OH * OW);
    d_ow.set(OW);
    d_cin.set(Cin);
    d_hw.set(H * W);
    d_w.set(W);
    d_cout.set(Cout);
    mkw16 = 65536 / KW + 1;
  }
};
bool conv2d_fwd_wants_db(DT dt, const void* x, const ConvShape& cs);
void transpose_w_fwd_launch(DT dt, const void* w, void* w_t2, int KHW, int Cin,
                            int Cout, hipStream_t s);
void conv2d_fwd_launch(DT dt, const void* x, const void* w, const void* w_t2,
                       const void* bias, void* y, const void* zero16,
                       float* stats, const ConvShape& cs, bool relu,
                       hipStream_t s);
bool conv2d_dgrad_wants_db(DT dt, const void* dy, const ConvShape& cs);
void transpose_w_dgrad_launch(DT dt, const void* w, void* w_t2d, int KHW,
                              int Cin, int Cout, hipStream_t s);
void conv2d_dgrad_launch(DT dt, const void* dy, const void* w_t,
                         const void* w_t2d, void* dx, const void* zero16,
                         const ConvShape& cs, hipStream_t s);
int conv2d_wgrad_zsplits(const ConvShape& cs);
void conv2d_wgrad_launch(DT dt, const void* x, const void* dy, void* dw_out,
                         DT out_dt, float* ws, int z, const void* zero16,
                         const ConvShape& cs, hipStream_t s);
void transpose_w_launch(DT dt, const void* w, void* w_t, int KH, int KW,
                        int Cin, int Cout, hipStream_t s);

// ---- batchnorm.hip ---------------------------------------------------------
int64_t bn_stats_ws_floats(DT dt, const void* x, int64_t rows, int cols);
void bn_stats_launch(DT dt, const void* x, float* mean, float* invstd,
                     float* rmean, float* rvar, float momentum, float* ws,
                     int64_t rows, int cols, float eps, hipStream_t s);
void bn_finalize_launch(const float* sum, const float* sumsq, float* mean,
                        float* invstd, float* rmean, float* rvar,
                        float momentum, int64_t rows, int cols, float eps,
                        hipStream_t s);
void bn_apply_launch(DT dt, const void* x, const float* mean,
                     const float* invstd, const float* gamma, const float* beta,
                     void* y, int64_t rows, int cols, bool relu, hipStream_t s);
void bn_apply_drop_launch(DT dt, const void* x, const float* mean,
                          const float* invstd, const float* gamma,
                          const float* beta, void* y, int64_t rows, int cols,
                          float p, uint64_t seed, const int64_t* ctr,
                          hipStream_t s);
void bn_infer_launch(DT dt, const void* x, const float* rmean,
                     const float* rvar, const float* gamma, const float* beta,
                     void* y, int64_t rows, int cols, float eps, bool relu,
                     hipStream_t s);
int64_t bn_bwd_ws_floats(DT dt, const void* x, const void* dy, int64_t rows,
                         int cols);
void bn_bwd_reduce_launch(DT dt, const void* x, const void* dy, const void* y_relu,
                          const float* mean, const float* invstd, float* sum_dy,
                          float* sum_dy_xhat, float* ws, int64_t rows, int cols,
                          float dy_scale, hipStream_t s);
void bn_bwd_apply_launch(DT dt, const void* x, const void* dy, const void* y_relu,
                         const float* mean, const float* invstd,
                         const float* gamma, const float* sum_dy,
                         const float* sum_dy_xhat, void* dx, const void* resid,
                         int64_t rows, int cols, float dy_scale,
                         hipStream_t s);

// ---- pool.hip --------------------------------------------------------------
struct PoolShape {
  int N, H, W, C, KH, KW, SH, SW, PH, PW, OH, OW;
};
void maxpool_fwd_launch(DT dt, const void* x, void* y, int32_t* idx,
                        const PoolShape& ps, hipStream_t s);
void maxpool_bwd_launch(DT dt, const void* dy, const int32_t* idx, float* dx_f32,
                        const PoolShape& ps, hipStream_t s);
void avgpool_fwd_launch(DT dt, const void* x, void* y, const PoolShape& ps,
                        hipStream_t s);
void avgpool_bwd_launch(DT dt, const void* dy, void* dx, const PoolShape& ps,
                        hipStream_t s);

// ---- loss.hip --------------------------------------------------------------
void ce_fwd_launch(DT dt, const void* logits, const int64_t* targets,
                   float* loss, float* lse, int64_t rows, int cols,
                   hipStream_t s);
void ce_bwd_launch(DT dt, const void* logits, const int64_t* targets,
                   const float* lse, const float* dloss, void* dlogits,
                   int64_t rows, int cols, hipStream_t s);
// kind: 0 = MSE, 1 = MAE, 2 = Huber
void ptloss_fwd_launch(DT dt, const void* pred, const void* tgt, float* out,
                       int64_t n, int kind, float delta, hipStream_t s);
void ptloss_bwd_launch(DT dt, const void* pred, const void* tgt,
                       const float* dloss, void* dpred, int64_t n, int kind,
                       float delta, hipStream_t s);

// ---- groupnorm.hip ---------------------------------------------------------
void gn_fwd_launch(DT dt, const void* x, const float* gamma, const float* beta,
                   void* y, float* mean, float* invstd, int64_t N, int64_t HW,
                   int C, int G, float eps, hipStream_t s);
void gn_bwd_launch(DT dt, const void* x, const void* dy, const float* mean,
                   const float* invstd, const float* gamma, float* s1,
                   float* s2, void* dx, float* dgamma, float* dbeta, int64_t N,
                   int64_t HW, int C, int G, hipStream_t s);

// ---- layernorm.hip ---------------------------------------------------------
void ln_fwd_launch(DT dt, const void* x, const float* gamma, const float* beta,
                   void* y, float* mean, float* invstd, int64_t rows, int cols,
                   float eps, hipStream_t s);
int64_t ln_bwd_ws_floats(DT dt, const void* x, const void* dy, int64_t rows,
                         int cols);
void ln_bwd_launch(DT dt, const void* x, const void* dy, const float* gamma,
                   const float* mean, const float* invstd, void* dx,
                   const void* resid, float* dgamma2, float* ws, int64_t rows,
                   int cols, hipStream_t s);

// ---- embedding.hip ---------------------------------------------------------
void embedding_fwd_launch(DT dt, const int64_t* ids, const void* table,
                          void* y, int64_t n_ids, int dim, hipStream_t s);
void embedding_bwd_launch(DT dt, const int64_t* ids, const void* dy,
                          float* dtable_f32, int64_t n_ids, int dim,
                          hipStream_t s);

// ---- attention.hip (bf16 only; D in {64,128}) ------------------------------
// i_str/o_str/do_str/w_str are {row(S), head, batch} element strides of
// the [B,H,S,D]-shaped (possibly non-contiguous, d-contiguous) views
void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, const void* zero16, int B, int H, int S,
                     int D, const int64_t* i_str, const int64_t* o_str,
                     bool causal, float scale, hipStream_t s);
void attn_bwd_launch(const void* q, const void* k, const void* v,
                     const void* o, const void* dout, const float* lse,
                     float* di, float* dq_ws, void* dq, void* dk, void* dv,
                     const void* zero16, int B, int H, int S, int D,
                     const int64_t* i_str, const int64_t* o_str,
                     const int64_t* do_str, const int64_t* w_str, bool causal,
                     float scale, hipStream_t s);

// ---- bmm.hip ---------------------------------------------------------------
void bmm_launch(DT dt, const void* a, const void* b, void* c,
                const void* zero16, int batch, int M, int N, int K, int lda,
                int ldb, int ldc, int64_t sa, int64_t sb, int64_t sc, bool ta,
                bool tb, hipStream_t s);

// ---- softmax.hip -----------------------------------------------------------
void smax_fwd_launch(DT dt, const void* x, void* y, int64_t rows, int cols,
                     int mrows, int qoff, float scale, bool causal,
                     hipStream_t s);
void smax_bwd_launch(DT dt, const void* p, const void* dy, void* dx,
                     int64_t rows, int cols, float scale, hipStream_t s);

// ---- decode.hip (bf16; D in {64,128}) --------------------------------------
void attn_decode_launch(const void* q, const void* k, const void* v, float* po,
                        float* ml, void* out, const int64_t* pos_ptr,
                        int len_static, int BH, int cap, int D, int splits,
                        float scale, hipStream_t s);

// ---- optim.hip -------------------------------------------------------------
void sgd_step_launch(DT dt_p, const void* grad, DT dt_g, void* param,
                     float* master, float* momentum_buf, int64_t n, float lr,
                     float momentum, float weight_decay, bool nesterov,
                     bool has_master, hipStream_t s);
void adam_step_launch(DT dt_p, const void* grad, DT dt_g, void* param,
                      float* master, float* m, float* v, int64_t n, int step,
                      const int64_t* step_dev, float lr, float beta1,
                      float beta2, float eps, float weight_decay, bool adamw,
                      bool has_master, hipStream_t s);
void adam_mt_launch(DT dt_p, DT dt_g, bool has_master, const int64_t* desc,
                    const int64_t* chunks, int nchunks, int step,
                    const int64_t* step_dev, float lr, float beta1,
                    float beta2, float eps, float weight_decay, bool adamw,
                    hipStream_t s);

}  // namespace tnn
