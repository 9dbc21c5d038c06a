// NHWC implicit-GEMM convolution on MFMA: fwd / dgrad / wgrad.
//
// Replaces the reference's cuDNN-frontend conv graphs
// (src/nn/layers_impl/cuda/cudnn_conv2d_ops.cu) and the legacy
// im2col->GEMM path (src/nn/layers_impl/legacy_conv2d_layer.cpp:137) with
// the fused form SURVEY §7 calls for: the im2col gather is folded into the
// GEMM's global->LDS staging, so no im2col buffer ever exists.
//
// GEMM views (row-major):
//   fwd  : y[M=N*OH*OW, Cout] = A[M, K=KH*KW*Cin] @ w[K, Cout]
//   dgrad: dx[M=N*H*W, Cin]   = A'[M, K=KH*KW*Cout] @ w_t[K, Cin]
//   wgrad: dw[K=KH*KW*Cin, Cout] = im2col(x)^T @ dy  (deterministic
//          split-M fp32 slabs + casting reduce -- no atomics)
//
// A-tiles are gathered on the fly: a 16B k-chunk stays within one (kh,kw)
// slice whenever Cin (fwd) / Cout (dgrad) is a multiple of the vector
// width -- which holds for every hot layer, and the python op zero-pads
// narrow-channel inputs (the RGB stem) up to the vector width; a scalar
// gather path remains for the rest.

#include "common.h"
#include "kernels.h"
#include "tile_gemm.h"

namespace tnn {

using namespace tile;

// POW2 resolved at compile time: the ?: form makes hipcc emit BOTH the
// shift and the division and select (measured 2x wgrad regression)
template <bool POW2>
DEV int idiv(int x, const IDiv& f) {
  if constexpr (POW2) return x >> f.lg;
  else if (f.magic)  // mul-shift reciprocal (exact for x < 2^26, see IDiv)
    return (int)(((unsigned long long)(unsigned)x * f.magic) >> f.sh);
  else
    return x / f.d;
}
template <bool POW2>
DEV int imod(int x, const IDiv& f) {
  if constexpr (POW2) return x & (f.d - 1);
  else return x - idiv<POW2>(x, f) * f.d;
}

// kidx -> (kh, kw) via mul-shift reciprocal (exact while kidx * KW < 2^16;
// kidx < KH*KW <= ~169 always). A runtime `%`/`/` by the non-pow2 KW costs
// ~25 VALU per gathered chunk (measured: dgrad 302 -> 446 TF from removing
// the stride modulo; this removes the remaining one).
DEV void kdecode(int kidx, const ConvShape& cs, int& kh, int& kw) {
  kh = (kidx * cs.mkw16) >> 16;
  kw = kidx - kh * cs.KW;
}

// ---------------------------------------------------------------------------
// double-buffered pure-glds forward: both operands DMA straight to LDS
// (weights pre-transposed to [Cout, K] so B rows are k-contiguous), the
// next chunk's DMAs overlap the current chunk's MFMAs. Counted
// s_waitcnt vmcnt + raw s_barrier per the CDNA4 glds idiom -- a
// __syncthreads() here would drain the in-flight next-chunk DMAs.
template <typename T, bool POW2, bool STATS = false>
__launch_bounds__(THREADS)
__global__ void k_conv_fwd_db(const T* __restrict__ X,
                              const T* __restrict__ WT2,
                              const float* __restrict__ bias_f32,
                              const T* __restrict__ bias_t, T* __restrict__ Y,
                              const T* __restrict__ zero16, ConvShape cs,
                              int act_kind, float* __restrict__ stats) {
  __shared__ alignas(16) T As[2][BM * BK];
  __shared__ alignas(16) T Bs[2][BN * BK];

  const int M = cs.N * cs.OH * cs.OW;
  const int K = cs.KH * cs.KW * cs.Cin;
  int tm_, tn_;
  // grouped tile ordering when the gathered activation stream exceeds
  // the 256 MiB L3 across its Cout/BN re-reads (see tile_gemm.h)
  if ((int64_t)cs.N * cs.OH * cs.OW * (cs.KH * cs.KW * cs.Cin) >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tm_, tn_);
  else {
    tm_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int m0 = tm_ * BM;
  const int n0 = tn_ * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  auto a_src = [&](int kk0, int rl, int kk) -> const T* {
    int gm = m0 + rl, gk = kk0 + kk;
    if (gm >= M || gk >= K) return zero16;
    int n = idiv<POW2>(gm, cs.d_ohow);
    int rem = gm - n * (cs.OH * cs.OW);
    int oh = idiv<POW2>(rem, cs.d_ow), ow = rem - oh * cs.OW;
    int ci = imod<POW2>(gk, cs.d_cin);
    int kidx = idiv<POW2>(gk, cs.d_cin);
    int kh, kw;
    kdecode(kidx, cs, kh, kw);
    int ih = oh * cs.SH - cs.PH + kh;
    int iw = ow * cs.SW - cs.PW + kw;
    if (ih < 0 || ih >= cs.H || iw < 0 || iw >= cs.W) return zero16;
    return &X[(((int64_t)n * cs.H + ih) * cs.W + iw) * cs.Cin + ci];
  };
  auto b_src = [&](int kk0, int rl, int kk) -> const T* {
    int gn = n0 + rl, gk = kk0 + kk;
    if (gn >= cs.Cout || gk >= K) return zero16;
    return &WT2[(int64_t)gn * K + gk];
  };
  auto stage = [&](int t, int which) {
    int kk0 = t * BK;
    glds_stage<T, BM>(As[which], wc,
                      [&](int rl, int kk) { return a_src(kk0, rl, kk); });
    glds_stage<T, BN>(Bs[which], wc,
                      [&](int rl, int kk) { return b_src(kk0, rl, kk); });
  };
  constexpr int NPER = glds_count<T, BM>() + glds_count<T, BN>();

  const int nch = (K + BK - 1) / BK;
  stage(0, 0);
  for (int t = 0; t < nch; ++t) {
    const int cur = t & 1;
    if (t + 1 < nch) {
      stage(t + 1, cur ^ 1);
      wait_vmcnt<NPER>();   // chunk t landed; t+1 stays in flight
    } else {
      wait_vmcnt<0>();
    }
    __builtin_amdgcn_s_barrier();
    mfma_compute_tile(As[cur], Bs[cur], wc, acc);
    __builtin_amdgcn_s_barrier();   // cur free for the t+2 stage
  }

  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < cs.Cout) {
      if (bias_f32) v += bias_f32[col];
      if (bias_t) v += VecIO<T>::to_f32(bias_t[col]);
      if (act_kind != ACT_LINEAR) v = act_apply(v, act_kind);
      Y[(int64_t)row * cs.Cout + col] = VecIO<T>::from_f32(v);
    }
  });

  if constexpr (STATS) {
    // per-channel sum/sumsq of the tile straight from the accumulators
    // (post-bias, linear act only): saves the BN stats pass's full read
    // of y. C/D fragment col = lane&15; lanes {l, l+16, l+32, l+48} share
    // a column -> two shfl_xor folds, then one atomic per (block, col).
    float* ssum = stats;
    float* ssumsq = stats + cs.Cout;
    const int cr = (wc.lane >> 4) * 4;
    const int cc = wc.lane & 15;
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      int col = n0 + wc.wcol0 + fn * 16 + cc;
      float bias = 0.0f;
      if (col < cs.Cout) {
        if (bias_f32) bias = bias_f32[col];
        if (bias_t) bias = VecIO<T>::to_f32(bias_t[col]);
      }
      float s = 0.0f, ss = 0.0f;
#pragma unroll
      for (int fm = 0; fm < FM; ++fm)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int row = m0 + wc.wrow0 + fm * 16 + cr + j;
          if (row < M && col < cs.Cout) {
            float v = acc[fm][fn][j] + bias;
            s += v;
            ss += v * v;
          }
        }
      s += __shfl_xor(s, 16, 64);
      ss += __shfl_xor(ss, 16, 64);
      s += __shfl_xor(s, 32, 64);
      ss += __shfl_xor(ss, 32, 64);
      if (wc.lane < 16 && col < cs.Cout) {
        atomicAdd(&ssum[col], s);
        atomicAdd(&ssumsq[col], ss);
      }
    }
  }
}

// ---------------------------------------------------------------------------
template <typename T, bool POW2, bool GLDS>
__launch_bounds__(THREADS)
__global__ void k_conv_fwd(const T* __restrict__ X, const T* __restrict__ Wt,
                           const float* __restrict__ bias_f32,
                           const T* __restrict__ bias_t, T* __restrict__ Y,
                           const T* __restrict__ zero16, ConvShape cs,
                           int act_kind) {
  constexpr int V = 16 / sizeof(T);
  __shared__ alignas(16) T As[BM * BK];
  __shared__ alignas(16) T Bs[BN * BK];

  const int M = cs.N * cs.OH * cs.OW;
  const int K = cs.KH * cs.KW * cs.Cin;
  int tm_, tn_;
  // grouped tile ordering when the gathered activation stream exceeds
  // the 256 MiB L3 across its Cout/BN re-reads (see tile_gemm.h)
  if ((int64_t)cs.N * cs.OH * cs.OW * (cs.KH * cs.KW * cs.Cin) >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tm_, tn_);
  else {
    tm_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int m0 = tm_ * BM;
  const int n0 = tn_ * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};
  using VecT = Pack16<T>;

  // fused-im2col source address for a 16B chunk (row, kk elem), or the
  // zero page when the chunk is out of image/padding/tail range; only
  // called on the GLDS path where Cin % V == 0 guarantees one (kh,kw)
  // slice per chunk.
  auto a_src = [&](int k0, int rl, int kk) -> const T* {
    int gm = m0 + rl, gk = k0 + kk;
    if (gm >= M || gk >= K) return zero16;
    int n = idiv<POW2>(gm, cs.d_ohow);
    int rem = gm - n * (cs.OH * cs.OW);
    int oh = idiv<POW2>(rem, cs.d_ow), ow = rem - oh * cs.OW;
    int ci = imod<POW2>(gk, cs.d_cin);
    int kidx = idiv<POW2>(gk, cs.d_cin);
    int kh, kw;
    kdecode(kidx, cs, kh, kw);
    int ih = oh * cs.SH - cs.PH + kh;
    int iw = ow * cs.SW - cs.PW + kw;
    if (ih < 0 || ih >= cs.H || iw < 0 || iw >= cs.W) return zero16;
    return &X[(((int64_t)n * cs.H + ih) * cs.W + iw) * cs.Cin + ci];
  };

  for (int k0 = 0; k0 < K; k0 += BK) {
    if constexpr (GLDS) {
      glds_stage_a<T>(As, wc, [&](int rl, int kk) { return a_src(k0, rl, kk); });
    } else {
    // ---- stage A: im2col gather ----
#pragma unroll
    for (int c = threadIdx.x; c < BM * (BK / V); c += THREADS) {
      int row = c / (BK / V);
      int kk = (c % (BK / V)) * V;
      int gm = m0 + row, gk = k0 + kk;
      VecT v = {};
      if (gm < M && gk < K) {
        int n = idiv<POW2>(gm, cs.d_ohow);
        int rem = gm - n * (cs.OH * cs.OW);
        int oh = idiv<POW2>(rem, cs.d_ow), ow = rem - oh * cs.OW;
        int ci = imod<POW2>(gk, cs.d_cin);
        int kidx = idiv<POW2>(gk, cs.d_cin);
        int kh, kw;
    kdecode(kidx, cs, kh, kw);
        if (ci + V <= cs.Cin && gk + V <= K && (cs.Cin % V) == 0 &&
            aligned16(X)) {
          int ih = oh * cs.SH - cs.PH + kh;
          int iw = ow * cs.SW - cs.PW + kw;
          if (ih >= 0 && ih < cs.H && iw >= 0 && iw < cs.W)
            v = *(const VecT*)&X[(((int64_t)n * cs.H + ih) * cs.W + iw) *
                                     cs.Cin + ci];
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j) {
            int k = gk + j;
            if (k < K) {
              int cij = imod<POW2>(k, cs.d_cin);
              int kj = idiv<POW2>(k, cs.d_cin);
              int kwj = kj % cs.KW, khj = kj / cs.KW;
              int ih = oh * cs.SH - cs.PH + khj;
              int iw = ow * cs.SW - cs.PW + kwj;
              if (ih >= 0 && ih < cs.H && iw >= 0 && iw < cs.W)
                v.e[j] = X[(((int64_t)n * cs.H + ih) * cs.W + iw) * cs.Cin + cij];
            }
          }
        }
      }
      *(VecT*)&As[lds_off<T>(row, kk)] = v;
    }
    }
    // ---- stage B (weights [K, Cout]) -> Bs[n][k] scatter-transpose ----
#pragma unroll
    for (int c = threadIdx.x; c < BK * (BN / V); c += THREADS) {
      int kk = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gk = k0 + kk, gn = n0 + nn;
      VecT v = {};
      if (gk < K) {
        const T* src = &Wt[(int64_t)gk * cs.Cout + gn];
        if (gn + V <= cs.Cout && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gn + j < cs.Cout) v.e[j] = Wt[(int64_t)gk * cs.Cout + gn + j];
        }
      }
#pragma unroll
      for (int j = 0; j < V; ++j)
        Bs[lds_off<T>(nn + j, kk)] = v.e[j];
    }
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < cs.Cout) {
      if (bias_f32) v += bias_f32[col];
      if (bias_t) v += VecIO<T>::to_f32(bias_t[col]);
      if (act_kind != ACT_LINEAR) v = act_apply(v, act_kind);
      Y[(int64_t)row * cs.Cout + col] = VecIO<T>::from_f32(v);
    }
  });
}

// ---------------------------------------------------------------------------
// dgrad: dx[n,ih,iw,ci] = sum_{kh,kw,co} dy[n,oh,ow,co] * w[kh,kw,ci,co]
// with oh = (ih+PH-kh)/SH when divisible. Wt here is the transposed weight
// [KH,KW,Cout,Cin] so B rows are k=(kh,kw,co) with ci contiguous.
// ---------------------------------------------------------------------------
template <typename T, bool POW2, bool GLDS, bool S2 = false, bool S1 = false>
__launch_bounds__(THREADS)
__global__ void k_conv_dgrad(const T* __restrict__ DY, const T* __restrict__ WT,
                             T* __restrict__ DX, const T* __restrict__ zero16,
                             ConvShape cs) {
  constexpr int V = 16 / sizeof(T);
  __shared__ alignas(16) T As[BM * BK];
  __shared__ alignas(16) T Bs[BN * BK];

  // S2: stride-2 parity decomposition. blockIdx.z selects the
  // (ih%2, iw%2) class so every (kh,kw) visited satisfies the stride
  // divisibility test by construction -- the generic kernel at stride 2
  // burns 3/4 of its MFMAs multiplying gathered zeros. The launcher
  // rewrites cs.d_hw/d_w to the per-class (Hc*Wc, Wc) divisors.
  const int cls_a = S2 ? ((int)blockIdx.z >> 1) : 0;
  const int cls_b = S2 ? ((int)blockIdx.z & 1) : 0;
  const int start_h = S2 ? ((cls_a + cs.PH) & 1) : 0;
  const int start_w = S2 ? ((cls_b + cs.PW) & 1) : 0;
  const int nkh = S2 ? ((cs.KH - start_h + 1) >> 1) : cs.KH;
  const int nkw = S2 ? ((cs.KW - start_w + 1) >> 1) : cs.KW;
  const int Hc = S2 ? (cs.H >> 1) : cs.H;
  const int Wc = S2 ? (cs.W >> 1) : cs.W;
  const int M = cs.N * Hc * Wc;
  const int K = nkh * nkw * cs.Cout;
  int tm_, tn_;
  // grouped tile ordering when the gathered activation stream exceeds
  // the 256 MiB L3 across its Cout/BN re-reads (see tile_gemm.h)
  if ((int64_t)cs.N * cs.OH * cs.OW * (cs.KH * cs.KW * cs.Cin) >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tm_, tn_);
  else {
    tm_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int m0 = tm_ * BM;
  const int n0 = tn_ * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};
  using VecT = Pack16<T>;

  auto a_src = [&](int k0, int rl, int kk) -> const T* {
    int gm = m0 + rl, gk = k0 + kk;
    if (gm >= M || gk >= K) return zero16;
    int n = idiv<POW2>(gm, cs.d_hw);
    int rem = gm - n * (Hc * Wc);
    int ih = idiv<POW2>(rem, cs.d_w), iw = rem - ih * Wc;
    int co = imod<POW2>(gk, cs.d_cout);
    int kidx = idiv<POW2>(gk, cs.d_cout);
    if constexpr (S2) {
      ih = 2 * ih + cls_a;
      iw = 2 * iw + cls_b;
      int kwi = kidx & (nkw - 1);              // nkw is 1 or 2
      int khi = nkw == 2 ? (kidx >> 1) : kidx;
      int kh = start_h + 2 * khi, kw = start_w + 2 * kwi;
      int th = ih + cs.PH - kh, tw = iw + cs.PW - kw;
      if (th < 0 || tw < 0) return zero16;
      int oh = th >> 1, ow = tw >> 1;          // divisible by construction
      if (oh >= cs.OH || ow >= cs.OW) return zero16;
      return &DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) * cs.Cout + co];
    } else {
      int kh, kw;
    kdecode(kidx, cs, kh, kw);
      int th = ih + cs.PH - kh, tw = iw + cs.PW - kw;
      if constexpr (S1) {  // stride 1: no divisibility test, oh == th
        if (th < 0 || tw < 0 || th >= cs.OH || tw >= cs.OW) return zero16;
        return &DY[(((int64_t)n * cs.OH + th) * cs.OW + tw) * cs.Cout + co];
      } else {
        if (th < 0 || tw < 0 || th % cs.SH || tw % cs.SW) return zero16;
        int oh = th / cs.SH, ow = tw / cs.SW;
        if (oh >= cs.OH || ow >= cs.OW) return zero16;
        return &DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) * cs.Cout + co];
      }
    }
  };

  for (int k0 = 0; k0 < K; k0 += BK) {
    if constexpr (GLDS) {
      glds_stage_a<T>(As, wc, [&](int rl, int kk) { return a_src(k0, rl, kk); });
    } else {
    // ---- stage A: gather dy with stride/padding inversion ----
#pragma unroll
    for (int c = threadIdx.x; c < BM * (BK / V); c += THREADS) {
      int row = c / (BK / V);
      int kk = (c % (BK / V)) * V;
      int gm = m0 + row, gk = k0 + kk;
      VecT v = {};
      if (gm < M && gk < K) {
        int n = idiv<POW2>(gm, cs.d_hw);
        int rem = gm - n * (cs.H * cs.W);
        int ih = idiv<POW2>(rem, cs.d_w), iw = rem - ih * cs.W;
        int co = imod<POW2>(gk, cs.d_cout);
        int kidx = idiv<POW2>(gk, cs.d_cout);
        int kh, kw;
    kdecode(kidx, cs, kh, kw);
        auto gather_one = [&](int khj, int kwj, int coj) -> T {
          int th = ih + cs.PH - khj, tw = iw + cs.PW - kwj;
          if (th < 0 || tw < 0 || th % cs.SH || tw % cs.SW) return T(0.0f);
          int oh = th / cs.SH, ow = tw / cs.SW;
          if (oh >= cs.OH || ow >= cs.OW) return T(0.0f);
          return DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) * cs.Cout + coj];
        };
        if (co + V <= cs.Cout && gk + V <= K && (cs.Cout % V) == 0 &&
            aligned16(DY)) {
          int th = ih + cs.PH - kh, tw = iw + cs.PW - kw;
          if (th >= 0 && tw >= 0 && th % cs.SH == 0 && tw % cs.SW == 0) {
            int oh = th / cs.SH, ow = tw / cs.SW;
            if (oh < cs.OH && ow < cs.OW)
              v = *(const VecT*)&DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) *
                                        cs.Cout + co];
          }
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j) {
            int k = gk + j;
            if (k < K) {
              int coj = imod<POW2>(k, cs.d_cout);
              int kj = idiv<POW2>(k, cs.d_cout);
              v.e[j] = gather_one(kj / cs.KW, kj % cs.KW, coj);
            }
          }
        }
      }
      *(VecT*)&As[lds_off<T>(row, kk)] = v;
    }
    }
    // ---- stage B (w_t [KH*KW*Cout, Cin]) -> Bs[ci][k] ----
#pragma unroll
    for (int c = threadIdx.x; c < BK * (BN / V); c += THREADS) {
      int kk = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gk = k0 + kk, gn = n0 + nn;
      VecT v = {};
      if (gk < K) {
        int64_t wrow = gk;
        if constexpr (S2) {
          // compacted (khi,kwi,co) -> full (kh*KW+kw)*Cout+co row of WT
          int co = imod<POW2>(gk, cs.d_cout);
          int kidx = idiv<POW2>(gk, cs.d_cout);
          int kwi = kidx & (nkw - 1);
          int khi = nkw == 2 ? (kidx >> 1) : kidx;
          wrow = (int64_t)((start_h + 2 * khi) * cs.KW + start_w + 2 * kwi) *
                     cs.Cout + co;
        }
        const T* src = &WT[wrow * cs.Cin + gn];
        if (gn + V <= cs.Cin && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gn + j < cs.Cin) v.e[j] = src[j];
        }
      }
#pragma unroll
      for (int j = 0; j < V; ++j)
        Bs[lds_off<T>(nn + j, kk)] = v.e[j];
    }
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < cs.Cin) {
      int64_t pix = row;
      if constexpr (S2) {
        int n = idiv<POW2>(row, cs.d_hw);
        int rem = row - n * (Hc * Wc);
        int ihp = idiv<POW2>(rem, cs.d_w);
        int ih = 2 * ihp + cls_a, iw = 2 * (rem - ihp * Wc) + cls_b;
        pix = ((int64_t)n * cs.H + ih) * cs.W + iw;
      }
      DX[pix * cs.Cin + col] = VecIO<T>::from_f32(v);
    }
  });
}

// ---------------------------------------------------------------------------
// double-buffered pure-glds dgrad (see k_conv_fwd_db): the weight comes
// pre-transposed to [Cin, KH*KW*Cout] so B rows are k-contiguous; with S2
// the compacted parity-class k index remaps into the full column space.
template <typename T, bool POW2, bool S2 = false, bool S1 = false>
__launch_bounds__(THREADS)
__global__ void k_conv_dgrad_db(const T* __restrict__ DY,
                                const T* __restrict__ WT2D,
                                T* __restrict__ DX,
                                const T* __restrict__ zero16, ConvShape cs) {
  __shared__ alignas(16) T As[2][BM * BK];
  __shared__ alignas(16) T Bs[2][BN * BK];

  const int cls_a = S2 ? ((int)blockIdx.z >> 1) : 0;
  const int cls_b = S2 ? ((int)blockIdx.z & 1) : 0;
  const int start_h = S2 ? ((cls_a + cs.PH) & 1) : 0;
  const int start_w = S2 ? ((cls_b + cs.PW) & 1) : 0;
  const int nkh = S2 ? ((cs.KH - start_h + 1) >> 1) : cs.KH;
  const int nkw = S2 ? ((cs.KW - start_w + 1) >> 1) : cs.KW;
  const int Hc = S2 ? (cs.H >> 1) : cs.H;
  const int Wc = S2 ? (cs.W >> 1) : cs.W;
  const int M = cs.N * Hc * Wc;
  const int K = nkh * nkw * cs.Cout;
  const int KFULL = cs.KH * cs.KW * cs.Cout;
  int tm_, tn_;
  // grouped tile ordering when the gathered activation stream exceeds
  // the 256 MiB L3 across its Cout/BN re-reads (see tile_gemm.h)
  if ((int64_t)cs.N * cs.OH * cs.OW * (cs.KH * cs.KW * cs.Cin) >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tm_, tn_);
  else {
    tm_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int m0 = tm_ * BM;
  const int n0 = tn_ * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  auto a_src = [&](int kk0, int rl, int kk) -> const T* {
    int gm = m0 + rl, gk = kk0 + kk;
    if (gm >= M || gk >= K) return zero16;
    int n = idiv<POW2>(gm, cs.d_hw);
    int rem = gm - n * (Hc * Wc);
    int ih = idiv<POW2>(rem, cs.d_w), iw = rem - ih * Wc;
    int co = imod<POW2>(gk, cs.d_cout);
    int kidx = idiv<POW2>(gk, cs.d_cout);
    if constexpr (S2) {
      ih = 2 * ih + cls_a;
      iw = 2 * iw + cls_b;
      int kwi = kidx & (nkw - 1);
      int khi = nkw == 2 ? (kidx >> 1) : kidx;
      int kh = start_h + 2 * khi, kw = start_w + 2 * kwi;
      int th = ih + cs.PH - kh, tw = iw + cs.PW - kw;
      if (th < 0 || tw < 0) return zero16;
      int oh = th >> 1, ow = tw >> 1;
      if (oh >= cs.OH || ow >= cs.OW) return zero16;
      return &DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) * cs.Cout + co];
    } else {
      int kh, kw;
    kdecode(kidx, cs, kh, kw);
      int th = ih + cs.PH - kh, tw = iw + cs.PW - kw;
      if constexpr (S1) {  // stride 1: no divisibility test, oh == th
        if (th < 0 || tw < 0 || th >= cs.OH || tw >= cs.OW) return zero16;
        return &DY[(((int64_t)n * cs.OH + th) * cs.OW + tw) * cs.Cout + co];
      } else {
        if (th < 0 || tw < 0 || th % cs.SH || tw % cs.SW) return zero16;
        int oh = th / cs.SH, ow = tw / cs.SW;
        if (oh >= cs.OH || ow >= cs.OW) return zero16;
        return &DY[(((int64_t)n * cs.OH + oh) * cs.OW + ow) * cs.Cout + co];
      }
    }
  };
  auto b_src = [&](int kk0, int rl, int kk) -> const T* {
    int gn = n0 + rl, gk = kk0 + kk;
    if (gn >= cs.Cin || gk >= K) return zero16;
    int64_t col = gk;
    if constexpr (S2) {
      int co = imod<POW2>(gk, cs.d_cout);
      int kidx = idiv<POW2>(gk, cs.d_cout);
      int kwi = kidx & (nkw - 1);
      int khi = nkw == 2 ? (kidx >> 1) : kidx;
      col = (int64_t)((start_h + 2 * khi) * cs.KW + start_w + 2 * kwi) *
                cs.Cout + co;
    }
    return &WT2D[(int64_t)gn * KFULL + col];
  };
  auto stage = [&](int t, int which) {
    int kk0 = t * BK;
    glds_stage<T, BM>(As[which], wc,
                      [&](int rl, int kk) { return a_src(kk0, rl, kk); });
    glds_stage<T, BN>(Bs[which], wc,
                      [&](int rl, int kk) { return b_src(kk0, rl, kk); });
  };
  constexpr int NPER = glds_count<T, BM>() + glds_count<T, BN>();

  const int nch = (K + BK - 1) / BK;
  stage(0, 0);
  for (int t = 0; t < nch; ++t) {
    const int cur = t & 1;
    if (t + 1 < nch) {
      stage(t + 1, cur ^ 1);
      wait_vmcnt<NPER>();
    } else {
      wait_vmcnt<0>();
    }
    __builtin_amdgcn_s_barrier();
    mfma_compute_tile(As[cur], Bs[cur], wc, acc);
    __builtin_amdgcn_s_barrier();
  }

  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < cs.Cin) {
      int64_t pix = row;
      if constexpr (S2) {
        int n = idiv<POW2>(row, cs.d_hw);
        int rem = row - n * (Hc * Wc);
        int ihp = idiv<POW2>(rem, cs.d_w);
        int ih = 2 * ihp + cls_a, iw = 2 * (rem - ihp * Wc) + cls_b;
        pix = ((int64_t)n * cs.H + ih) * cs.W + iw;
      }
      DX[pix * cs.Cin + col] = VecIO<T>::from_f32(v);
    }
  });
}

// ---------------------------------------------------------------------------
// wgrad: dw[(kh,kw,ci), co] += sum_m x_gather * dy ; fp32 atomics over
// grid.z m-slices.
// ---------------------------------------------------------------------------
template <typename T, bool POW2>
__launch_bounds__(THREADS, 3)  // cap VGPRs: unconstrained allocation hit 221-256 VGPR = 1-2 waves/SIMD
__global__ void k_conv_wgrad(const T* __restrict__ X, const T* __restrict__ DY,
                             float* __restrict__ DW, ConvShape cs) {
  constexpr int V = 16 / sizeof(T);
  __shared__ alignas(16) T As[BM * BK];  // rows = (kh,kw,ci), k = m
  __shared__ alignas(16) T Bs[BN * BK];  // rows = co, k = m

  const int M = cs.N * cs.OH * cs.OW;         // reduction dim
  const int Kout = cs.KH * cs.KW * cs.Cin;    // output rows
  int tr_, tn_;
  // same grouped ordering: the gathered-x A operand [Kout, M] is
  // re-read gridDim.y times without it
  if ((int64_t)(cs.KH * cs.KW * cs.Cin) * cs.N * cs.OH * cs.OW >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tr_, tn_);
  else {
    tr_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int r0 = tr_ * BM;
  const int n0 = tn_ * BN;
  const int m_begin = (int)((int64_t)M * blockIdx.z / gridDim.z);
  const int m_end = (int)((int64_t)M * (blockIdx.z + 1) / gridDim.z);
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};
  using VecT = Pack16<T>;

  for (int k0 = m_begin; k0 < m_end; k0 += BK) {
    // ---- stage A^T: x rows ci-contiguous, scatter into As[ko][m] ----
#pragma unroll 1  // full unroll quadruples live address state (spills)
    for (int c = threadIdx.x; c < BK * (BM / V); c += THREADS) {
      int mm = c / (BM / V);
      int rr = (c % (BM / V)) * V;
      int gm = k0 + mm;
      int gr = r0 + rr;
      VecT v = {};
      if (gm < m_end && gr < Kout) {
        int n = idiv<POW2>(gm, cs.d_ohow);
        int rem = gm - n * (cs.OH * cs.OW);
        int oh = idiv<POW2>(rem, cs.d_ow), ow = rem - oh * cs.OW;
        int ci = imod<POW2>(gr, cs.d_cin);
        int kidx = idiv<POW2>(gr, cs.d_cin);
        int kh, kw;
    kdecode(kidx, cs, kh, kw);
        if (ci + V <= cs.Cin && gr + V <= Kout && (cs.Cin % V) == 0 &&
            aligned16(X)) {
          int ih = oh * cs.SH - cs.PH + kh;
          int iw = ow * cs.SW - cs.PW + kw;
          if (ih >= 0 && ih < cs.H && iw >= 0 && iw < cs.W)
            v = *(const VecT*)&X[(((int64_t)n * cs.H + ih) * cs.W + iw) *
                                     cs.Cin + ci];
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j) {
            int r = gr + j;
            if (r < Kout) {
              int cij = imod<POW2>(r, cs.d_cin);
              int kj = idiv<POW2>(r, cs.d_cin);
              int kwj = kj % cs.KW, khj = kj / cs.KW;
              int ih = oh * cs.SH - cs.PH + khj;
              int iw = ow * cs.SW - cs.PW + kwj;
              if (ih >= 0 && ih < cs.H && iw >= 0 && iw < cs.W)
                v.e[j] = X[(((int64_t)n * cs.H + ih) * cs.W + iw) * cs.Cin + cij];
            }
          }
        }
      }
#pragma unroll
      for (int j = 0; j < V; ++j)
        As[lds_off<T>(rr + j, mm)] = v.e[j];
    }
    // ---- stage B: dy[m][co] -> Bs[co][m] ----
#pragma unroll
    for (int c = threadIdx.x; c < BK * (BN / V); c += THREADS) {
      int mm = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gm = k0 + mm;
      int gn = n0 + nn;
      VecT v = {};
      if (gm < m_end) {
        const T* src = &DY[(int64_t)gm * cs.Cout + gn];
        if (gn + V <= cs.Cout && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gn + j < cs.Cout) v.e[j] = DY[(int64_t)gm * cs.Cout + gn + j];
        }
      }
#pragma unroll
      for (int j = 0; j < V; ++j)
        Bs[lds_off<T>(nn + j, mm)] = v.e[j];
    }
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  float* out = DW + (int64_t)blockIdx.z * Kout * cs.Cout;
  epilogue_visit(wc, acc, r0, n0, [&](int row, int col, float v) {
    if (row < Kout && col < cs.Cout) out[(int64_t)row * cs.Cout + col] = v;
  });
}

// ---------------------------------------------------------------------------
// vector-guaranteed wgrad (Cin%V==0, Cout%V==0, 16B-aligned tensors): stages
// in three phases -- addresses (zero-page for OOB/padding), then all 16B
// loads, then the LDS scatter -- so RA+RB global loads stay in flight.
// The generic kernel's `#pragma unroll 1` staging (needed there to avoid
// 221-256 VGPR spills from per-iteration gather state) serializes to one
// load in flight per wave; holding only pointers between phases keeps the
// live state at ~24 VGPR.
// ---------------------------------------------------------------------------
template <typename T, bool POW2>
__launch_bounds__(THREADS, 3)
__global__ void k_conv_wgrad_vec(const T* __restrict__ X,
                                 const T* __restrict__ DY,
                                 float* __restrict__ DW,
                                 const T* __restrict__ zero16, ConvShape cs) {
  constexpr int V = 16 / sizeof(T);
  constexpr int RA = BK * (BM / V) / THREADS;
  constexpr int RB = BK * (BN / V) / THREADS;
  __shared__ alignas(16) T As[BM * BK];  // rows = (kh,kw,ci), k = m
  __shared__ alignas(16) T Bs[BN * BK];  // rows = co, k = m
  const int M = cs.N * cs.OH * cs.OW;
  const int Kout = cs.KH * cs.KW * cs.Cin;
  int tr_, tn_;
  // same grouped ordering: the gathered-x A operand [Kout, M] is
  // re-read gridDim.y times without it
  if ((int64_t)(cs.KH * cs.KW * cs.Cin) * cs.N * cs.OH * cs.OW >
      (int64_t)32 * 1024 * 1024)
    tile::tile_remap_xcd<4>(tr_, tn_);
  else {
    tr_ = blockIdx.x;
    tn_ = blockIdx.y;
  }
  const int r0 = tr_ * BM;
  const int n0 = tn_ * BN;
  const int m_begin = (int)((int64_t)M * blockIdx.z / gridDim.z);
  const int m_end = (int)((int64_t)M * (blockIdx.z + 1) / gridDim.z);
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};
  using VecT = Pack16<T>;

  // bf16: each thread owns one 8-row group x 4 consecutive m columns --
  // after the loads an in-thread 8x4 transpose packs 4 m-values per row
  // into one 8B ds_write_b64 (vs 8x ds_write_b16 per load; LDS
  // instruction issue was wgrad's post-phase-separation bottleneck:
  // 48 b16 writes vs 16 MFMAs per thread-chunk). The XOR swizzle flips
  // byte bits 4-6 only, so the 8B alignment survives. fp32 keeps the
  // generic scatter.
  //
  // Software-pipelined (T14 rotate): tile t's registers are written to
  // LDS, then tile t+1's loads are ISSUED into the same (now dead)
  // registers before tile t's MFMA phase — the gather latency lands
  // under the MFMAs instead of being exposed at the head of every
  // chunk (the attention kernels' TStage split, applied here).
  constexpr bool TR = sizeof(T) == 2;
  // lane mapping: consecutive lanes share the m-quad and span the row
  // groups: the 16 lanes' 16B global loads form one contiguous 256B run
  // (the conflict-free same-row/mm-strided write mapping was tried and
  // lost more to broken global coalescing than its bank relief won)
  const int a_rr = TR ? (threadIdx.x & (BM / V - 1)) * V : 0;
  const int a_mm0 = TR ? (threadIdx.x / (BM / V)) * 4 : 0;
  VecT va[RA], vb[RB];
  auto load_tile = [&](int k0) {
    const T* asrc[RA];
#pragma unroll
    for (int it = 0; it < RA; ++it) {
      int mm, rr;
      if constexpr (TR) {
        mm = a_mm0 + it;
        rr = a_rr;
      } else {
        int c = threadIdx.x + it * THREADS;
        mm = c / (BM / V);
        rr = (c % (BM / V)) * V;
      }
      int gm = k0 + mm, gr = r0 + rr;
      const T* p = zero16;
      if (gm < m_end && gr < Kout) {
        int n = idiv<POW2>(gm, cs.d_ohow);
        int rem = gm - n * (cs.OH * cs.OW);
        int oh = idiv<POW2>(rem, cs.d_ow), ow = rem - oh * cs.OW;
        int ci = imod<POW2>(gr, cs.d_cin);
        int kidx = idiv<POW2>(gr, cs.d_cin);
        int kh, kw;
        kdecode(kidx, cs, kh, kw);
        int ih = oh * cs.SH - cs.PH + kh;
        int iw = ow * cs.SW - cs.PW + kw;
        if (ih >= 0 && ih < cs.H && iw >= 0 && iw < cs.W)
          p = &X[(((int64_t)n * cs.H + ih) * cs.W + iw) * cs.Cin + ci];
      }
      asrc[it] = p;
    }
    const T* bsrc[RB];
#pragma unroll
    for (int it = 0; it < RB; ++it) {
      int c = threadIdx.x + it * THREADS;
      int mm = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gm = k0 + mm;
      bsrc[it] = (gm < m_end && n0 + nn + V <= cs.Cout)
                     ? &DY[(int64_t)gm * cs.Cout + n0 + nn]
                     : zero16;
    }
#pragma unroll
    for (int it = 0; it < RA; ++it) va[it] = *(const VecT*)asrc[it];
#pragma unroll
    for (int it = 0; it < RB; ++it) vb[it] = *(const VecT*)bsrc[it];
  };

  load_tile(m_begin);
  for (int k0 = m_begin; k0 < m_end; k0 += BK) {
    if constexpr (TR) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        struct alignas(8) H4 { T e[4]; } h;
#pragma unroll
        for (int q = 0; q < 4; ++q) h.e[q] = va[q].e[j];
        *(H4*)&As[lds_off<T>(a_rr + j, a_mm0)] = h;
      }
    } else {
#pragma unroll
      for (int it = 0; it < RA; ++it) {
        int c = threadIdx.x + it * THREADS;
        int mm = c / (BM / V);
        int rr = (c % (BM / V)) * V;
#pragma unroll
        for (int j = 0; j < V; ++j) As[lds_off<T>(rr + j, mm)] = va[it].e[j];
      }
    }
#pragma unroll
    for (int it = 0; it < RB; ++it) {
      int c = threadIdx.x + it * THREADS;
      int mm = c / (BN / V);
      int nn = (c % (BN / V)) * V;
#pragma unroll
      for (int j = 0; j < V; ++j) Bs[lds_off<T>(nn + j, mm)] = vb[it].e[j];
    }
    if (k0 + BK < m_end) load_tile(k0 + BK);
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  float* out = DW + (int64_t)blockIdx.z * Kout * cs.Cout;
  epilogue_visit(wc, acc, r0, n0, [&](int row, int col, float v) {
    if (row < Kout && col < cs.Cout) out[(int64_t)row * cs.Cout + col] = v;
  });
}

// ---------------------------------------------------------------------------
// batched LDS-tiled 2D transpose: out[b][c][r] = in[b][r][c]. 32x32 tiles,
// coalesced on both sides (the naive elementwise form writes strided and
// measured 2.6 TB/s; this hits ~5).
template <typename T>
__global__ void k_transpose2d_tiled(const T* __restrict__ in,
                                    T* __restrict__ out, int rows, int cols) {
  __shared__ T tile[32][33];
  const int c0 = blockIdx.x * 32, r0 = blockIdx.y * 32;
  const int64_t slice = (int64_t)blockIdx.z * rows * cols;
  const T* src = in + slice;
  T* dst = out + slice;
  const int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;  // 32x8
#pragma unroll
  for (int d = 0; d < 32; d += 8) {
    int r = r0 + ty + d, c = c0 + tx;
    if (r < rows && c < cols) tile[ty + d][tx] = src[(int64_t)r * cols + c];
  }
  __syncthreads();
#pragma unroll
  for (int d = 0; d < 32; d += 8) {
    int c = c0 + ty + d, r = r0 + tx;
    if (c < cols && r < rows) dst[(int64_t)c * rows + r] = tile[tx][ty + d];
  }
}

// ---------------------------------------------------------------------------
// [KH,KW,Ci,Co] -> [Ci, KH*KW*Co] (k-contiguous rows for the dgrad B glds)
template <typename T>
__global__ void k_transpose_w_dgrad(const T* __restrict__ W,
                                    T* __restrict__ WT2D, int KHW, int Cin,
                                    int Cout) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t n = (int64_t)KHW * Cin * Cout;
  if (i >= n) return;
  int co = i % Cout;
  int ci = (i / Cout) % Cin;
  int64_t khw = i / ((int64_t)Cin * Cout);
  WT2D[((int64_t)ci * KHW + khw) * Cout + co] = W[i];
}

// co stays the inner dim on both sides: move whole 16B chunks, coalesced
// reads AND writes
template <typename T>
__global__ void k_transpose_w_dgrad_vec(const T* __restrict__ W,
                                        T* __restrict__ WT2D, int KHW,
                                        int Cin, int Cout) {
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  const int cov = Cout / V;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t n = (int64_t)KHW * Cin * cov;
  if (i >= n) return;
  int c = i % cov;
  int ci = (i / cov) % Cin;
  int64_t khw = i / ((int64_t)Cin * cov);
  ((P*)WT2D)[((int64_t)ci * KHW + khw) * cov + c] =
      ((const P*)W)[((int64_t)khw * Cin + ci) * cov + c];
}

// ---------------------------------------------------------------------------
static bool all_pow2(const ConvShape& cs) {
  return cs.d_ohow.lg >= 0 && cs.d_ow.lg >= 0 && cs.d_cin.lg >= 0 &&
         cs.d_hw.lg >= 0 && cs.d_w.lg >= 0 && cs.d_cout.lg >= 0;
}

bool conv2d_fwd_wants_db(DT dt, const void* x, const ConvShape& cs) {
  static const bool db_on = []() {
    const char* e = getenv("TNN_FWD_DB");
    return !(e && atoi(e) == 0);
  }();
  if (!db_on) return false;
  static const int maxk = []() {
    const char* e = getenv("TNN_DB_MAXK");
    return e ? atoi(e) : 2304;
  }();
  int v = dt == DT::F32 ? 4 : 8;
  // 2x LDS buffers cap occupancy at 3 blocks/CU: a win while the k-loop is
  // short (measured +8% at K<=1152, -15% at K=4608 where the single-buffer
  // kernel's 6 resident blocks hide latency better)
  return cs.KH * cs.KW * cs.Cin <= maxk && all_pow2(cs) &&
         cs.Cin % v == 0 && (((uintptr_t)x & 15) == 0);
}

void transpose_w_fwd_launch(DT dt, const void* w, void* w_t2, int KHW, int Cin,
                            int Cout, hipStream_t s) {
  // [KHW*Cin, Cout] -> [Cout, KHW*Cin] is a plain 2D transpose
  int rows = KHW * Cin;
  dim3 grid((Cout + 31) / 32, (rows + 31) / 32, 1);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_transpose2d_tiled<float>, grid, dim3(256), 0, s,
                       (const float*)w, (float*)w_t2, rows, Cout);
  else
    hipLaunchKernelGGL(k_transpose2d_tiled<bf16>, grid, dim3(256), 0, s,
                       (const bf16*)w, (bf16*)w_t2, rows, Cout);
}

void conv2d_fwd_launch(DT dt, const void* x, const void* w, const void* w_t2,
                       const void* bias, void* y, const void* zero16,
                       float* stats, const ConvShape& cs, bool relu,
                       hipStream_t s) {
  int M = cs.N * cs.OH * cs.OW;
  dim3 grid(ceil_div(M, BM), ceil_div(cs.Cout, BN));
  int act = relu ? ACT_RELU : ACT_LINEAR;
  bool p2 = all_pow2(cs);
  if (dt == DT::F32) {
    if (w_t2) {
      auto kern = stats ? k_conv_fwd_db<float, true, true>
                        : k_conv_fwd_db<float, true>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const float*)x,
                         (const float*)w_t2, (const float*)bias,
                         (const float*)nullptr, (float*)y,
                         (const float*)zero16, cs, act, stats);
      return;
    }
    bool g = p2 && cs.Cin % 4 == 0 && (((uintptr_t)x & 15) == 0);
    auto kern = g ? k_conv_fwd<float, true, true>
                  : (p2 ? k_conv_fwd<float, true, false>
                        : k_conv_fwd<float, false, false>);
    hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s,
                       (const float*)x, (const float*)w, (const float*)bias,
                       (const float*)nullptr, (float*)y, (const float*)zero16,
                       cs, act);
  } else {
    if (w_t2) {
      auto kern = stats ? k_conv_fwd_db<bf16, true, true>
                        : k_conv_fwd_db<bf16, true>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const bf16*)x,
                         (const bf16*)w_t2, (const float*)nullptr,
                         (const bf16*)bias, (bf16*)y, (const bf16*)zero16, cs,
                         act, stats);
      return;
    }
    bool g = p2 && cs.Cin % 8 == 0 && (((uintptr_t)x & 15) == 0);
    auto kern = g ? k_conv_fwd<bf16, true, true>
                  : (p2 ? k_conv_fwd<bf16, true, false>
                        : k_conv_fwd<bf16, false, false>);
    hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s,
                       (const bf16*)x, (const bf16*)w, (const float*)nullptr,
                       (const bf16*)bias, (bf16*)y, (const bf16*)zero16, cs,
                       act);
  }
}

bool conv2d_dgrad_wants_db(DT dt, const void* dy, const ConvShape& cs) {
  static const bool db_on = []() {
    const char* e = getenv("TNN_FWD_DB");
    return !(e && atoi(e) == 0);
  }();
  if (!db_on) return false;
  static const int maxk = []() {
    const char* e = getenv("TNN_DB_MAXK_DGRAD");
    return e ? atoi(e) : 4608;  // dgrad ties-or-wins with DB up to here
  }();
  int v = dt == DT::F32 ? 4 : 8;
  return cs.KH * cs.KW * cs.Cout <= maxk && all_pow2(cs) &&
         cs.Cout % v == 0 && (((uintptr_t)dy & 15) == 0);
}

void transpose_w_dgrad_launch(DT dt, const void* w, void* w_t2d, int KHW,
                              int Cin, int Cout, hipStream_t s) {
  int V = dt == DT::F32 ? 4 : 8;
  if (Cout % V == 0 && (((uintptr_t)w & 15) == 0)) {
    int64_t n = (int64_t)KHW * Cin * (Cout / V);
    int blocks = (int)((n + 255) / 256);
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_transpose_w_dgrad_vec<float>, dim3(blocks),
                         dim3(256), 0, s, (const float*)w, (float*)w_t2d, KHW,
                         Cin, Cout);
    else
      hipLaunchKernelGGL(k_transpose_w_dgrad_vec<bf16>, dim3(blocks),
                         dim3(256), 0, s, (const bf16*)w, (bf16*)w_t2d, KHW,
                         Cin, Cout);
    return;
  }
  int64_t n = (int64_t)KHW * Cin * Cout;
  int blocks = (int)((n + 255) / 256);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_transpose_w_dgrad<float>, dim3(blocks), dim3(256), 0,
                       s, (const float*)w, (float*)w_t2d, KHW, Cin, Cout);
  else
    hipLaunchKernelGGL(k_transpose_w_dgrad<bf16>, dim3(blocks), dim3(256), 0,
                       s, (const bf16*)w, (bf16*)w_t2d, KHW, Cin, Cout);
}

void conv2d_dgrad_launch(DT dt, const void* dy, const void* w_t,
                         const void* w_t2d, void* dx, const void* zero16,
                         const ConvShape& cs, hipStream_t s) {
  int M = cs.N * cs.H * cs.W;
  dim3 grid(ceil_div(M, BM), ceil_div(cs.Cin, BN));
  bool p2 = all_pow2(cs);
  // stride-2 parity decomposition (see k_conv_dgrad<..., S2>): 4 classes in
  // grid.z over an M/4-row problem with only the stride-compatible (kh,kw)
  // taps in K. Requires even spatial dims; routed through the glds variant.
  bool s2 = p2 && cs.SH == 2 && cs.SW == 2 && cs.H % 2 == 0 && cs.W % 2 == 0;
  ConvShape cs2 = cs;
  if (s2) {
    cs2.d_hw.set((cs.H / 2) * (cs.W / 2));
    cs2.d_w.set(cs.W / 2);
    s2 = cs2.d_hw.lg >= 0 && cs2.d_w.lg >= 0;
  }
  dim3 grid_s2(ceil_div(cs.N * (cs.H / 2) * (cs.W / 2), BM),
               ceil_div(cs.Cin, BN), 4);
  if (dt == DT::F32) {
    bool g = p2 && cs.Cout % 4 == 0 && (((uintptr_t)dy & 15) == 0);
    if (w_t2d && s2) {
      hipLaunchKernelGGL((k_conv_dgrad_db<float, true, true>), grid_s2,
                         dim3(THREADS), 0, s, (const float*)dy,
                         (const float*)w_t2d, (float*)dx, (const float*)zero16,
                         cs2);
      return;
    }
    if (w_t2d) {
      auto kern = cs.SH == 1 && cs.SW == 1
                      ? k_conv_dgrad_db<float, true, false, true>
                      : k_conv_dgrad_db<float, true>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const float*)dy,
                         (const float*)w_t2d, (float*)dx,
                         (const float*)zero16, cs);
      return;
    }
    if (g && s2) {
      hipLaunchKernelGGL((k_conv_dgrad<float, true, true, true>), grid_s2,
                         dim3(THREADS), 0, s, (const float*)dy,
                         (const float*)w_t, (float*)dx, (const float*)zero16,
                         cs2);
      return;
    }
    bool s1 = cs.SH == 1 && cs.SW == 1;
    auto kern = g ? (s1 ? k_conv_dgrad<float, true, true, false, true>
                        : k_conv_dgrad<float, true, true>)
                  : (p2 ? k_conv_dgrad<float, true, false>
                        : k_conv_dgrad<float, false, false>);
    hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s,
                       (const float*)dy, (const float*)w_t, (float*)dx,
                       (const float*)zero16, cs);
  } else {
    bool g = p2 && cs.Cout % 8 == 0 && (((uintptr_t)dy & 15) == 0);
    if (w_t2d && s2) {
      hipLaunchKernelGGL((k_conv_dgrad_db<bf16, true, true>), grid_s2,
                         dim3(THREADS), 0, s, (const bf16*)dy,
                         (const bf16*)w_t2d, (bf16*)dx, (const bf16*)zero16,
                         cs2);
      return;
    }
    if (w_t2d) {
      auto kern = cs.SH == 1 && cs.SW == 1
                      ? k_conv_dgrad_db<bf16, true, false, true>
                      : k_conv_dgrad_db<bf16, true>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const bf16*)dy,
                         (const bf16*)w_t2d, (bf16*)dx, (const bf16*)zero16,
                         cs);
      return;
    }
    if (g && s2) {
      hipLaunchKernelGGL((k_conv_dgrad<bf16, true, true, true>), grid_s2,
                         dim3(THREADS), 0, s, (const bf16*)dy,
                         (const bf16*)w_t, (bf16*)dx, (const bf16*)zero16,
                         cs2);
      return;
    }
    bool s1 = cs.SH == 1 && cs.SW == 1;
    auto kern = g ? (s1 ? k_conv_dgrad<bf16, true, true, false, true>
                        : k_conv_dgrad<bf16, true, true>)
                  : (p2 ? k_conv_dgrad<bf16, true, false>
                        : k_conv_dgrad<bf16, false, false>);
    hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s,
                       (const bf16*)dy, (const bf16*)w_t, (bf16*)dx,
                       (const bf16*)zero16, cs);
  }
}

int conv2d_wgrad_zsplits(const ConvShape& cs) {
  int M = cs.N * cs.OH * cs.OW;
  int Kout = cs.KH * cs.KW * cs.Cin;
  int base = ceil_div(Kout, BM) * ceil_div(cs.Cout, BN);
  int want = 2048;
  int z = base >= want ? 1
                       : std::min(ceil_div(M, 4 * BK), ceil_div(want, base));
  return std::max(z, 1);
}

void conv2d_wgrad_launch(DT dt, const void* x, const void* dy, void* dw_out,
                         DT out_dt, float* ws, int z, const void* zero16,
                         const ConvShape& cs, hipStream_t s) {
  int Kout = cs.KH * cs.KW * cs.Cin;
  // bf16 output always goes through the (casting) reduce; fp32 with z==1
  // writes the slab target directly
  bool direct = z == 1 && out_dt == DT::F32;
  float* target = direct ? (float*)dw_out : ws;
  dim3 grid(ceil_div(Kout, BM), ceil_div(cs.Cout, BN), z);
  bool p2 = all_pow2(cs);
  if (dt == DT::F32) {
    bool vec = p2 && cs.Cin % 4 == 0 && cs.Cout % 4 == 0 &&
               (((uintptr_t)x & 15) == 0) && (((uintptr_t)dy & 15) == 0);
    if (vec) {
      hipLaunchKernelGGL((k_conv_wgrad_vec<float, true>), grid, dim3(THREADS),
                         0, s, (const float*)x, (const float*)dy, target,
                         (const float*)zero16, cs);
    } else {
      auto kern = p2 ? k_conv_wgrad<float, true> : k_conv_wgrad<float, false>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const float*)x,
                         (const float*)dy, target, cs);
    }
  } else {
    bool vec = p2 && cs.Cin % 8 == 0 && cs.Cout % 8 == 0 &&
               (((uintptr_t)x & 15) == 0) && (((uintptr_t)dy & 15) == 0);
    if (vec) {
      hipLaunchKernelGGL((k_conv_wgrad_vec<bf16, true>), grid, dim3(THREADS),
                         0, s, (const bf16*)x, (const bf16*)dy, target,
                         (const bf16*)zero16, cs);
    } else {
      auto kern = p2 ? k_conv_wgrad<bf16, true> : k_conv_wgrad<bf16, false>;
      hipLaunchKernelGGL(kern, grid, dim3(THREADS), 0, s, (const bf16*)x,
                         (const bf16*)dy, target, cs);
    }
  }
  if (!direct)
    splitk_reduce_launch(ws, dw_out, out_dt, z, (int64_t)Kout * cs.Cout, s);
}

void transpose_w_launch(DT dt, const void* w, void* w_t, int KH, int KW,
                        int Cin, int Cout, hipStream_t s) {
  // batched [Cin, Cout] -> [Cout, Cin] transpose per (kh,kw) slice
  dim3 grid((Cout + 31) / 32, (Cin + 31) / 32, KH * KW);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_transpose2d_tiled<float>, grid, dim3(256), 0, s,
                       (const float*)w, (float*)w_t, Cin, Cout);
  else
    hipLaunchKernelGGL(k_transpose2d_tiled<bf16>, grid, dim3(256), 0, s,
                       (const bf16*)w, (bf16*)w_t, Cin, Cout);
}

}  // namespace tnn
