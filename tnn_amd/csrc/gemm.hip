// MFMA GEMM kernels: NN (fused bias+activation epilogue), NT (dgrad),
// TN (wgrad: deterministic split-M fp32 slabs + casting reduce), plus
// M=1 GEMV forms for the decode path.
// Replaces the reference's cublasGemmEx / cuDNN-frontend matmul call sites
// (src/math/cuda/gemm.cu:64, src/math/cuda/cudnn_gemm.cu:285,
// src/nn/layers_impl/cuda/dense_ops.cu:17-117).

#include "common.h"
#include "kernels.h"
#include "tile_gemm.h"

namespace tnn {

using namespace tile;

// ---------------------------------------------------------------------------
// NN / NT kernel: C[M,N] = act(A[M,K] @ B + bias)
//   TRANS_B = false: B is [K, N] row-major (staged transposed into LDS)
//   TRANS_B = true:  B is [N, K] row-major (staged directly; C = A @ B^T)
// ---------------------------------------------------------------------------

template <typename T, bool TRANS_B, bool GLDS>
__launch_bounds__(THREADS)
__global__ void k_gemm(const T* __restrict__ A, const T* __restrict__ B,
                       const float* __restrict__ bias_f32,
                       const T* __restrict__ bias_t, T* __restrict__ C,
                       const T* __restrict__ zero16,
                       const T* __restrict__ resid, int M, int N, int K,
                       int act_kind) {
  constexpr int V = 16 / sizeof(T);  // elems per 16B vector
  __shared__ alignas(16) T As[BM * BK];
  __shared__ alignas(16) T Bs[BN * BK];

  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  using VecT = Pack16<T>;

  for (int k0 = 0; k0 < K; k0 += BK) {
    if constexpr (GLDS) {
      // K % V == 0 and A 16B-aligned guaranteed by the launcher
      glds_stage_a<T>(As, wc, [&](int rl, int kk) -> const T* {
        int gm = m0 + rl, gk = k0 + kk;
        if (gm >= M || gk >= K) return zero16;
        return &A[(int64_t)gm * K + gk];
      });
    } else {
    // ---- stage A: [BM][BK], vectorized, zero-filled at edges ----
#pragma unroll
    for (int c = threadIdx.x; c < BM * (BK / V); c += THREADS) {
      int row = c / (BK / V);
      int kk = (c % (BK / V)) * V;
      int gm = m0 + row, gk = k0 + kk;
      VecT v = {};
      if (gm < M) {
        const T* src = &A[(int64_t)gm * K + gk];
        if (gk + V <= K && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gk + j < K) v.e[j] = A[(int64_t)gm * K + gk + j];
        }
      }
      *(VecT*)&As[lds_off<T>(row, kk)] = v;
    }
    }
    // ---- stage B into Bs[n][k] ----
    if constexpr (TRANS_B && GLDS) {
      // B[N,K] rows are already the image rows: same DMA form as A
      glds_stage<T, BN>(Bs, wc, [&](int rl, int kk) -> const T* {
        int gn = n0 + rl, gk = k0 + kk;
        if (gn >= N || gk >= K) return zero16;
        return &B[(int64_t)gn * K + gk];
      });
    } else if constexpr (TRANS_B) {
      // B[N,K]: rows are output columns; contiguous copy
#pragma unroll
      for (int c = threadIdx.x; c < BN * (BK / V); c += THREADS) {
        int col = c / (BK / V);
        int kk = (c % (BK / V)) * V;
        int gn = n0 + col, gk = k0 + kk;
        VecT v = {};
        if (gn < N) {
          const T* src = &B[(int64_t)gn * K + gk];
          if (gk + V <= K && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gk + j < K) v.e[j] = B[(int64_t)gn * K + gk + j];
          }
        }
        *(VecT*)&Bs[lds_off<T>(col, kk)] = v;
      }
    } else {
      // B[K,N]: load n-contiguous vectors, scatter-transpose into LDS
#pragma unroll 1
      for (int c = threadIdx.x; c < BK * (BN / V); c += THREADS) {
        int kk = c / (BN / V);
        int nn = (c % (BN / V)) * V;
        int gk = k0 + kk, gn = n0 + nn;
        VecT v = {};
        if (gk < K) {
          const T* src = &B[(int64_t)gk * N + gn];
          if (gn + V <= N && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gn + j < N) v.e[j] = B[(int64_t)gk * N + gn + j];
          }
        }
#pragma unroll
        for (int j = 0; j < V; ++j)
          Bs[lds_off<T>(nn + j, kk)] = v.e[j];
      }
    }
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  // ---- epilogue: bias + activation + residual + cast store ----
  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < N) {
      if (bias_f32) v += bias_f32[col];
      if (bias_t) v += VecIO<T>::to_f32(bias_t[col]);
      if (act_kind != ACT_LINEAR) v = act_apply(v, act_kind);
      if (resid) v += VecIO<T>::to_f32(resid[(int64_t)row * N + col]);
      C[(int64_t)row * N + col] = VecIO<T>::from_f32(v);
    }
  });
}

// ---------------------------------------------------------------------------
// TN kernel: C[K,N] += A[M,K]^T @ B[M,N], fp32 atomic accumulate.
// Grid.z slices the M reduction (split-K in the GEMM sense) so small-output
// wgrads still fill 256 CUs (SURVEY §7 hard part 1).
// ---------------------------------------------------------------------------

template <typename T>
__launch_bounds__(THREADS, 3)  // cap VGPRs: unconstrained allocation hit 221-256 VGPR = 1-2 waves/SIMD
__global__ void k_gemm_tn(const T* __restrict__ A, const T* __restrict__ B,
                          float* __restrict__ C, int M, int N, int K) {
  constexpr int V = 16 / sizeof(T);
  __shared__ alignas(16) T As[BM * BK];  // rows = K-dim, k = m-chunk
  __shared__ alignas(16) T Bs[BN * BK];

  const int r0 = blockIdx.x * BM;  // output row = A column
  const int n0 = blockIdx.y * BN;
  // 32-bit m index: int64 div/mod in the staging loop costs ~100 VALU
  // cycles per chunk (measured: wgrad at 1/4 the MFMA residency of fwd)
  const int m_begin = (int)((int64_t)M * blockIdx.z / gridDim.z);
  const int m_end = (int)((int64_t)M * (blockIdx.z + 1) / gridDim.z);
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  using VecT = Pack16<T>;

  for (int k0 = m_begin; k0 < m_end; k0 += BK) {
    // ---- stage A^T: load A[m][r..r+V] (contiguous), scatter to As[r][m] ----
#pragma unroll 1  // full unroll quadruples live address state (spills)
    for (int c = threadIdx.x; c < BK * (BM / V); c += THREADS) {
      int mm = c / (BM / V);
      int rr = (c % (BM / V)) * V;
      int gm = k0 + mm;
      int gr = r0 + rr;
      VecT v = {};
      if (gm < m_end) {
        const T* src = &A[(int64_t)gm * K + gr];
        if (gr + V <= K && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gr + j < K) v.e[j] = A[(int64_t)gm * K + gr + j];
        }
      }
#pragma unroll
      for (int j = 0; j < V; ++j)
        As[lds_off<T>(rr + j, mm)] = v.e[j];
    }
    // ---- stage B[m][n] -> Bs[n][m] ----
#pragma unroll 1
    for (int c = threadIdx.x; c < BK * (BN / V); c += THREADS) {
      int mm = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gm = k0 + mm;
      int gn = n0 + nn;
      VecT v = {};
      if (gm < m_end) {
        const T* src = &B[(int64_t)gm * N + gn];
        if (gn + V <= N && aligned16(src)) {
          v = *(const VecT*)src;
        } else {
#pragma unroll
          for (int j = 0; j < V; ++j)
            if (gn + j < N) v.e[j] = B[(int64_t)gm * N + gn + j];
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

  // slab write: block z owns slab z of the fp32 workspace (C when z==1);
  // a separate reduce pass sums slabs — deterministic, no atomic contention
  float* out = C + (int64_t)blockIdx.z * K * N;
  epilogue_visit(wc, acc, r0, n0, [&](int row, int col, float v) {
    if (row < K && col < N) out[(int64_t)row * N + col] = v;
  });
}

// double-buffered pure-glds NT gemm (see k_conv_fwd_db): A[M,K] and
// B[N,K] rows are both k-contiguous, so the next chunk's LDS-DMAs overlap
// the current chunk's MFMAs. Dispatched for K <= 3072 (2x LDS buffers cap
// occupancy at 3 blocks/CU; long k-loops prefer the 6-block single-buffer
// kernel).
template <typename T>
__launch_bounds__(THREADS)
__global__ void k_gemm_nt_db(const T* __restrict__ A, const T* __restrict__ B,
                             const float* __restrict__ bias_f32,
                             const T* __restrict__ bias_t, T* __restrict__ C,
                             const T* __restrict__ zero16,
                             const T* __restrict__ resid, int M, int N, int K,
                             int act_kind) {
  __shared__ alignas(16) T As[2][BM * BK];
  __shared__ alignas(16) T Bs[2][BN * BK];
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  auto stage = [&](int t, int which) {
    int kk0 = t * BK;
    glds_stage<T, BM>(As[which], wc, [&](int rl, int kk) -> const T* {
      int gm = m0 + rl, gk = kk0 + kk;
      if (gm >= M || gk >= K) return zero16;
      return &A[(int64_t)gm * K + gk];
    });
    glds_stage<T, BN>(Bs[which], wc, [&](int rl, int kk) -> const T* {
      int gn = n0 + rl, gk = kk0 + kk;
      if (gn >= N || gk >= K) return zero16;
      return &B[(int64_t)gn * K + gk];
    });
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
    if (row < M && col < N) {
      if (bias_f32) v += bias_f32[col];
      if (bias_t) v += VecIO<T>::to_f32(bias_t[col]);
      if (act_kind != ACT_LINEAR) v = act_apply(v, act_kind);
      if (resid) v += VecIO<T>::to_f32(resid[(int64_t)row * N + col]);
      C[(int64_t)row * N + col] = VecIO<T>::from_f32(v);
    }
  });
}

// M=1 NT matvec (decode-path linears): one wave per output row, 16B loads
// over K, cross-lane reduce. The MFMA tile kernel wastes 127/128 of its
// A-tile rows at M=1 and is ~30x slower on gpt2 decode shapes.
template <typename T>
__launch_bounds__(256)
__global__ void k_gemv_nt(const T* __restrict__ x, const T* __restrict__ B,
                          const float* __restrict__ bias_f32,
                          const T* __restrict__ bias_t, T* __restrict__ y,
                          int N, int K, int act_kind) {
  constexpr int V = 16 / sizeof(T);
  using VecT = Pack16<T>;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int n = blockIdx.x * 4 + wave;
  if (n >= N) return;
  const T* row = &B[(int64_t)n * K];
  float acc = 0.0f;
  if ((K % V) == 0 && aligned16(row) && aligned16(x)) {
    for (int k = lane * V; k < K; k += 64 * V) {
      VecT vb = *(const VecT*)&row[k];
      VecT vx = *(const VecT*)&x[k];
#pragma unroll
      for (int j = 0; j < V; ++j)
        acc += VecIO<T>::to_f32(vx.e[j]) * VecIO<T>::to_f32(vb.e[j]);
    }
  } else {
    for (int k = lane; k < K; k += 64)
      acc += VecIO<T>::to_f32(x[k]) * VecIO<T>::to_f32(row[k]);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    acc += __shfl_xor(acc, off, 64);
  if (lane == 0) {
    if (bias_f32) acc += bias_f32[n];
    if (bias_t) acc += VecIO<T>::to_f32(bias_t[n]);
    if (act_kind != ACT_LINEAR) acc = act_apply(acc, act_kind);
    y[n] = VecIO<T>::from_f32(acc);
  }
}

// K-split NT gemm for huge-K underfilled shapes (the GPT-2 vocab-head
// dgrad: dx[4096,768] = dy[4096,50264] @ W^T — 384 blocks and a 785-
// chunk k-loop on the plain kernel, 313 TF). grid.z slices K; each
// slice writes an fp32 slab, the split-K reduce (shared with wgrad)
// sums and casts. Double-buffered glds staging (both operands' rows are
// k-contiguous).
template <typename T>
__launch_bounds__(THREADS)
__global__ void k_gemm_nt_z(const T* __restrict__ A, const T* __restrict__ B,
                            float* __restrict__ Cws,
                            const T* __restrict__ zero16, int M, int N,
                            int K) {
  __shared__ alignas(16) T As[2][BM * BK];
  __shared__ alignas(16) T Bs[2][BN * BK];
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int Z = gridDim.z;
  const int k_lo = (int)(((int64_t)K * blockIdx.z / Z) / BK * BK);
  const int k_hi = blockIdx.z + 1 == Z
                       ? K
                       : (int)(((int64_t)K * (blockIdx.z + 1) / Z) / BK * BK);
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  auto stage = [&](int kk0, int which) {
    glds_stage<T, BM>(As[which], wc, [&](int rl, int kk) -> const T* {
      int gm = m0 + rl, gk = kk0 + kk;
      if (gm >= M || gk >= k_hi) return zero16;
      return &A[(int64_t)gm * K + gk];
    });
    glds_stage<T, BN>(Bs[which], wc, [&](int rl, int kk) -> const T* {
      int gn = n0 + rl, gk = kk0 + kk;
      if (gn >= N || gk >= k_hi) return zero16;
      return &B[(int64_t)gn * K + gk];
    });
  };
  constexpr int NPER = glds_count<T, BM>() + glds_count<T, BN>();

  const int nch = (k_hi - k_lo + BK - 1) / BK;
  stage(k_lo, 0);
  for (int t = 0; t < nch; ++t) {
    const int cur = t & 1;
    if (t + 1 < nch) {
      stage(k_lo + (t + 1) * BK, cur ^ 1);
      wait_vmcnt<NPER>();
    } else {
      wait_vmcnt<0>();
    }
    __builtin_amdgcn_s_barrier();
    mfma_compute_tile(As[cur], Bs[cur], wc, acc);
    __builtin_amdgcn_s_barrier();
  }

  float* out = Cws + (int64_t)blockIdx.z * M * N;
  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < N) out[(int64_t)row * N + col] = v;
  });
}

// ---------------------------------------------------------------------------
// 32x32x16-MFMA NT GEMM (bf16): the 16x16x32 tile core needs 3-4 waves
// per SIMD to cover its ~17 cyc/SIMD single-wave issue floor, but
// v_mfma_f32_32x32x16_bf16 sustains PEAK issue at ONE wave/SIMD
// (MI355X_MICROARCH.md cycle table) and touches half the LDS bytes per
// FLOP (A/B reuse 32 wide, not 16). Block tile 128x128 (2x2 waves of
// 64x64, FM=FN=2 fragments of 32x32), BK=64, double-buffered glds
// staging, optional K-split over blockIdx.z into fp32 slabs (finalized
// by k_splitk_fin_ep with the bias/act/residual epilogue).
namespace nt32 {
constexpr int BM2 = 128, BN2 = 64, BK2 = 64;
#ifndef GEMM_REMAP_GM
#define GEMM_REMAP_GM 8
#endif

// C/D map for 32x32x16 (guide: col = lane&31, row = (reg&3) + 8*(reg>>2)
// + 4*(lane>>5), reg in [0,16))
template <typename T>
__launch_bounds__(256)
__global__ void k_gemm_nt32(const T* __restrict__ A, const T* __restrict__ B,
                            const float* __restrict__ bias_f32,
                            const T* __restrict__ bias_t, T* __restrict__ C,
                            float* __restrict__ Cws,
                            const T* __restrict__ zero16,
                            const T* __restrict__ resid, int M, int N, int K,
                            int act_kind) {
  using namespace tile;
  __shared__ alignas(16) T As[2][BM2 * BK2];
  __shared__ alignas(16) T Bs[2][BN2 * BK2];
  int tm, tn;
  // grouped ordering pays only when the re-read streams exceed the
  // 256 MiB L3 (vocab-head-sized operands measured +9%; L3-resident
  // shapes were flat-to-negative)
  if ((int64_t)M * N + (int64_t)N * K > (int64_t)48 * 1024 * 1024)
    tile::tile_remap<GEMM_REMAP_GM>(tm, tn);
  else {
    tm = blockIdx.x;
    tn = blockIdx.y;
  }
  const int m0 = tm * BM2;
  const int n0 = tn * BN2;
  const int Z = gridDim.z;
  const int k_lo = Z == 1 ? 0 : (int)(((int64_t)K * blockIdx.z / Z) / BK2 * BK2);
  const int k_hi = (Z == 1 || blockIdx.z + 1 == Z)
                       ? K
                       : (int)(((int64_t)K * (blockIdx.z + 1) / Z) / BK2 * BK2);
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wrow0 = wid * 32;   // 4 M-stacked waves: 32 rows x 64 cols each
  const WaveCoord wc;  // only for glds_stage lane bookkeeping

  f32x16 acc[2] = {};

  auto stage = [&](int kk0, int which) {
    glds_stage<T, BM2>(As[which], wc, [&](int rl, int kk) -> const T* {
      int gm = m0 + rl, gk = kk0 + kk;
      if (gm >= M || gk >= k_hi) return zero16;
      return &A[(int64_t)gm * K + gk];
    });
    glds_stage<T, BN2>(Bs[which], wc, [&](int rl, int kk) -> const T* {
      int gn = n0 + rl, gk = kk0 + kk;
      if (gn >= N || gk >= k_hi) return zero16;
      return &B[(int64_t)gn * K + gk];
    });
  };
  constexpr int NPER = glds_count<T, BM2>() + glds_count<T, BN2>();

  const int nch = (k_hi - k_lo + BK2 - 1) / BK2;
  stage(k_lo, 0);
  for (int t = 0; t < nch; ++t) {
    const int cur = t & 1;
    if (t + 1 < nch) {
      stage(k_lo + (t + 1) * BK2, cur ^ 1);
      wait_vmcnt<NPER>();
    } else {
      wait_vmcnt<0>();
    }
    __builtin_amdgcn_s_barrier();
    // a-frag: lane supplies row (lane&31), k (lane>>5)*8..+8
    const int ar = lane & 31;
    const int kh = (lane >> 5) * 8;
#pragma unroll
    for (int ks = 0; ks < BK2 / 16; ++ks) {
      const int kb = ks * 16 + kh;
      bf16x8 a, b[2];
      a = *(const bf16x8*)&As[cur][lds_off<T>(wrow0 + ar, kb)];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        b[fn] = *(const bf16x8*)&Bs[cur][lds_off<T>(fn * 32 + ar, kb)];
#pragma unroll
      for (int fn = 0; fn < 2; ++fn)
        acc[fn] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a, b[fn], acc[fn], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
  }

  // epilogue
  const int cc = lane & 31;
  const int r4 = (lane >> 5) * 4;
#pragma unroll
    for (int fn = 0; fn < 2; ++fn)
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int row = m0 + wrow0 + (reg & 3) + 8 * (reg >> 2) + r4;
        const int col = n0 + fn * 32 + cc;
        if (row >= M || col >= N) continue;
        float vv = acc[fn][reg];
        if (Z > 1) {
          Cws[(int64_t)blockIdx.z * M * N + (int64_t)row * N + col] = vv;
        } else {
          if (bias_f32) vv += bias_f32[col];
          else if (bias_t) vv += VecIO<T>::to_f32(bias_t[col]);
          if (resid) vv += VecIO<T>::to_f32(resid[(int64_t)row * N + col]);
          C[(int64_t)row * N + col] = VecIO<T>::from_f32(act_apply(vv, act_kind));
        }
      }
}
}  // namespace nt32

// split-K for moderately underfilled FORWARD NT shapes (e.g. the
// transformer's [B*S, 768] projections: 384 blocks on 256 CUs ran at
// ~170 TF while full-grid launches of the same kernel reach ~2.3x
// that): fill to ~3 blocks/CU, epilogue moves into the fused finalize.
int gemm_nt_fsplits(DT dt, int M, int N, int K) {
  const int base = ceil_div(M, BM) * ceil_div(N, BN);
  if (base >= 768 || K < 4 * BK) return 1;
  const bool g = dt == DT::F32 ? K % 4 == 0 : K % 8 == 0;
  if (!g) return 1;
  return std::min(ceil_div(768, base), K / (2 * BK));
}

// fused split-K finalize: fold the Z fp32 slabs, add bias / residual,
// apply the activation, cast to T (the bias/act/resid epilogue the
// direct kernels do, relocated here for the z-split route)
template <typename T>
__launch_bounds__(256)
__global__ void k_splitk_fin_ep(const float4* __restrict__ ws,
                                const float* __restrict__ bias_f32,
                                const T* __restrict__ bias_t,
                                const T* __restrict__ resid,
                                T* __restrict__ out, int z, int N,
                                int64_t n4, int act_kind) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  struct alignas(4 * sizeof(T)) T4 { T e[4]; };
  for (; i < n4; i += stride) {
    float4 a{0.0f, 0.0f, 0.0f, 0.0f};
    for (int sl = 0; sl < z; ++sl) {
      float4 v = ws[(int64_t)sl * n4 + i];
      a.x += v.x; a.y += v.y; a.z += v.z; a.w += v.w;
    }
    float r[4] = {a.x, a.y, a.z, a.w};
    const int col0 = (int)((i * 4) % N);
    T4 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      if (bias_f32) r[j] += bias_f32[col0 + j];
      else if (bias_t) r[j] += VecIO<T>::to_f32(bias_t[col0 + j]);
      if (resid) r[j] += VecIO<T>::to_f32(resid[i * 4 + j]);
      o.e[j] = VecIO<T>::from_f32(act_apply(r[j], act_kind));
    }
    ((T4*)out)[i] = o;
  }
}

void gemm_nt_zf_launch(DT dt, const void* a, const void* b, float* ws,
                       const void* bias_f32_or_t, const void* resid, void* c,
                       int z, const void* zero16, int M, int N, int K,
                       int act_kind, hipStream_t s) {
  dim3 grid(ceil_div(M, BM), ceil_div(N, BN), z);
  const int64_t n4 = (int64_t)M * N / 4;  // N % 4 == 0 enforced by caller
  int blocks = (int)std::min<int64_t>((n4 + 255) / 256, (int64_t)2048);
  if (dt == DT::F32) {
    hipLaunchKernelGGL(k_gemm_nt_z<float>, grid, dim3(THREADS), 0, s,
                       (const float*)a, (const float*)b, ws,
                       (const float*)zero16, M, N, K);
    hipLaunchKernelGGL(k_splitk_fin_ep<float>, dim3(blocks), dim3(256), 0, s,
                       (const float4*)ws, (const float*)bias_f32_or_t,
                       (const float*)nullptr, (const float*)resid, (float*)c,
                       z, N, n4, act_kind);
  } else {
    hipLaunchKernelGGL(k_gemm_nt_z<bf16>, grid, dim3(THREADS), 0, s,
                       (const bf16*)a, (const bf16*)b, ws,
                       (const bf16*)zero16, M, N, K);
    hipLaunchKernelGGL(k_splitk_fin_ep<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const float4*)ws, (const float*)nullptr,
                       (const bf16*)bias_f32_or_t, (const bf16*)resid,
                       (bf16*)c, z, N, n4, act_kind);
  }
}

// K-splits for the 32-wide NT kernel: fill to ~2 blocks/CU (its LDS
// footprint caps residency at 2)
int gemm_nt32_zsplits(int M, int N, int K) {
  using namespace nt32;
  const int base = ceil_div(M, BM2) * ceil_div(N, BN2);
  if (base >= 512 || K < 4 * BK2) return 1;
  if (K % 8 != 0) return 1;
  return std::min(ceil_div(512, base), K / (2 * BK2));
}

void gemm_nt32_launch(const void* a, const void* b, float* ws,
                      const void* bias_f32, const void* bias_t,
                      const void* resid, void* c, int z, const void* zero16,
                      int M, int N, int K, int act_kind, hipStream_t s) {
  using namespace nt32;
  dim3 grid(ceil_div(M, BM2), ceil_div(N, BN2), z);
  hipLaunchKernelGGL(k_gemm_nt32<bf16>, grid, dim3(256), 0, s, (const bf16*)a,
                     (const bf16*)b, (const float*)bias_f32,
                     (const bf16*)bias_t, (bf16*)c, ws, (const bf16*)zero16,
                     (const bf16*)resid, M, N, K, act_kind);
  if (z > 1) {
    const int64_t n4 = (int64_t)M * N / 4;
    int blocks = (int)std::min<int64_t>((n4 + 255) / 256, (int64_t)2048);
    hipLaunchKernelGGL(k_splitk_fin_ep<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const float4*)ws, (const float*)bias_f32,
                       (const bf16*)bias_t, (const bf16*)resid, (bf16*)c, z,
                       N, n4, act_kind);
  }
}


int gemm_nt_zsplits(DT dt, int M, int N, int K) {
  const int base = ceil_div(M, BM) * ceil_div(N, BN);
  if (base >= 768 || K < 4 * BK) return 1;
  const bool g = dt == DT::F32 ? K % 4 == 0 : K % 8 == 0;
  if (!g) return 1;
  return std::min(ceil_div(K, 4 * BK), ceil_div(1024, base));
}

void gemm_nt_z_launch(DT dt, const void* a, const void* b, float* ws,
                      void* c_out, DT out_dt, int z, const void* zero16,
                      int M, int N, int K, hipStream_t s) {
  dim3 grid(ceil_div(M, BM), ceil_div(N, BN), z);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_gemm_nt_z<float>, grid, dim3(THREADS), 0, s,
                       (const float*)a, (const float*)b, ws,
                       (const float*)zero16, M, N, K);
  else
    hipLaunchKernelGGL(k_gemm_nt_z<bf16>, grid, dim3(THREADS), 0, s,
                       (const bf16*)a, (const bf16*)b, ws,
                       (const bf16*)zero16, M, N, K);
  splitk_reduce_launch(ws, c_out, out_dt, z, (int64_t)M * N, s);
}

// Single-launch M=1 NN matvec: out[N] = x[K] @ B[K,N], x cached in LDS,
// bias/activation/residual fused into the epilogue. The two-kernel
// K-split form below measured 23us per decode linear (its finalize is a
// 48-deep serial slab reduce on 3 blocks); this is one latency-bound
// pass (~2-3us). Blocks own 64 n-columns; the 4 thread-rows split K and
// fold through LDS. K <= GEMV1_MAX_K (x LDS cache); larger K falls back.
constexpr int GEMV1_MAX_K = 8192;

template <typename T>
__launch_bounds__(256)
__global__ void k_gemv_nn1(const T* __restrict__ x, const T* __restrict__ B,
                           const float* __restrict__ bias_f32,
                           const T* __restrict__ bias_t,
                           const T* __restrict__ res, T* __restrict__ y,
                           float* __restrict__ part,
                           const float* __restrict__ ln_g,
                           const float* __restrict__ ln_b, float ln_eps,
                           int N, int K, int act_kind) {
  __shared__ float xs[GEMV1_MAX_K];
  __shared__ float red[4][64];
  const int c = threadIdx.x & 63;
  const int ks = threadIdx.x >> 6;
  const int n = blockIdx.x * 64 + c;
  // this block's K slice (grid.y = nsplit; 4-aligned so the thread-row
  // interleave stays uniform)
  const int nsplit = gridDim.y;
  const int k_lo = (int)(((int64_t)K * blockIdx.y / nsplit) & ~3LL);
  const int k_hi = blockIdx.y + 1 == nsplit
                       ? K
                       : (int)(((int64_t)K * (blockIdx.y + 1) / nsplit) & ~3LL);
  if (ln_g) {
    // fused LayerNorm on the x vector (decode: saves the separate ln
    // kernel + its memory round trip; every split block recomputes the
    // row stats — trivial next to the per-kernel floor)
    for (int k = threadIdx.x; k < K; k += 256)
      xs[k] = VecIO<T>::to_f32(x[k]);
    __syncthreads();
    float sm = 0.0f, sq = 0.0f;
    for (int k = threadIdx.x; k < K; k += 256) {
      const float v = xs[k];
      sm += v;
      sq += v * v;
    }
    sm = block_reduce_sum(sm, red[0]);
    __shared__ float bmean;
    if (threadIdx.x == 0) bmean = sm / (float)K;
    __syncthreads();
    sq = block_reduce_sum(sq, red[0]);
    __shared__ float binv;
    if (threadIdx.x == 0)
      binv = rsqrtf(fmaxf(sq / (float)K - bmean * bmean, 0.0f) + ln_eps);
    __syncthreads();
    const float mu = bmean, is = binv;
    for (int k = k_lo + (int)threadIdx.x; k < k_hi; k += 256)
      xs[k] = (xs[k] - mu) * is * ln_g[k] + ln_b[k];
    __syncthreads();
  } else {
    for (int k = k_lo + (int)threadIdx.x; k < k_hi; k += 256)
      xs[k] = VecIO<T>::to_f32(x[k]);
    __syncthreads();
  }
  float acc = 0.0f;
  if (n < N) {
    int k = k_lo + ks;
    // batched 8-deep loads, then the FMAs: keeps 8 loads in flight per
    // thread instead of a serialized load-fma chain
    for (; k + 32 <= k_hi; k += 32) {
      float bv[8];
#pragma unroll
      for (int u = 0; u < 8; ++u)
        bv[u] = VecIO<T>::to_f32(B[(int64_t)(k + 4 * u) * N + n]);
#pragma unroll
      for (int u = 0; u < 8; ++u) acc += xs[k + 4 * u] * bv[u];
    }
    for (; k < k_hi; k += 4)
      acc += xs[k] * VecIO<T>::to_f32(B[(int64_t)k * N + n]);
  }
  red[ks][c] = acc;
  __syncthreads();
  if (ks == 0 && n < N) {
    float a = red[0][c] + red[1][c] + red[2][c] + red[3][c];
    if (part) {
      part[(int64_t)blockIdx.y * N + n] = a;
      return;
    }
    if (bias_f32) a += bias_f32[n];
    if (bias_t) a += VecIO<T>::to_f32(bias_t[n]);
    if (act_kind != ACT_LINEAR) a = act_apply(a, act_kind);
    if (res) a += VecIO<T>::to_f32(res[n]);
    y[n] = VecIO<T>::from_f32(a);
  }
}

// parallel finalize over n: sum <= 16 fp32 partial rows + fused epilogue
template <typename T>
__launch_bounds__(256)
__global__ void k_gemv_fin2(const float* __restrict__ part,
                            const float* __restrict__ bias_f32,
                            const T* __restrict__ bias_t,
                            const T* __restrict__ res, T* __restrict__ y,
                            int N, int nsplit, int act_kind) {
  const int n = blockIdx.x * 256 + threadIdx.x;
  if (n >= N) return;
  float a0 = 0.0f, a1 = 0.0f, a2 = 0.0f, a3 = 0.0f;
  int s = 0;
  for (; s + 4 <= nsplit; s += 4) {
    a0 += part[(int64_t)s * N + n];
    a1 += part[(int64_t)(s + 1) * N + n];
    a2 += part[(int64_t)(s + 2) * N + n];
    a3 += part[(int64_t)(s + 3) * N + n];
  }
  for (; s < nsplit; ++s) a0 += part[(int64_t)s * N + n];
  float a = (a0 + a1) + (a2 + a3);
  if (bias_f32) a += bias_f32[n];
  if (bias_t) a += VecIO<T>::to_f32(bias_t[n]);
  if (act_kind != ACT_LINEAR) a = act_apply(a, act_kind);
  if (res) a += VecIO<T>::to_f32(res[n]);
  y[n] = VecIO<T>::from_f32(a);
}

int gemv_nn1_nsplit(int N, int K) {
  // fill the chip: blocks = (N/64) * nsplit ~ 256; keep slices >= 64 rows
  int ncb = ceil_div(N, 64);
  int ns = std::max(1, std::min(16, 256 / ncb));
  ns = std::min(ns, std::max(1, K / 64));
  return ns;
}

void gemv_nn1_launch(DT dt, const void* x, const void* b, const void* bias,
                     const void* res, void* y, float* part, int nsplit,
                     const float* ln_g, const float* ln_b, float ln_eps, int N,
                     int K, int act_kind, hipStream_t s) {
  dim3 g(ceil_div(N, 64), nsplit);
  dim3 fg(ceil_div(N, 256));
  if (dt == DT::F32) {
    hipLaunchKernelGGL(k_gemv_nn1<float>, g, dim3(256), 0, s,
                       (const float*)x, (const float*)b, (const float*)bias,
                       (const float*)nullptr, (const float*)res, (float*)y,
                       nsplit > 1 ? part : nullptr, ln_g, ln_b, ln_eps, N, K,
                       act_kind);
    if (nsplit > 1)
      hipLaunchKernelGGL(k_gemv_fin2<float>, fg, dim3(256), 0, s, part,
                         (const float*)bias, (const float*)nullptr,
                         (const float*)res, (float*)y, N, nsplit, act_kind);
  } else {
    hipLaunchKernelGGL(k_gemv_nn1<bf16>, g, dim3(256), 0, s, (const bf16*)x,
                       (const bf16*)b, (const float*)nullptr,
                       (const bf16*)bias, (const bf16*)res, (bf16*)y,
                       nsplit > 1 ? part : nullptr, ln_g, ln_b, ln_eps, N, K,
                       act_kind);
    if (nsplit > 1)
      hipLaunchKernelGGL(k_gemv_fin2<bf16>, fg, dim3(256), 0, s, part,
                         (const float*)nullptr, (const bf16*)bias,
                         (const bf16*)res, (bf16*)y, N, nsplit, act_kind);
  }
}

// Legacy two-kernel K-split matvec (kept for K > GEMV1_MAX_K): fp32
// partials then a finalize pass.
template <typename T>
__launch_bounds__(256)
__global__ void k_gemv_nn_part(const T* __restrict__ x, const T* __restrict__ B,
                               float* __restrict__ ws, int N, int K) {
  int n = blockIdx.x * 256 + threadIdx.x;
  int ks = blockIdx.y, KS = gridDim.y;
  if (n >= N) return;
  int k0 = (int)((int64_t)K * ks / KS), k1 = (int)((int64_t)K * (ks + 1) / KS);
  float a0 = 0.0f, a1 = 0.0f;
  int k = k0;
  for (; k + 2 <= k1; k += 2) {
    a0 += VecIO<T>::to_f32(x[k]) * VecIO<T>::to_f32(B[(int64_t)k * N + n]);
    a1 += VecIO<T>::to_f32(x[k + 1]) *
          VecIO<T>::to_f32(B[(int64_t)(k + 1) * N + n]);
  }
  if (k < k1) a0 += VecIO<T>::to_f32(x[k]) * VecIO<T>::to_f32(B[(int64_t)k * N + n]);
  ws[(int64_t)ks * N + n] = a0 + a1;
}

template <typename T>
__launch_bounds__(256)
__global__ void k_gemv_nn_fin(const float* __restrict__ ws,
                              const float* __restrict__ bias_f32,
                              const T* __restrict__ bias_t, T* __restrict__ y,
                              int N, int KS, int act_kind) {
  int n = blockIdx.x * 256 + threadIdx.x;
  if (n >= N) return;
  float a = 0.0f;
  for (int s = 0; s < KS; ++s) a += ws[(int64_t)s * N + n];
  if (bias_f32) a += bias_f32[n];
  if (bias_t) a += VecIO<T>::to_f32(bias_t[n]);
  if (act_kind != ACT_LINEAR) a = act_apply(a, act_kind);
  y[n] = VecIO<T>::from_f32(a);
}

// vector-guaranteed TN gemm (K%V==0, N%V==0, aligned): 3-phase staging --
// addresses (zero-page OOB), all 16B loads, LDS scatter -- keeps RA+RB
// loads in flight where `#pragma unroll 1` serializes to one (see
// k_conv_wgrad_vec).
template <typename T>
__launch_bounds__(THREADS, 3)
__global__ void k_gemm_tn_vec(const T* __restrict__ A, const T* __restrict__ B,
                              float* __restrict__ C,
                              const T* __restrict__ zero16, int M, int N,
                              int K) {
  constexpr int V = 16 / sizeof(T);
  constexpr int RA = BK * (BM / V) / THREADS;
  constexpr int RB = BK * (BN / V) / THREADS;
  __shared__ alignas(16) T As[BM * BK];
  __shared__ alignas(16) T Bs[BN * BK];
  const int r0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const int m_begin = (int)((int64_t)M * blockIdx.z / gridDim.z);
  const int m_end = (int)((int64_t)M * (blockIdx.z + 1) / gridDim.z);
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};
  using VecT = Pack16<T>;

  for (int k0 = m_begin; k0 < m_end; k0 += BK) {
    // bf16: (8-row group x 4 consecutive m) per thread + in-thread
    // transpose -> ds_write_b64 (see k_conv_wgrad_vec)
    constexpr bool TR = sizeof(T) == 2;
    // lane mapping: consecutive lanes share the m-quad and span the row
    // groups: the 16 lanes' 16B global loads form one contiguous 256B run
    // (the conflict-free same-row/mm-strided write mapping was tried and
    // lost more to broken global coalescing than its bank relief won)
    const int a_rr = TR ? (threadIdx.x & (BM / V - 1)) * V : 0;
    const int a_mm0 = TR ? (threadIdx.x / (BM / V)) * 4 : 0;
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
      asrc[it] = (gm < m_end && gr < K) ? &A[(int64_t)gm * K + gr] : zero16;
    }
    const T* bsrc[RB];
#pragma unroll
    for (int it = 0; it < RB; ++it) {
      int c = threadIdx.x + it * THREADS;
      int mm = c / (BN / V);
      int nn = (c % (BN / V)) * V;
      int gm = k0 + mm, gn = n0 + nn;
      bsrc[it] = (gm < m_end && gn < N) ? &B[(int64_t)gm * N + gn] : zero16;
    }
    VecT va[RA], vb[RB];
#pragma unroll
    for (int it = 0; it < RA; ++it) va[it] = *(const VecT*)asrc[it];
#pragma unroll
    for (int it = 0; it < RB; ++it) vb[it] = *(const VecT*)bsrc[it];
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
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  float* out = C + (int64_t)blockIdx.z * K * N;
  epilogue_visit(wc, acc, r0, n0, [&](int row, int col, float v) {
    if (row < K && col < N) out[(int64_t)row * N + col] = v;
  });
}

template <typename OUT>
__global__ void k_splitk_reduce(const float* __restrict__ ws,
                                OUT* __restrict__ out, int z, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float acc = 0.0f;
    for (int s = 0; s < z; ++s) acc += ws[(int64_t)s * n + i];
    out[i] = VecIO<OUT>::from_f32(acc);
  }
}

// float4 variant with 4 independent slab accumulators: the scalar kernel is
// latency-bound (one dependent 4B load per z iteration measured ~1.25 TB/s);
// 16B loads x 4-deep MLP reach the HBM roofline.
template <typename OUT>
__global__ void k_splitk_reduce4(const float4* __restrict__ ws,
                                 OUT* __restrict__ out, int z, int64_t n4) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    float4 a0{0, 0, 0, 0}, a1{0, 0, 0, 0}, a2{0, 0, 0, 0}, a3{0, 0, 0, 0};
    int s = 0;
    for (; s + 4 <= z; s += 4) {
      float4 v0 = ws[(int64_t)s * n4 + i];
      float4 v1 = ws[(int64_t)(s + 1) * n4 + i];
      float4 v2 = ws[(int64_t)(s + 2) * n4 + i];
      float4 v3 = ws[(int64_t)(s + 3) * n4 + i];
      a0.x += v0.x; a0.y += v0.y; a0.z += v0.z; a0.w += v0.w;
      a1.x += v1.x; a1.y += v1.y; a1.z += v1.z; a1.w += v1.w;
      a2.x += v2.x; a2.y += v2.y; a2.z += v2.z; a2.w += v2.w;
      a3.x += v3.x; a3.y += v3.y; a3.z += v3.z; a3.w += v3.w;
    }
    for (; s < z; ++s) {
      float4 v = ws[(int64_t)s * n4 + i];
      a0.x += v.x; a0.y += v.y; a0.z += v.z; a0.w += v.w;
    }
    float r[4];
    r[0] = (a0.x + a1.x) + (a2.x + a3.x);
    r[1] = (a0.y + a1.y) + (a2.y + a3.y);
    r[2] = (a0.z + a1.z) + (a2.z + a3.z);
    r[3] = (a0.w + a1.w) + (a2.w + a3.w);
    struct alignas(4 * sizeof(OUT)) O4 { OUT e[4]; } o;
#pragma unroll
    for (int j = 0; j < 4; ++j) o.e[j] = VecIO<OUT>::from_f32(r[j]);
    ((O4*)out)[i] = o;
  }
}

// tall fold for small outputs with many K-slices (e.g. early-conv wgrad:
// n4 a few hundred, z in the hundreds): the flat kernel gets <16 blocks and
// one serial z-chain per thread. Here 256/CPB thread-rows split z per
// column, then an LDS tree folds the rows — same shape as k_slab_fin.
template <typename OUT, int CPB>
__global__ void k_splitk_fold(const float4* __restrict__ ws,
                              OUT* __restrict__ out, int z, int64_t n4) {
  constexpr int RPB = 256 / CPB;
  __shared__ float4 sh[256];
  const int cl = threadIdx.x % CPB;
  const int r = threadIdx.x / CPB;
  const int64_t c = (int64_t)blockIdx.x * CPB + cl;
  float4 a{0.0f, 0.0f, 0.0f, 0.0f};
  if (c < n4)
    for (int sl = r; sl < z; sl += RPB) {
      float4 v = ws[(int64_t)sl * n4 + c];
      a.x += v.x; a.y += v.y; a.z += v.z; a.w += v.w;
    }
  sh[r * CPB + cl] = a;
  __syncthreads();
  for (int h = RPB / 2; h > 0; h >>= 1) {
    if (r < h) {
      float4 o = sh[(r + h) * CPB + cl];
      float4 m = sh[r * CPB + cl];
      m.x += o.x; m.y += o.y; m.z += o.z; m.w += o.w;
      sh[r * CPB + cl] = m;
    }
    __syncthreads();
  }
  if (r == 0 && c < n4) {
    float4 m = sh[cl];
    struct alignas(4 * sizeof(OUT)) O4 { OUT e[4]; } o;
    o.e[0] = VecIO<OUT>::from_f32(m.x);
    o.e[1] = VecIO<OUT>::from_f32(m.y);
    o.e[2] = VecIO<OUT>::from_f32(m.z);
    o.e[3] = VecIO<OUT>::from_f32(m.w);
    ((O4*)out)[c] = o;
  }
}

template <typename OUT>
static void splitk_fold_launch(const float4* ws, OUT* out, int z, int64_t n4,
                               hipStream_t s) {
#define FOLD_CASE(CPB)                                                    \
  hipLaunchKernelGGL((k_splitk_fold<OUT, CPB>),                           \
                     dim3((unsigned)((n4 + CPB - 1) / CPB)), dim3(256), 0, \
                     s, ws, out, z, n4)
  if (n4 >= 128 * 32) FOLD_CASE(32);
  else if (n4 >= 128 * 16) FOLD_CASE(16);
  else if (n4 >= 128 * 8) FOLD_CASE(8);
  else if (n4 >= 128 * 4) FOLD_CASE(4);
  else if (n4 >= 128 * 2) FOLD_CASE(2);
  else FOLD_CASE(1);
#undef FOLD_CASE
}

void splitk_reduce_launch(const float* ws, void* out, DT out_dt, int z,
                          int64_t n, hipStream_t s) {
  if ((n & 3) == 0 && n < 4 * 32768 && z >= 8) {
    // small output x deep z: parallelize over z (see k_splitk_fold)
    int64_t n4 = n >> 2;
    if (out_dt == DT::F32)
      splitk_fold_launch((const float4*)ws, (float*)out, z, n4, s);
    else
      splitk_fold_launch((const float4*)ws, (bf16*)out, z, n4, s);
    return;
  }
  if ((n & 3) == 0) {
    int64_t n4 = n >> 2;
    int blocks = (int)std::min<int64_t>((n4 + 255) / 256, (int64_t)2048);
    if (out_dt == DT::F32)
      hipLaunchKernelGGL(k_splitk_reduce4<float>, dim3(blocks), dim3(256), 0,
                         s, (const float4*)ws, (float*)out, z, n4);
    else
      hipLaunchKernelGGL(k_splitk_reduce4<bf16>, dim3(blocks), dim3(256), 0,
                         s, (const float4*)ws, (bf16*)out, z, n4);
    return;
  }
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (out_dt == DT::F32)
    hipLaunchKernelGGL(k_splitk_reduce<float>, dim3(blocks), dim3(256), 0, s,
                       ws, (float*)out, z, n);
  else
    hipLaunchKernelGGL(k_splitk_reduce<bf16>, dim3(blocks), dim3(256), 0, s,
                       ws, (bf16*)out, z, n);
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------

void gemm_launch(DT dt, const void* a, const void* b, const void* bias,
                 void* c, const void* zero16, const void* resid, int M, int N,
                 int K, bool trans_b, int act_kind, hipStream_t s) {
  if (M == 1 && trans_b) {  // decode-path matvec
    dim3 gg(ceil_div(N, 4));
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_gemv_nt<float>, gg, dim3(256), 0, s,
                         (const float*)a, (const float*)b, (const float*)bias,
                         (const float*)nullptr, (float*)c, N, K, act_kind);
    else
      hipLaunchKernelGGL(k_gemv_nt<bf16>, gg, dim3(256), 0, s, (const bf16*)a,
                         (const bf16*)b, (const float*)nullptr,
                         (const bf16*)bias, (bf16*)c, N, K, act_kind);
    return;
  }
  dim3 grid(ceil_div(M, BM), ceil_div(N, BN));
  dim3 blk(THREADS);
  if (dt == DT::F32) {
    bool g = K % 4 == 0 && (((uintptr_t)a & 15) == 0);
    if (trans_b && g && K <= 3072 && (((uintptr_t)b & 15) == 0)) {
      hipLaunchKernelGGL(k_gemm_nt_db<float>, grid, blk, 0, s,
                         (const float*)a, (const float*)b, (const float*)bias,
                         (const float*)nullptr, (float*)c,
                         (const float*)zero16, (const float*)resid, M, N, K,
                         act_kind);
      return;
    }
    auto kern = trans_b ? (g ? k_gemm<float, true, true>
                             : k_gemm<float, true, false>)
                        : (g ? k_gemm<float, false, true>
                             : k_gemm<float, false, false>);
    hipLaunchKernelGGL(kern, grid, blk, 0, s, (const float*)a, (const float*)b,
                       (const float*)bias, (const float*)nullptr, (float*)c,
                       (const float*)zero16, (const float*)resid, M, N, K,
                       act_kind);
  } else {
    bool g = K % 8 == 0 && (((uintptr_t)a & 15) == 0);
    if (trans_b && g && K <= 3072 && (((uintptr_t)b & 15) == 0)) {
      hipLaunchKernelGGL(k_gemm_nt_db<bf16>, grid, blk, 0, s, (const bf16*)a,
                         (const bf16*)b, (const float*)nullptr,
                         (const bf16*)bias, (bf16*)c, (const bf16*)zero16,
                         (const bf16*)resid, M, N, K, act_kind);
      return;
    }
    auto kern = trans_b ? (g ? k_gemm<bf16, true, true>
                             : k_gemm<bf16, true, false>)
                        : (g ? k_gemm<bf16, false, true>
                             : k_gemm<bf16, false, false>);
    hipLaunchKernelGGL(kern, grid, blk, 0, s, (const bf16*)a, (const bf16*)b,
                       (const float*)nullptr, (const bf16*)bias, (bf16*)c,
                       (const bf16*)zero16, (const bf16*)resid, M, N, K,
                       act_kind);
  }
}

int gemv_nn_ksplits(int N, int K) {
  int ncb = ceil_div(N, 256);
  int ks = std::max(1, 512 / ncb);
  ks = std::min(ks, std::max(1, K / 16));
  return ks;
}

void gemv_nn_launch(DT dt, const void* x, const void* b, const void* bias,
                    void* y, float* ws, int ks, int N, int K, int act_kind,
                    hipStream_t s) {
  dim3 pg(ceil_div(N, 256), ks);
  dim3 fg(ceil_div(N, 256));
  if (dt == DT::F32) {
    hipLaunchKernelGGL(k_gemv_nn_part<float>, pg, dim3(256), 0, s,
                       (const float*)x, (const float*)b, ws, N, K);
    hipLaunchKernelGGL(k_gemv_nn_fin<float>, fg, dim3(256), 0, s, ws,
                       (const float*)bias, (const float*)nullptr, (float*)y, N,
                       ks, act_kind);
  } else {
    hipLaunchKernelGGL(k_gemv_nn_part<bf16>, pg, dim3(256), 0, s,
                       (const bf16*)x, (const bf16*)b, ws, N, K);
    hipLaunchKernelGGL(k_gemv_nn_fin<bf16>, fg, dim3(256), 0, s, ws,
                       (const float*)nullptr, (const bf16*)bias, (bf16*)y, N,
                       ks, act_kind);
  }
}

int gemm_tn_zsplits(int M, int N, int K) {
  int base = ceil_div(K, BM) * ceil_div(N, BN);
  int want = 2048;
  int z = base >= want ? 1 : std::min(ceil_div(M, BK), ceil_div(want, base));
  return std::max(z, 1);
}

void gemm_tn_launch(DT dt, const void* a, const void* b, void* c_out,
                    DT out_dt, float* ws, int z, const void* zero16, int M,
                    int N, int K, hipStream_t s) {
  dim3 grid(ceil_div(K, BM), ceil_div(N, BN), z);
  bool direct = z == 1 && out_dt == DT::F32;
  float* target = direct ? (float*)c_out : ws;
  if (dt == DT::F32) {
    if (K % 4 == 0 && N % 4 == 0 && (((uintptr_t)a & 15) == 0) &&
        (((uintptr_t)b & 15) == 0))
      hipLaunchKernelGGL(k_gemm_tn_vec<float>, grid, dim3(THREADS), 0, s,
                         (const float*)a, (const float*)b, target,
                         (const float*)zero16, M, N, K);
    else
      hipLaunchKernelGGL(k_gemm_tn<float>, grid, dim3(THREADS), 0, s,
                         (const float*)a, (const float*)b, target, M, N, K);
  } else {
    if (K % 8 == 0 && N % 8 == 0 && (((uintptr_t)a & 15) == 0) &&
        (((uintptr_t)b & 15) == 0))
      hipLaunchKernelGGL(k_gemm_tn_vec<bf16>, grid, dim3(THREADS), 0, s,
                         (const bf16*)a, (const bf16*)b, target,
                         (const bf16*)zero16, M, N, K);
    else
      hipLaunchKernelGGL(k_gemm_tn<bf16>, grid, dim3(THREADS), 0, s,
                         (const bf16*)a, (const bf16*)b, target, M, N, K);
  }
  if (!direct)
    splitk_reduce_launch(ws, c_out, out_dt, z, (int64_t)K * N, s);
}

// ---------------------------------------------------------------------------
// MFMA layout self-test: single-wave D = A@B for the exact fragment maps
// the tile core assumes (run once on real hardware; asymmetric inputs).
// ---------------------------------------------------------------------------

__global__ void k_mfma_selftest(const bf16* A, const bf16* B, float* D) {
  // A [16][32] row-major, B [32][16] row-major, D [16][16]
  int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (__bf16)A[(l & 15) * 32 + (l >> 4) * 8 + j];
    b[j] = (__bf16)B[((l >> 4) * 8 + j) * 16 + (l & 15)];
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) D[((l >> 4) * 4 + j) * 16 + (l & 15)] = acc[j];
}

__global__ void k_mfma_selftest_f32(const float* A, const float* B, float* D) {
  // A [16][4], B [4][16], D [16][16]
  int l = threadIdx.x;
  float a = A[(l & 15) * 4 + (l >> 4)];
  float b = B[(l >> 4) * 16 + (l & 15)];
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) D[((l >> 4) * 4 + j) * 16 + (l & 15)] = acc[j];
}

void mfma_selftest_launch(const void* a, const void* b, float* d,
                          hipStream_t s) {
  hipLaunchKernelGGL(k_mfma_selftest, dim3(1), dim3(64), 0, s, (const bf16*)a,
                     (const bf16*)b, d);
}

void mfma_selftest_f32_launch(const float* a, const float* b, float* d,
                              hipStream_t s) {
  hipLaunchKernelGGL(k_mfma_selftest_f32, dim3(1), dim3(64), 0, s, a, b, d);
}

}  // namespace tnn
