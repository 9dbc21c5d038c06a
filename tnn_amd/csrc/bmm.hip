// Strided-batched MFMA GEMM — the substrate for the materialized-scores
// attention family (reference gemm_strided_batched_ex,
// src/math/cuda/gemm.cu:84-105, and the per-head batched QK^T / scores.V
// GEMMs in src/nn/blocks_impl/attention_block.cpp:144-147).
//
// One kernel, four operand layouts (TA/TB), same MFMA tile core and
// XOR-swizzled LDS images as tile_gemm.h:
//   A logical [M,K]: TA=0 element (m,k) at A[m*lda+k] (rows k-contiguous),
//                    TA=1 at A[k*lda+m] (a transpose view; scatter-staged)
//   B logical [K,N]: TB=0 element (k,n) at B[k*ldb+n] (scatter-staged),
//                    TB=1 at B[n*ldb+k] (rows k-contiguous — the QK^T form)
//   C [M,N] natural, row stride ldc; per-matrix strides sa/sb/sc.
// Grid: (ceil(M/BM), ceil(N/BN), batch).

#include "common.h"
#include "kernels.h"
#include "tile_gemm.h"

namespace tnn {

using namespace tile;

template <typename T, bool TA, bool TB>
__launch_bounds__(THREADS)
__global__ void k_bmm(const T* __restrict__ A, const T* __restrict__ B,
                      T* __restrict__ C, int M, int N, int K, int lda, int ldb,
                      int ldc, int64_t sa, int64_t sb, int64_t sc) {
  constexpr int V = 16 / sizeof(T);
  using VecT = Pack16<T>;
  __shared__ alignas(16) T As[BM * BK];
  __shared__ alignas(16) T Bs[BN * BK];

  const T* a = A + blockIdx.z * sa;
  const T* b = B + blockIdx.z * sb;
  T* c = C + blockIdx.z * sc;

  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A tile [BM][BK] ----
    if constexpr (!TA) {
      // rows k-contiguous: vectorized row loads
#pragma unroll
      for (int t = threadIdx.x; t < BM * (BK / V); t += THREADS) {
        int row = t / (BK / V);
        int kk = (t % (BK / V)) * V;
        int gm = m0 + row, gk = k0 + kk;
        VecT v = {};
        if (gm < M) {
          const T* src = &a[(int64_t)gm * lda + gk];
          if (gk + V <= K && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gk + j < K) v.e[j] = a[(int64_t)gm * lda + gk + j];
          }
        }
        *(VecT*)&As[lds_off<T>(row, kk)] = v;
      }
    } else {
      // memory m-contiguous for fixed k: load m-vectors, scatter-transpose
#pragma unroll 1
      for (int t = threadIdx.x; t < BK * (BM / V); t += THREADS) {
        int kk = t / (BM / V);
        int mm = (t % (BM / V)) * V;
        int gk = k0 + kk, gm = m0 + mm;
        VecT v = {};
        if (gk < K) {
          const T* src = &a[(int64_t)gk * lda + gm];
          if (gm + V <= M && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gm + j < M) v.e[j] = a[(int64_t)gk * lda + gm + j];
          }
        }
#pragma unroll
        for (int j = 0; j < V; ++j) As[lds_off<T>(mm + j, kk)] = v.e[j];
      }
    }
    // ---- stage B tile Bs[n][k] ----
    if constexpr (TB) {
      // B^T memory: n rows k-contiguous — direct image rows
#pragma unroll
      for (int t = threadIdx.x; t < BN * (BK / V); t += THREADS) {
        int col = t / (BK / V);
        int kk = (t % (BK / V)) * V;
        int gn = n0 + col, gk = k0 + kk;
        VecT v = {};
        if (gn < N) {
          const T* src = &b[(int64_t)gn * ldb + gk];
          if (gk + V <= K && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gk + j < K) v.e[j] = b[(int64_t)gn * ldb + gk + j];
          }
        }
        *(VecT*)&Bs[lds_off<T>(col, kk)] = v;
      }
    } else {
      // B[K,N]: n-contiguous vectors, scatter-transpose into Bs[n][k]
#pragma unroll 1
      for (int t = threadIdx.x; t < BK * (BN / V); t += THREADS) {
        int kk = t / (BN / V);
        int nn = (t % (BN / V)) * V;
        int gk = k0 + kk, gn = n0 + nn;
        VecT v = {};
        if (gk < K) {
          const T* src = &b[(int64_t)gk * ldb + gn];
          if (gn + V <= N && aligned16(src)) {
            v = *(const VecT*)src;
          } else {
#pragma unroll
            for (int j = 0; j < V; ++j)
              if (gn + j < N) v.e[j] = b[(int64_t)gk * ldb + gn + j];
          }
        }
#pragma unroll
        for (int j = 0; j < V; ++j) Bs[lds_off<T>(nn + j, kk)] = v.e[j];
      }
    }
    __syncthreads();
    mfma_compute_tile(As, Bs, wc, acc);
    __syncthreads();
  }

  epilogue_visit(wc, acc, m0, n0, [&](int row, int col, float v) {
    if (row < M && col < N)
      c[(int64_t)row * ldc + col] = VecIO<T>::from_f32(v);
  });
}

// double-buffered glds variant for the NT layout (both operands' rows
// k-contiguous — the QK^T / dO@V^T attention shapes): tile t+1's
// LDS-DMAs overlap tile t's MFMAs (same structure as k_gemm_nt_db).
template <typename T>
__launch_bounds__(THREADS)
__global__ void k_bmm_nt_db(const T* __restrict__ A, const T* __restrict__ B,
                            T* __restrict__ C, const T* __restrict__ zero16,
                            int M, int N, int K, int lda, int ldb, int ldc,
                            int64_t sa, int64_t sb, int64_t sc) {
  __shared__ alignas(16) T As[2][BM * BK];
  __shared__ alignas(16) T Bs[2][BN * BK];
  const T* a = A + blockIdx.z * sa;
  const T* b = B + blockIdx.z * sb;
  T* c = C + blockIdx.z * sc;
  const int m0 = blockIdx.x * BM;
  const int n0 = blockIdx.y * BN;
  const WaveCoord wc;
  f32x4 acc[FM][FN] = {};

  auto stage = [&](int t, int which) {
    const int kk0 = t * BK;
    glds_stage<T, BM>(As[which], wc, [&](int rl, int kk) -> const T* {
      int gm = m0 + rl, gk = kk0 + kk;
      if (gm >= M || gk >= K) return zero16;
      return &a[(int64_t)gm * lda + gk];
    });
    glds_stage<T, BN>(Bs[which], wc, [&](int rl, int kk) -> const T* {
      int gn = n0 + rl, gk = kk0 + kk;
      if (gn >= N || gk >= K) return zero16;
      return &b[(int64_t)gn * ldb + gk];
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
    if (row < M && col < N)
      c[(int64_t)row * ldc + col] = VecIO<T>::from_f32(v);
  });
}

void bmm_launch(DT dt, const void* a, const void* b, void* c,
                const void* zero16, int batch, int M, int N, int K, int lda,
                int ldb, int ldc, int64_t sa, int64_t sb, int64_t sc, bool ta,
                bool tb, hipStream_t s) {
  dim3 grid(ceil_div(M, BM), ceil_div(N, BN), batch);
  dim3 blk(THREADS);
  // NT fast path: rows of both operands k-contiguous and 16B-clean
  if (!ta && tb) {
    const int V = dt == DT::F32 ? 4 : 8;
    const bool clean = K % V == 0 && lda % V == 0 && ldb % V == 0 &&
                       sa % V == 0 && sb % V == 0 &&
                       (((uintptr_t)a & 15) == 0) && (((uintptr_t)b & 15) == 0);
    if (clean && K <= 3072) {
      if (dt == DT::F32)
        hipLaunchKernelGGL(k_bmm_nt_db<float>, grid, blk, 0, s,
                           (const float*)a, (const float*)b, (float*)c,
                           (const float*)zero16, M, N, K, lda, ldb, ldc, sa,
                           sb, sc);
      else
        hipLaunchKernelGGL(k_bmm_nt_db<bf16>, grid, blk, 0, s,
                           (const bf16*)a, (const bf16*)b, (bf16*)c,
                           (const bf16*)zero16, M, N, K, lda, ldb, ldc, sa,
                           sb, sc);
      return;
    }
  }
#define LAUNCH_T(T)                                                          \
  do {                                                                       \
    auto kern = ta ? (tb ? k_bmm<T, true, true> : k_bmm<T, true, false>)     \
                   : (tb ? k_bmm<T, false, true> : k_bmm<T, false, false>);  \
    hipLaunchKernelGGL(kern, grid, blk, 0, s, (const T*)a, (const T*)b,      \
                       (T*)c, M, N, K, lda, ldb, ldc, sa, sb, sc);           \
  } while (0)
  if (dt == DT::F32) LAUNCH_T(float);
  else LAUNCH_T(bf16);
#undef LAUNCH_T
}

}  // namespace tnn
