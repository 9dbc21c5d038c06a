// Flash attention forward + backward, hand-written CDNA4 MFMA kernels
// (replaces reference cuDNN-frontend SDPA graphs,
// src/nn/blocks_impl/cuda/cudnn_flash_attention_ops.cu:81-130; the
// reference outsourced both passes — SURVEY §7 hard part 2).
//
// Layout: q/k/v are [B, H, S, D] contiguous bf16, D in {64, 128}.
// Forward (k_attn_fwd): one 256-thread block owns a 64-row Q tile of one
// (b, h); waves own 16 q rows each. Per KV tile of 64: QK^T on MFMA
// (K staged natural [kv][D] — exactly the B^T-operand layout), online
// softmax on the score fragments (row stats via 16-lane shfl_xor),
// P staged per-wave through LDS to re-shape into the A-operand layout,
// then P@V on MFMA with V staged transposed [D][kv]. Outputs O and
// per-row lse for the backward.
//
// Backward (k_attn_bwd): blocks own a KV tile; waves own 16 kv rows and
// accumulate dK/dV in registers across the q-tile loop (no atomics);
// dQ contributions go to an fp32 workspace via atomics. S^T = K Q^T is
// recomputed from the stored lse (no S*S materialization anywhere).
//
// All LDS images use the same XOR swizzle as tile_gemm.h (bank-conflict
// free for both the vector fragment reads and the transpose scatters).

#include "common.h"
#include "kernels.h"
#include "tile_gemm.h"

namespace tnn {
namespace attn {

constexpr int BQ = 64;    // q rows per block (fwd) / q tile (bwd)
constexpr int BKV = 64;   // kv rows per tile
constexpr int THREADS = 256;

using tile::Pack16;
using tile::aligned16;

// LDS element offset with the tile_gemm XOR swizzle over a [rows][C] image
// (C*sizeof(T) >= 128 so the XOR stays in-row).
template <typename T, int C>
DEV int aoff(int row, int col) {
  int byte = col * (int)sizeof(T);
  byte ^= ((((row >> 3) ^ row) & 7) << 4);
  return row * C + byte / (int)sizeof(T);
}

// cooperative stage of a [ROWS][D] bf16 tile from global (row-major,
// row stride D) into the swizzled LDS image via global_load_lds: lane-
// linear 1KiB regions, swizzle applied to the per-lane source address,
// rows >= nrows redirected to the zero page (same scheme as
// tile::glds_stage; see tile_gemm.h).
template <int ROWS, int D>
DEV void stage_tile(const bf16* __restrict__ g, bf16* lds, int nrows,
                    const bf16* __restrict__ zero16) {
  constexpr int RB = D * 2;            // bytes per image row
  constexpr int RPK = 1024 / RB;       // rows per 1 KiB region
  constexpr int LPR = RB / 16;         // lanes per row
  constexpr int NREG = ROWS * RB / 1024;
  constexpr int NPW = NREG / 4;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int i = 0; i < NPW; ++i) {
    const int j = wid * NPW + i;
    const int rl = RPK * j + lane / LPR;
    const int byte_in_row = (lane % LPR) * 16;
    const int col = (byte_in_row ^ ((((rl >> 3) ^ rl) & 7) << 4)) / 2;
    const bf16* src = rl < nrows ? &g[rl * D + col] : zero16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)&lds[j * 512],
        16, 0, 0);
  }
}

// stage transposed: global [ROWS][D] -> LDS image [D][ROWS]
template <int ROWS, int D>
DEV void stage_tile_t(const bf16* __restrict__ g, bf16* lds, int nrows) {
  constexpr int V = 8;
#pragma unroll
  for (int c = threadIdx.x; c < ROWS * (D / V); c += THREADS) {
    int row = c / (D / V);
    int col = (c % (D / V)) * V;
    Pack16<bf16> v = {};
    if (row < nrows) v = *(const Pack16<bf16>*)&g[row * D + col];
#pragma unroll
    for (int j = 0; j < V; ++j) lds[aoff<bf16, ROWS>(col + j, row)] = v.e[j];
  }
}

// S_acc[fn] += A_rows x B_rows^T over D (both images [*][D] swizzled);
// wave computes 16 rows (arow0..+16) x 64 cols.
template <int D, int FN>
DEV void mma_nt(const bf16* Alds, const bf16* Blds, int arow0, int lane,
                f32x4 (&acc)[FN]) {
  const int r = lane & 15;
#pragma unroll
  for (int ks = 0; ks < D / 32; ++ks) {
    const int kb = ks * 32 + (lane >> 4) * 8;
    bf16x8 a = *(const bf16x8*)&Alds[aoff<bf16, D>(arow0 + r, kb)];
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      bf16x8 b = *(const bf16x8*)&Blds[aoff<bf16, D>(fn * 16 + r, kb)];
      acc[fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fn], 0, 0, 0);
    }
  }
}

// O_acc[fo] += P[16 rows x 64] @ V^T-image[D][64]; P rows read at
// arow0 + r within a [*][BKV] swizzled image (absolute-row swizzle keys)
template <int D>
DEV void mma_pv(const bf16* Plds, const bf16* VTlds, int arow0, int lane,
                f32x4 (&acc)[D / 16]) {
  const int r = lane & 15;
#pragma unroll
  for (int ks = 0; ks < BKV / 32; ++ks) {
    const int kb = ks * 32 + (lane >> 4) * 8;
    bf16x8 a = *(const bf16x8*)&Plds[aoff<bf16, BKV>(arow0 + r, kb)];
#pragma unroll
    for (int fo = 0; fo < D / 16; ++fo) {
      bf16x8 b = *(const bf16x8*)&VTlds[aoff<bf16, BKV>(fo * 16 + r, kb)];
      acc[fo] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fo], 0, 0, 0);
    }
  }
}

// write a wave's 16x64 f32 fragment set (C/D layout) into a [16][64]
// bf16 LDS image
template <int FN>
DEV void frag_to_lds(const f32x4 (&acc)[FN], bf16* lds, int lane) {
  const int cr = (lane >> 4) * 4;
  const int cc = lane & 15;
#pragma unroll
  for (int fn = 0; fn < FN; ++fn)
#pragma unroll
    for (int j = 0; j < 4; ++j)
      lds[aoff<bf16, 64>(cr + j, fn * 16 + cc)] = f2bf(acc[fn][j]);
}

// row-wise reduce over the 16 lanes that share a score row
DEV float row_reduce_max(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}
DEV float row_reduce_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, 64);
  return v;
}

// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(THREADS)
__global__ void k_attn_fwd(const bf16* __restrict__ Q, const bf16* __restrict__ K,
                           const bf16* __restrict__ V, bf16* __restrict__ O,
                           float* __restrict__ LSE,
                           const bf16* __restrict__ zero16, int S,
                           float scale) {
  constexpr int FN = BKV / 16;   // 4 score col fragments
  constexpr int FO = D / 16;     // output col fragments
  __shared__ alignas(16) bf16 q_lds[BQ * D];
  __shared__ alignas(16) bf16 k_lds[BKV * D];
  __shared__ alignas(16) bf16 vt_lds[D * BKV];
  __shared__ alignas(16) bf16 p_lds[4][16 * BKV];

  const int q0 = blockIdx.x * BQ;
  const int64_t bh = blockIdx.y;
  const bf16* q = Q + bh * S * D;
  const bf16* k = K + bh * S * D;
  const bf16* v = V + bh * S * D;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wrow = wid * 16;                 // wave's q-row offset in tile

  stage_tile<BQ, D>(q + (int64_t)q0 * D, q_lds, S - q0, zero16);
  // per-lane row stats for the 4 rows this lane's fragments touch share
  // one (m, l) per row; every lane keeps its row's copy (cr group)
  float m_run[4], l_run[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    m_run[j] = -INFINITY;
    l_run[j] = 0.0f;
  }
  f32x4 o_acc[FO] = {};

  const int kv_end = CAUSAL ? min(S, q0 + BQ) : S;
  for (int kv0 = 0; kv0 < kv_end; kv0 += BKV) {
    __syncthreads();
    stage_tile<BKV, D>(k + (int64_t)kv0 * D, k_lds, S - kv0, zero16);
    stage_tile_t<BKV, D>(v + (int64_t)kv0 * D, vt_lds, S - kv0);
    __syncthreads();

    f32x4 s_acc[FN] = {};
    mma_nt<D, FN>(q_lds, k_lds, wrow, lane, s_acc);

    // mask + online softmax on the C/D fragment layout:
    // element (fn, j): row = (lane>>4)*4+j, col = fn*16 + (lane&15)
    const int cr = (lane >> 4) * 4;
    const int cc = lane & 15;
    float p_new[FN][4];
    float alpha[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int grow = q0 + wrow + cr + j;
      float mx = -INFINITY;
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        int gcol = kv0 + fn * 16 + cc;
        float sv = s_acc[fn][j] * scale;
        if (gcol >= S || (CAUSAL && gcol > grow)) sv = -INFINITY;
        p_new[fn][j] = sv;
        mx = fmaxf(mx, sv);
      }
      mx = row_reduce_max(mx);
      float m_new = fmaxf(m_run[j], mx);
      // all-masked row (causal, q < kv0): keep everything unchanged
      float a = (m_new == -INFINITY) ? 1.0f : __expf(m_run[j] - m_new);
      if (m_run[j] == -INFINITY) a = 0.0f;
      if (m_new == -INFINITY) a = 1.0f;
      float rsum = 0.0f;
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        float p = (p_new[fn][j] == -INFINITY) ? 0.0f
                                              : __expf(p_new[fn][j] - m_new);
        p_new[fn][j] = p;
        rsum += p;
      }
      rsum = row_reduce_sum(rsum);
      l_run[j] = l_run[j] * a + rsum;
      m_run[j] = m_new;
      alpha[j] = a;
    }
    // stash P for the PV matmul (per-wave buffer, no cross-wave barrier)
    {
      f32x4 pf[FN];
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
#pragma unroll
        for (int j = 0; j < 4; ++j) pf[fn][j] = p_new[fn][j];
      frag_to_lds<FN>(pf, p_lds[wid], lane);
    }
#pragma unroll
    for (int fo = 0; fo < FO; ++fo)
#pragma unroll
      for (int j = 0; j < 4; ++j) o_acc[fo][j] *= alpha[j];
    // wave-local use of p_lds written by the same wave: needs only an LDS
    // data-dependency wait, which the compiler inserts
    mma_pv<D>(p_lds[wid], vt_lds, 0, lane, o_acc);
  }

  // epilogue: O = O / l; write rows < S; LSE = m + log(l)
  const int cr = (lane >> 4) * 4;
  const int cc = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int grow = q0 + wrow + cr + j;
    if (grow >= S) continue;
    float inv_l = l_run[j] > 0.0f ? 1.0f / l_run[j] : 0.0f;
#pragma unroll
    for (int fo = 0; fo < FO; ++fo)
      O[bh * S * D + (int64_t)grow * D + fo * 16 + cc] =
          f2bf(o_acc[fo][j] * inv_l);
    if (cc == 0)
      LSE[bh * S + grow] =
          l_run[j] > 0.0f ? m_run[j] + __logf(l_run[j]) : -INFINITY;
  }
}

// ---------------------------------------------------------------------------
// Di = rowsum(dO * O) per (b,h,row)
__global__ void k_attn_dot(const bf16* __restrict__ dO,
                           const bf16* __restrict__ O, float* __restrict__ Di,
                           int64_t rows, int D) {
  int64_t row = (int64_t)blockIdx.x * blockDim.x / 64 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  float acc = 0.0f;
  for (int d = lane; d < D; d += 64)
    acc += bf2f(dO[row * D + d]) * bf2f(O[row * D + d]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) Di[row] = acc;
}

// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(THREADS)
__global__ void k_attn_bwd(const bf16* __restrict__ Q, const bf16* __restrict__ K,
                           const bf16* __restrict__ V, const bf16* __restrict__ dO,
                           const float* __restrict__ LSE,
                           const float* __restrict__ Di,
                           float* __restrict__ dQw, bf16* __restrict__ dK,
                           bf16* __restrict__ dV,
                           const bf16* __restrict__ zero16, int S,
                           float scale) {
  constexpr int FN = BQ / 16;    // 4 q-col fragments
  constexpr int FO = D / 16;
  __shared__ alignas(16) bf16 k_lds[BKV * D];
  __shared__ alignas(16) bf16 v_lds[BKV * D];
  __shared__ alignas(16) bf16 kt_lds[D * BKV];
  __shared__ alignas(16) bf16 q_lds[BQ * D];
  __shared__ alignas(16) bf16 qt_lds[D * BQ];
  __shared__ alignas(16) bf16 dot_lds[BQ * D];   // dO natural
  __shared__ alignas(16) bf16 dott_lds[D * BQ];  // dO transposed
  __shared__ alignas(16) bf16 p_lds[4][16 * BQ];   // P^T rows (per wave)
  __shared__ alignas(16) bf16 ds_lds[BQ * BKV];    // dS natural [q][kv]
  __shared__ float lse_lds[BQ];
  __shared__ float di_lds[BQ];

  const int kv0 = blockIdx.x * BKV;
  const int64_t bh = blockIdx.y;
  const bf16* q = Q + bh * S * D;
  const bf16* k = K + bh * S * D;
  const bf16* v = V + bh * S * D;
  const bf16* dout = dO + bh * S * D;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wrow = wid * 16;                 // wave's kv-row offset

  stage_tile<BKV, D>(k + (int64_t)kv0 * D, k_lds, S - kv0, zero16);
  stage_tile<BKV, D>(v + (int64_t)kv0 * D, v_lds, S - kv0, zero16);
  stage_tile_t<BKV, D>(k + (int64_t)kv0 * D, kt_lds, S - kv0);

  f32x4 dk_acc[FO] = {}, dv_acc[FO] = {};

  const int q_start = CAUSAL ? (kv0 / BQ) * BQ : 0;
  for (int qt = q_start; qt < S; qt += BQ) {
    __syncthreads();
    stage_tile<BQ, D>(q + (int64_t)qt * D, q_lds, S - qt, zero16);
    stage_tile_t<BQ, D>(q + (int64_t)qt * D, qt_lds, S - qt);
    stage_tile<BQ, D>(dout + (int64_t)qt * D, dot_lds, S - qt, zero16);
    stage_tile_t<BQ, D>(dout + (int64_t)qt * D, dott_lds, S - qt);
    for (int i = threadIdx.x; i < BQ; i += THREADS) {
      int g = qt + i;
      lse_lds[i] = g < S ? LSE[bh * S + g] : INFINITY;
      di_lds[i] = g < S ? Di[bh * S + g] : 0.0f;
    }
    __syncthreads();

    // S^T[kv][q] = K Q^T ; P^T = exp(S^T*scale - lse[q])
    f32x4 st_acc[FN] = {};
    mma_nt<D, FN>(k_lds, q_lds, wrow, lane, st_acc);
    const int cr = (lane >> 4) * 4;
    const int cc = lane & 15;
    float pt[FN][4];
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int grow = kv0 + wrow + cr + j;      // kv index
        int gcol = qt + fn * 16 + cc;        // q index
        float sv = st_acc[fn][j] * scale;
        bool dead = gcol >= S || grow >= S || (CAUSAL && gcol < grow);
        float lse = lse_lds[fn * 16 + cc];
        pt[fn][j] = dead ? 0.0f : __expf(sv - lse);
      }
    // dV += P^T @ dO  (A = P^T from LDS, B = dO^T image)
    {
      f32x4 pf[FN];
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
#pragma unroll
        for (int j = 0; j < 4; ++j) pf[fn][j] = pt[fn][j];
      frag_to_lds<FN>(pf, p_lds[wid], lane);
    }
    mma_pv<D>(p_lds[wid], dott_lds, 0, lane, dv_acc);

    // dP^T[kv][q] = V dO^T
    f32x4 dpt_acc[FN] = {};
    mma_nt<D, FN>(v_lds, dot_lds, wrow, lane, dpt_acc);

    // dS^T = P^T * (dP^T - Di[q]) * scale
    float dst[FN][4];
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        dst[fn][j] = pt[fn][j] * (dpt_acc[fn][j] - di_lds[fn * 16 + cc]) * scale;

    // dK += dS^T @ Q  (A = dS^T via LDS, B = Q^T image)
    {
      f32x4 df[FN];
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
#pragma unroll
        for (int j = 0; j < 4; ++j) df[fn][j] = dst[fn][j];
      frag_to_lds<FN>(df, p_lds[wid], lane);
    }
    mma_pv<D>(p_lds[wid], qt_lds, 0, lane, dk_acc);

    // dQ[q] += dS[q][kv] @ K[kv][d]: build the dS natural image (scatter
    // from the dS^T fragments), then each wave computes 16 q rows
    __syncthreads();  // protect ds_lds reuse across iterations
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int kvr = wrow + cr + j;
        int qc = fn * 16 + cc;
        ds_lds[aoff<bf16, BKV>(qc, kvr)] = f2bf(dst[fn][j]);
      }
    __syncthreads();
    f32x4 dq_acc[FO] = {};
    mma_pv<D>(ds_lds, kt_lds, wrow, lane, dq_acc);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int grow = qt + wrow + cr + j;
      if (grow >= S) continue;
#pragma unroll
      for (int fo = 0; fo < FO; ++fo)
        atomicAdd(&dQw[bh * S * D + (int64_t)grow * D + fo * 16 + cc],
                  dq_acc[fo][j]);
    }
  }

  // write dK, dV (each kv row owned by exactly this block's wave)
  const int cr = (lane >> 4) * 4;
  const int cc = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int grow = kv0 + wrow + cr + j;
    if (grow >= S) continue;
#pragma unroll
    for (int fo = 0; fo < FO; ++fo) {
      dK[bh * S * D + (int64_t)grow * D + fo * 16 + cc] = f2bf(dk_acc[fo][j]);
      dV[bh * S * D + (int64_t)grow * D + fo * 16 + cc] = f2bf(dv_acc[fo][j]);
    }
  }
}

}  // namespace attn
}  // namespace tnn

namespace tnn {

using namespace attn;

void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, const void* zero16, int BH, int S, int D,
                     bool causal, float scale, hipStream_t s) {
  dim3 grid((S + BQ - 1) / BQ, BH);
#define LAUNCH(DD, CC)                                                     \
  hipLaunchKernelGGL((k_attn_fwd<DD, CC>), grid, dim3(THREADS), 0, s,      \
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,       \
                     (bf16*)o, lse, (const bf16*)zero16, S, scale)
  if (D == 64) { if (causal) LAUNCH(64, true); else LAUNCH(64, false); }
  else if (D == 128) { if (causal) LAUNCH(128, true); else LAUNCH(128, false); }
#undef LAUNCH
}

void attn_bwd_launch(const void* q, const void* k, const void* v,
                     const void* o, const void* dout, const float* lse,
                     float* di, float* dq_ws, void* dk, void* dv,
                     const void* zero16, int BH, int S, int D, bool causal,
                     float scale, hipStream_t s) {
  int64_t rows = (int64_t)BH * S;
  hipLaunchKernelGGL(k_attn_dot, dim3((rows * 64 + 255) / 256), dim3(256), 0,
                     s, (const bf16*)dout, (const bf16*)o, di, rows, D);
  dim3 grid((S + BKV - 1) / BKV, BH);
#define LAUNCH(DD, CC)                                                     \
  hipLaunchKernelGGL((k_attn_bwd<DD, CC>), grid, dim3(THREADS), 0, s,      \
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,       \
                     (const bf16*)dout, lse, di, dq_ws, (bf16*)dk,         \
                     (bf16*)dv, (const bf16*)zero16, S, scale)
  if (D == 64) { if (causal) LAUNCH(64, true); else LAUNCH(64, false); }
  else if (D == 128) { if (causal) LAUNCH(128, true); else LAUNCH(128, false); }
#undef LAUNCH
}

}  // namespace tnn
