// Flash attention forward + backward, hand-written CDNA4 MFMA kernels
// (replaces reference cuDNN-frontend SDPA graphs,
// src/nn/blocks_impl/cuda/cudnn_flash_attention_ops.cu:81-130; the
// reference outsourced both passes — SURVEY §7 hard part 2).
//
// Layout: q/k/v are [B, H, S, D] contiguous bf16, D in {64, 128}.
//
// Forward (k_attn_fwd, round-2 structure): one 512-thread / 8-wave block
// owns a 128-row Q tile of one (b, h); each wave owns 16 q rows, so one
// K/V staging pass feeds 128 q rows (2x the round-1 4-wave/64-row
// structure). Q is staged once through LDS and hoisted into per-wave
// MFMA A-fragments in REGISTERS for the whole KV loop (no Q re-reads);
// its LDS buffer is then re-used as the 8 per-wave P buffers. Per KV
// tile of 64: K staged natural [kv][D] via global_load_lds (the
// B^T-operand layout), V transposed to [D][kv] with an in-thread 4x8
// transpose + one ds_write_b64 per d-row — NOT a per-element b16
// scatter (the round-1 LDS-conflict source, PERFORMANCE.md ladder 18's
// wgrad trick applied here). QK^T on MFMA, online softmax on the score
// fragments, P staged per-wave to LDS, then P@V on MFMA.
//
// Backward (k_attn_bwd): blocks own a KV tile; waves own 16 kv rows and
// accumulate dK/dV in registers across the q-tile loop (no atomics);
// dQ contributions go to an fp32 workspace via atomics. S^T = K Q^T is
// recomputed from the stored lse (no S*S materialization anywhere).
// All transposed stagings use the same b64 transpose-pack; the dS
// natural-image scatter packs 4 kv-contiguous values per ds_write_b64.
//
// LDS images use the tile_gemm XOR swizzle (bank-conflict free for the
// vector fragment reads and the b64 transpose writes).

#include "common.h"
#include "kernels.h"
#include "tile_gemm.h"

namespace tnn {
namespace attn {

constexpr int BKV = 64;   // kv rows per tile
constexpr int BWD_BQ = 64;     // q tile (bwd)
constexpr int BWD_THREADS = 256;

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
// tile::glds_stage; see tile_gemm.h). NW = waves in the block.
template <int ROWS, int D, int NW>
DEV void stage_tile(const bf16* __restrict__ g, bf16* lds, int nrows,
                    const bf16* __restrict__ zero16, int64_t rs = D) {
  constexpr int RB = D * 2;            // bytes per image row
  constexpr int RPK = 1024 / RB;       // rows per 1 KiB region
  constexpr int LPR = RB / 16;         // lanes per row
  constexpr int NREG = ROWS * RB / 1024;
  constexpr int NPW = (NREG + NW - 1) / NW;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int i = 0; i < NPW; ++i) {
    const int j = wid * NPW + i;
    if (NREG % NW != 0 && j >= NREG) break;
    const int rl = RPK * j + lane / LPR;
    const int byte_in_row = (lane % LPR) * 16;
    const int col = (byte_in_row ^ ((((rl >> 3) ^ rl) & 7) << 4)) / 2;
    const bf16* src = rl < nrows ? &g[rl * rs + col] : zero16;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)&lds[j * 512],
        16, 0, 0);
  }
}

// stage transposed: global [ROWS][D] -> LDS image [D][ROWS] with an
// in-thread 4x8 transpose: each active thread loads 8 d-contiguous bf16
// from 4 consecutive kv rows and writes 8 ds_write_b64 quads (4
// kv-contiguous values at one d row) — 4x fewer LDS instructions and
// conflict-free groups vs the per-element b16 scatter. Split into
// load()/write() halves (T14 issue-early / write-late): the loads go
// out before a compute phase, the LDS writes land after it.
template <int ROWS, int D, int NT>
struct TStage {
  static constexpr int CH = (ROWS / 4) * (D / 8);  // (4-row x 8-col) chunks
  static constexpr int PER = (CH + NT - 1) / NT;
  Pack16<bf16> v[PER][4];

  DEV void load(const bf16* __restrict__ g, int nrows, int64_t rs = D) {
#pragma unroll
    for (int i = 0; i < PER; ++i) {
      const int c = threadIdx.x + i * NT;
      if (CH % NT != 0 && c >= CH) break;
      const int kv0 = (c / (D / 8)) * 4;
      const int d0 = (c % (D / 8)) * 8;
#pragma unroll
      for (int q = 0; q < 4; ++q) {
        if (kv0 + q < nrows)
          v[i][q] = *(const Pack16<bf16>*)&g[(kv0 + q) * rs + d0];
        else
          v[i][q] = {};
      }
    }
  }

  DEV void write(bf16* lds) {
    struct alignas(8) H4 { bf16 e[4]; };
#pragma unroll
    for (int i = 0; i < PER; ++i) {
      const int c = threadIdx.x + i * NT;
      if (CH % NT != 0 && c >= CH) break;
      const int kv0 = (c / (D / 8)) * 4;
      const int d0 = (c % (D / 8)) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        H4 h;
#pragma unroll
        for (int q = 0; q < 4; ++q) h.e[q] = v[i][q].e[j];
        *(H4*)&lds[aoff<bf16, ROWS>(d0 + j, kv0)] = h;
      }
    }
  }
};

template <int ROWS, int D, int NT>
DEV void stage_tile_t(const bf16* __restrict__ g, bf16* lds, int nrows,
                      int64_t rs = D) {
  TStage<ROWS, D, NT> st;
  st.load(g, nrows, rs);
  st.write(lds);
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

// same with the A fragments already hoisted to registers
template <int D, int FN>
DEV void mma_nt_areg(const bf16x8 (&a)[D / 32], const bf16* Blds, int lane,
                     f32x4 (&acc)[FN]) {
  const int r = lane & 15;
#pragma unroll
  for (int ks = 0; ks < D / 32; ++ks) {
    const int kb = ks * 32 + (lane >> 4) * 8;
#pragma unroll
    for (int fn = 0; fn < FN; ++fn) {
      bf16x8 b = *(const bf16x8*)&Blds[aoff<bf16, D>(fn * 16 + r, kb)];
      acc[fn] =
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[ks], b, acc[fn], 0, 0, 0);
    }
  }
}

// O_acc[fo] += P[16 rows x 64] @ V^T-image[NF*16 rows][64]; P rows read
// at arow0 + r within a [*][BKV] swizzled image (absolute-row swizzle
// keys). NF may exceed D/16: the fwd kernel appends a ones-row fragment
// so the softmax row-sum rides the same MFMAs (l = acc[NF-1] col 0).
template <int NF>
DEV void mma_pv(const bf16* Plds, const bf16* VTlds, int arow0, int lane,
                f32x4 (&acc)[NF]) {
  const int r = lane & 15;
#pragma unroll
  for (int ks = 0; ks < BKV / 32; ++ks) {
    const int kb = ks * 32 + (lane >> 4) * 8;
    bf16x8 a = *(const bf16x8*)&Plds[aoff<bf16, BKV>(arow0 + r, kb)];
#pragma unroll
    for (int fo = 0; fo < NF; ++fo) {
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
// NW waves own 16 q rows each (BQ = NW*16); one K/V staging pass per KV
// tile feeds the whole block's q rows. NW=4 keeps 3 blocks/CU resident
// (latency hiding via TLP), NW=8 halves K/V restaging per q row.
template <int D, bool CAUSAL, int NW>
__launch_bounds__(NW * 64)
__global__ void k_attn_fwd(const bf16* __restrict__ Q, const bf16* __restrict__ K,
                           const bf16* __restrict__ V, bf16* __restrict__ O,
                           float* __restrict__ LSE,
                           const bf16* __restrict__ zero16, int S, int H,
                           int64_t i_rs, int64_t i_hs, int64_t i_bs,
                           int64_t o_rs, int64_t o_hs, int64_t o_bs,
                           float scale) {
  constexpr int BQ = NW * 16;    // q rows per block
  constexpr int THREADS = NW * 64;
  constexpr int FN = BKV / 16;   // 4 score col fragments
  constexpr int FO = D / 16;     // output col fragments
  // q_lds (BQ*D) is read once into registers, then re-used as the NW
  // per-wave P buffers (NW * 16 * BKV = BQ * 64 <= BQ * D elements).
  // K and V^T are double-buffered: tile t+1's K LDS-DMAs and V register
  // loads are issued before tile t's compute and land after it (glds
  // 2-buffer + T14 split — the stall-bound single-buffer chain measured
  // WAIT_ANY 0.52-0.61, profiles/pmc_attention_r1.md successor).
  __shared__ alignas(16) bf16 q_lds[BQ * D];
  __shared__ alignas(16) bf16 k_lds[2][BKV * D];
  // V^T image carries 16 extra rows: row D is all-ones, D+1..D+15 zero —
  // the PV MFMA's last fragment then computes the softmax row-sum in its
  // col 0, deleting the dependent __shfl sum chain entirely (l rides the
  // same alpha-rescale recurrence as O).
  __shared__ alignas(16) bf16 vt_lds[2][(D + 16) * BKV];
  bf16* p_lds = q_lds;  // aliased after the Q fragments are hoisted

  const int q0 = blockIdx.x * BQ;
  const int64_t bh = blockIdx.y;
  // q/k/v share one stride tuple (BHSD-contiguous, a BSHD view, or three
  // slices of a merged [B,S,3*H*D] QKV buffer -- bases differ, strides
  // don't); rows stay d-contiguous in every layout
  const int64_t ibase = (bh / H) * i_bs + (bh % H) * i_hs;
  const bf16* q = Q + ibase;
  const bf16* k = K + ibase;
  const bf16* v = V + ibase;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wrow = wid * 16;                 // wave's q-row offset in tile

  stage_tile<BQ, D, NW>(q + (int64_t)q0 * i_rs, q_lds, S - q0, zero16, i_rs);
  // constant ones/zero tail rows of both V^T buffers (written once)
  for (int i = threadIdx.x; i < 16 * BKV; i += THREADS) {
    const int rr = i / BKV, col = i % BKV;
    const bf16 val = f2bf(rr == 0 ? 1.0f : 0.0f);
    vt_lds[0][aoff<bf16, BKV>(D + rr, col)] = val;
    vt_lds[1][aoff<bf16, BKV>(D + rr, col)] = val;
  }
  __syncthreads();
  // hoist this wave's Q fragments into registers for the whole KV loop
  bf16x8 q_frag[D / 32];
  {
    const int r = lane & 15;
#pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      const int kb = ks * 32 + (lane >> 4) * 8;
      q_frag[ks] = *(const bf16x8*)&q_lds[aoff<bf16, D>(wrow + r, kb)];
    }
  }
  __syncthreads();  // q_lds free -> p_lds

  float m_run[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) m_run[j] = -INFINITY;
  f32x4 o_acc[FO + 1] = {};  // acc[FO] col 0 = running softmax denominator

  const int kv_end = CAUSAL ? min(S, q0 + BQ) : S;
  const int ntiles = (kv_end + BKV - 1) / BKV;
  TStage<BKV, D, THREADS> vst;

  // prologue: stage tile 0 into buffer 0
  stage_tile<BKV, D, NW>(k, k_lds[0], S, zero16, i_rs);
  vst.load(v, S, i_rs);
  vst.write(vt_lds[0]);
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int cur = t & 1;
    const int kv0 = t * BKV;
    if (t + 1 < ntiles) {
      // issue tile t+1's K LDS-DMAs and V loads; they stay in flight
      // under this tile's compute (no ordinary-load USE until the
      // write at the bottom, so hipcc keeps them outstanding)
      stage_tile<BKV, D, NW>(k + (int64_t)(kv0 + BKV) * i_rs, k_lds[cur ^ 1],
                             S - kv0 - BKV, zero16, i_rs);
      vst.load(v + (int64_t)(kv0 + BKV) * i_rs, S - kv0 - BKV, i_rs);
    }

    f32x4 s_acc[FN] = {};
    mma_nt_areg<D, FN>(q_frag, k_lds[cur], lane, s_acc);

    // mask + online softmax IN PLACE on the C/D score fragments:
    // element (fn, j): row = (lane>>4)*4+j, col = fn*16 + (lane&15)
    const int cr = (lane >> 4) * 4;
    const int cc = lane & 15;
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
        s_acc[fn][j] = sv;
        mx = fmaxf(mx, sv);
      }
      mx = row_reduce_max(mx);
      float m_new = fmaxf(m_run[j], mx);
      // all-masked row (causal, q < kv0): keep everything unchanged
      float a = (m_new == -INFINITY) ? 1.0f : __expf(m_run[j] - m_new);
      if (m_run[j] == -INFINITY) a = 0.0f;
      if (m_new == -INFINITY) a = 1.0f;
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        float p = (s_acc[fn][j] == -INFINITY) ? 0.0f
                                              : __expf(s_acc[fn][j] - m_new);
        s_acc[fn][j] = p;
      }
      m_run[j] = m_new;
      alpha[j] = a;
    }
    // stash P for the PV matmul (per-wave buffer, no cross-wave barrier)
    frag_to_lds<FN>(s_acc, p_lds + wid * 16 * BKV, lane);
#pragma unroll
    for (int fo = 0; fo <= FO; ++fo)
#pragma unroll
      for (int j = 0; j < 4; ++j) o_acc[fo][j] *= alpha[j];
    // wave-local use of p_lds written by the same wave: needs only an LDS
    // data-dependency wait, which the compiler inserts
    mma_pv<FO + 1>(p_lds + wid * 16 * BKV, vt_lds[cur], 0, lane, o_acc);

    if (t + 1 < ntiles)
      vst.write(vt_lds[cur ^ 1]);  // waits the in-flight V loads here
    __syncthreads();               // drains the K glds; releases buffers
  }

  // epilogue: O = O / l; write rows < S; LSE = m + log(l)
  const int cr = (lane >> 4) * 4;
  const int cc = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    // broadcast the denominator (held in col 0 of the l-fragment) to the
    // row's 16 lanes
    float l = __shfl(o_acc[FO][j], lane & 48, 64);
    const int grow = q0 + wrow + cr + j;
    if (grow >= S) continue;
    float inv_l = l > 0.0f ? 1.0f / l : 0.0f;
#pragma unroll
    for (int fo = 0; fo < FO; ++fo)
      O[(bh / H) * o_bs + (bh % H) * o_hs + (int64_t)grow * o_rs + fo * 16 +
        cc] = f2bf(o_acc[fo][j] * inv_l);
    if (cc == 0)
      LSE[bh * S + grow] = l > 0.0f ? m_run[j] + __logf(l) : -INFINITY;
  }
}

// ---------------------------------------------------------------------------
// Di = rowsum(dO * O) per (b,h,row)
__global__ void k_attn_dot(const bf16* __restrict__ dO,
                           const bf16* __restrict__ O, float* __restrict__ Di,
                           int64_t rows, int S, int H, int D,
                           int64_t do_rs, int64_t do_hs, int64_t do_bs,
                           int64_t o_rs, int64_t o_hs, int64_t o_bs) {
  int64_t row = (int64_t)blockIdx.x * blockDim.x / 64 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const int64_t bh = row / S, sr = row % S;
  const bf16* dop = dO + (bh / H) * do_bs + (bh % H) * do_hs + sr * do_rs;
  const bf16* op = O + (bh / H) * o_bs + (bh % H) * o_hs + sr * o_rs;
  float acc = 0.0f;
  for (int d = lane; d < D; d += 64) acc += bf2f(dop[d]) * bf2f(op[d]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) Di[row] = acc;
}

// ---------------------------------------------------------------------------
template <int D, bool CAUSAL>
__launch_bounds__(BWD_THREADS)
__global__ void k_attn_bwd(const bf16* __restrict__ Q, const bf16* __restrict__ K,
                           const bf16* __restrict__ V, const bf16* __restrict__ dO,
                           const float* __restrict__ LSE,
                           const float* __restrict__ Di,
                           float* __restrict__ dQw, bf16* __restrict__ dK,
                           bf16* __restrict__ dV,
                           const bf16* __restrict__ zero16, int S, int H,
                           int64_t i_rs, int64_t i_hs, int64_t i_bs,
                           int64_t do_rs, int64_t do_hs, int64_t do_bs,
                           int64_t w_rs, int64_t w_hs, int64_t w_bs,
                           float scale) {
  constexpr int FN = BWD_BQ / 16;  // 4 q-col fragments
  constexpr int FO = D / 16;
  constexpr int NW = BWD_THREADS / 64;
  __shared__ alignas(16) bf16 k_lds[BKV * D];
  __shared__ alignas(16) bf16 v_lds[BKV * D];
  __shared__ alignas(16) bf16 kt_lds[D * BKV];
  __shared__ alignas(16) bf16 q_lds[BWD_BQ * D];
  __shared__ alignas(16) bf16 qt_lds[D * BWD_BQ];
  __shared__ alignas(16) bf16 dot_lds[BWD_BQ * D];   // dO natural
  __shared__ alignas(16) bf16 dott_lds[D * BWD_BQ];  // dO transposed
  __shared__ alignas(16) bf16 p_lds[4][16 * BWD_BQ];   // P^T rows (per wave)
  __shared__ alignas(16) bf16 ds_lds[BWD_BQ * BKV];    // dS natural [q][kv]
  __shared__ float lse_lds[BWD_BQ];
  __shared__ float di_lds[BWD_BQ];

  const int kv0 = blockIdx.x * BKV;
  const int64_t bh = blockIdx.y;
  const int64_t ibase = (bh / H) * i_bs + (bh % H) * i_hs;
  const bf16* q = Q + ibase;
  const bf16* k = K + ibase;
  const bf16* v = V + ibase;
  const bf16* dout = dO + (bh / H) * do_bs + (bh % H) * do_hs;
  // dK/dV land strided (w_* tuple: BHSD-dense, a BSHD buffer, or slices
  // of one merged [B,S,3*H*D] dQKV buffer)
  bf16* dkp = dK + (bh / H) * w_bs + (bh % H) * w_hs;
  bf16* dvp = dV + (bh / H) * w_bs + (bh % H) * w_hs;
  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wrow = wid * 16;                 // wave's kv-row offset

  stage_tile<BKV, D, NW>(k + (int64_t)kv0 * i_rs, k_lds, S - kv0, zero16,
                         i_rs);
  stage_tile<BKV, D, NW>(v + (int64_t)kv0 * i_rs, v_lds, S - kv0, zero16,
                         i_rs);
  stage_tile_t<BKV, D, BWD_THREADS>(k + (int64_t)kv0 * i_rs, kt_lds, S - kv0,
                                    i_rs);

  // hoist this wave's K/V MFMA A-fragments into registers for the whole
  // q-tile loop (re-read from LDS every tile otherwise). D==64 only:
  // at D==128 the +32 VGPR would push past the 256 cap and spill.
  bf16x8 k_frag[D == 64 ? D / 32 : 1], v_frag[D == 64 ? D / 32 : 1];
  if constexpr (D == 64) {
    __syncthreads();
    const int r = lane & 15;
#pragma unroll
    for (int ks = 0; ks < D / 32; ++ks) {
      const int kb = ks * 32 + (lane >> 4) * 8;
      k_frag[ks] = *(const bf16x8*)&k_lds[aoff<bf16, D>(wrow + r, kb)];
      v_frag[ks] = *(const bf16x8*)&v_lds[aoff<bf16, D>(wrow + r, kb)];
    }
  }

  f32x4 dk_acc[FO] = {}, dv_acc[FO] = {};

  // T14 split on the transposed q/dO stagings: tile t+1's register loads
  // are issued before tile t's compute and written to LDS at the top of
  // the next iteration (the natural q/dO images go through glds at the
  // top — single-buffered, their latency is what the split loads cover).
  TStage<BWD_BQ, D, BWD_THREADS> qtst, dotst;
  const int q_start = CAUSAL ? (kv0 / BWD_BQ) * BWD_BQ : 0;
  if (q_start < S) {
    qtst.load(q + (int64_t)q_start * i_rs, S - q_start, i_rs);
    dotst.load(dout + (int64_t)q_start * do_rs, S - q_start, do_rs);
  }
  for (int qt = q_start; qt < S; qt += BWD_BQ) {
    __syncthreads();
    stage_tile<BWD_BQ, D, NW>(q + (int64_t)qt * i_rs, q_lds, S - qt, zero16,
                              i_rs);
    stage_tile<BWD_BQ, D, NW>(dout + (int64_t)qt * do_rs, dot_lds, S - qt,
                              zero16, do_rs);
    qtst.write(qt_lds);
    dotst.write(dott_lds);
    for (int i = threadIdx.x; i < BWD_BQ; i += BWD_THREADS) {
      int g = qt + i;
      lse_lds[i] = g < S ? LSE[bh * S + g] : INFINITY;
      di_lds[i] = g < S ? Di[bh * S + g] : 0.0f;
    }
    __syncthreads();
    if (qt + BWD_BQ < S) {
      qtst.load(q + (int64_t)(qt + BWD_BQ) * i_rs, S - qt - BWD_BQ, i_rs);
      dotst.load(dout + (int64_t)(qt + BWD_BQ) * do_rs, S - qt - BWD_BQ,
                 do_rs);
    }

    // S^T[kv][q] = K Q^T ; P^T = exp(S^T*scale - lse[q]) — in place
    f32x4 st_acc[FN] = {};
    if constexpr (D == 64)
      mma_nt_areg<D, FN>(k_frag, q_lds, lane, st_acc);
    else
      mma_nt<D, FN>(k_lds, q_lds, wrow, lane, st_acc);
    const int cr = (lane >> 4) * 4;
    const int cc = lane & 15;
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int grow = kv0 + wrow + cr + j;      // kv index
        int gcol = qt + fn * 16 + cc;        // q index
        float sv = st_acc[fn][j] * scale;
        bool dead = gcol >= S || grow >= S || (CAUSAL && gcol < grow);
        float lse = lse_lds[fn * 16 + cc];
        st_acc[fn][j] = dead ? 0.0f : __expf(sv - lse);
      }
    // dV += P^T @ dO  (A = P^T from LDS, B = dO^T image)
    frag_to_lds<FN>(st_acc, p_lds[wid], lane);
    mma_pv<FO>(p_lds[wid], dott_lds, 0, lane, dv_acc);

    // dP^T[kv][q] = V dO^T
    f32x4 dpt_acc[FN] = {};
    if constexpr (D == 64)
      mma_nt_areg<D, FN>(v_frag, dot_lds, lane, dpt_acc);
    else
      mma_nt<D, FN>(v_lds, dot_lds, wrow, lane, dpt_acc);

    // dS^T = P^T * (dP^T - Di[q]) * scale — in place into dpt_acc
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        dpt_acc[fn][j] = st_acc[fn][j] *
                         (dpt_acc[fn][j] - di_lds[fn * 16 + cc]) * scale;

    // dK += dS^T @ Q  (A = dS^T via LDS, B = Q^T image)
    frag_to_lds<FN>(dpt_acc, p_lds[wid], lane);
    mma_pv<FO>(p_lds[wid], qt_lds, 0, lane, dk_acc);

    // dQ[q] += dS[q][kv] @ K[kv][d]: build the dS natural image from the
    // dS^T fragments. Element (kv = wrow+cr+j, q = fn*16+cc) goes to
    // image row q, col kv — j is kv-contiguous, so pack the 4 j-values
    // of each fragment into ONE ds_write_b64 (vs 4 b16 scatters).
    __syncthreads();  // protect ds_lds reuse across iterations
    {
      struct alignas(8) H4 { bf16 e[4]; };
#pragma unroll
      for (int fn = 0; fn < FN; ++fn) {
        H4 h;
#pragma unroll
        for (int j = 0; j < 4; ++j) h.e[j] = f2bf(dpt_acc[fn][j]);
        *(H4*)&ds_lds[aoff<bf16, BKV>(fn * 16 + cc, wrow + cr)] = h;
      }
    }
    __syncthreads();
    f32x4 dq_acc[FO] = {};
    mma_pv<FO>(ds_lds, kt_lds, wrow, lane, dq_acc);
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
      dkp[(int64_t)grow * w_rs + fo * 16 + cc] = f2bf(dk_acc[fo][j]);
      dvp[(int64_t)grow * w_rs + fo * 16 + cc] = f2bf(dv_acc[fo][j]);
    }
  }
}

}  // namespace attn
}  // namespace tnn

namespace tnn {

using namespace attn;

void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, const void* zero16, int B, int H, int S,
                     int D, const int64_t* i_str, const int64_t* o_str,
                     bool causal, float scale, hipStream_t s) {
  // wave-count selection: TNN_ATTN_WAVES overrides (4 or 8); default 8
  // (with the double-buffered staging + MFMA-carried softmax sum the
  // 8-wave/128-row-Q form wins every measured shape — see profiles/)
  static int nw = [] {
    const char* e = getenv("TNN_ATTN_WAVES");
    return (e && e[0] == '4') ? 4 : 8;
  }();
#define LAUNCH(DD, CC, NWV)                                                  \
  hipLaunchKernelGGL((k_attn_fwd<DD, CC, NWV>),                              \
                     dim3((S + NWV * 16 - 1) / (NWV * 16), B * H),           \
                     dim3(NWV * 64), 0, s, (const bf16*)q, (const bf16*)k,   \
                     (const bf16*)v, (bf16*)o, lse, (const bf16*)zero16, S,  \
                     H, i_str[0], i_str[1], i_str[2], o_str[0], o_str[1],    \
                     o_str[2], scale)
#define PICK(DD, CC)                                                         \
  do {                                                                       \
    if (nw == 8 && S >= 128) LAUNCH(DD, CC, 8); /* short seqs (ViT S=65) */  \
    else LAUNCH(DD, CC, 4);  /* would idle half an 8-wave block */           \
  } while (0)
  if (D == 64) { if (causal) PICK(64, true); else PICK(64, false); }
  else if (D == 128) { if (causal) PICK(128, true); else PICK(128, false); }
#undef PICK
#undef LAUNCH
}

// dense fp32 [B,H,S,D] workspace -> strided bf16 dq (w_* tuple)
__global__ void k_cast_dq(const float* __restrict__ ws, bf16* __restrict__ dq,
                          int64_t n, int S, int H, int D, int64_t w_rs,
                          int64_t w_hs, int64_t w_bs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int d = (int)(i % D);
    const int64_t r = i / D;
    const int sr = (int)(r % S);
    const int64_t bh = r / S;
    dq[(bh / H) * w_bs + (bh % H) * w_hs + (int64_t)sr * w_rs + d] =
        f2bf(ws[i]);
  }
}

void attn_bwd_launch(const void* q, const void* k, const void* v,
                     const void* o, const void* dout, const float* lse,
                     float* di, float* dq_ws, void* dq, void* dk, void* dv,
                     const void* zero16, int B, int H, int S, int D,
                     const int64_t* i_str, const int64_t* o_str,
                     const int64_t* do_str, const int64_t* w_str, bool causal,
                     float scale, hipStream_t s) {
  int64_t rows = (int64_t)B * H * S;
  hipLaunchKernelGGL(k_attn_dot, dim3((rows * 64 + 255) / 256), dim3(256), 0,
                     s, (const bf16*)dout, (const bf16*)o, di, rows, S, H, D,
                     do_str[0], do_str[1], do_str[2], o_str[0], o_str[1],
                     o_str[2]);
  dim3 grid((S + BKV - 1) / BKV, B * H);
#define LAUNCH(DD, CC)                                                     \
  hipLaunchKernelGGL((k_attn_bwd<DD, CC>), grid, dim3(BWD_THREADS), 0, s,  \
                     (const bf16*)q, (const bf16*)k, (const bf16*)v,       \
                     (const bf16*)dout, lse, di, dq_ws, (bf16*)dk,         \
                     (bf16*)dv, (const bf16*)zero16, S, H, i_str[0],       \
                     i_str[1], i_str[2], do_str[0], do_str[1], do_str[2],  \
                     w_str[0], w_str[1], w_str[2], scale)
  if (D == 64) { if (causal) LAUNCH(64, true); else LAUNCH(64, false); }
  else if (D == 128) { if (causal) LAUNCH(128, true); else LAUNCH(128, false); }
#undef LAUNCH
  const int64_t n = (int64_t)B * H * S * D;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)4096);
  hipLaunchKernelGGL(k_cast_dq, dim3(blocks), dim3(256), 0, s, dq_ws,
                     (bf16*)dq, n, S, H, D, w_str[0], w_str[1], w_str[2]);
}

}  // namespace tnn
