// MFMA block-tile GEMM core shared by the dense GEMM and the implicit-GEMM
// convolution kernels (replaces reference cublasGemmEx call sites,
// src/math/cuda/gemm.cu:64, and the cuDNN-frontend conv graphs).
//
// Structure (cdna_hip_programming.md §5 canonical GEMM):
//   - 256 threads = 4 waves in a 2x2 grid; block tile BM=128 x BN=64, BK=64.
//   - A staged in LDS as [BM][BK] row-major; B staged TRANSPOSED as
//     [BN][BK] so both operands read contiguous k-vectors per lane.
//   - LDS banking: element (row, col) lives at byte
//     (col*sizeof(T)) ^ ((((row>>3)^row)&7)<<4) within its 128B+ row.
//     The XOR key mixes low AND mid row bits so BOTH access patterns are
//     spread across banks: fragment reads touch 16 consecutive rows
//     (key varies via row&7) and transpose-scatter writes touch rows at
//     stride 8 across lanes (key varies via row>>3). Plain +pad schemes
//     can't fix the scatter side: any 16B-aligned row stride puts an
//     8-row lane stride on one bank (measured 3e8 conflict cycles).
//   - bf16: v_mfma_f32_16x16x32_bf16; fp32: v_mfma_f32_16x16x4_f32
//     (exact f32 at the f32 vector rate — no xf32 on gfx950).
//   - Each wave owns a 64x32 sub-tile: 4x2 fragments, f32x4 accumulators.
#pragma once

#include "common.h"

namespace tile {

constexpr int BM = 128, BN = 64, BK = 64, THREADS = 256;
constexpr int WAVES_M = 2, WAVES_N = 2;      // wave grid
constexpr int WM = BM / WAVES_M;             // 64 rows per wave
constexpr int WN = BN / WAVES_N;             // 32 cols per wave
constexpr int FM = WM / 16;                  // 4 row fragments
constexpr int FN = WN / 16;                  // 2 col fragments

// 16-byte staging pack (dwordx4 load/store); ext_vector_type cannot hold
// the __hip_bfloat16 struct, an aligned POD array can.
template <typename T> struct alignas(16) Pack16 {
  T e[16 / sizeof(T)];
};

template <typename T>
DEV bool aligned16(const T* p) {
  return ((unsigned long long)(const void*)p & 15ull) == 0;
}

// element offset of tile element (row, col) in the swizzled [rows][BK]
// image; the XOR only touches byte bits 4-6, so any 16B-aligned col keeps
// its vector contiguous and the swizzle stays inside the row.
template <typename T>
DEV int lds_off(int row, int col) {
  int byte = col * (int)sizeof(T);
  byte ^= ((((row >> 3) ^ row) & 7) << 4);
  return row * BK + byte / (int)sizeof(T);
}


template <typename T> struct LDSBytes {
  static constexpr int A = BM * BK * sizeof(T);
  static constexpr int B = BN * BK * sizeof(T);
  static constexpr int total = A + B;
};

// L2-locality tile remap (grouped ordering): dispatch-linear block ids
// sweep GM m-tiles x ALL n-tiles in panels, so a panel's A rows stay
// L2-resident while every B column streams once per panel — without
// this an [M,N,K] GEMM re-reads A (N/BN) times from HBM (measured: the
// fc1-shaped GEMM's whole runtime equals that traffic at 8 TB/s).
template <int GM>
DEV void tile_remap(int& tm, int& tn) {
  const int nbx = gridDim.x, nby = gridDim.y;
  const int id = blockIdx.y * nbx + blockIdx.x;  // dispatch order, x fastest
  const int per_group = GM * nby;
  const int group = id / per_group;
  const int first_m = group * GM;
  const int gm = min(nbx - first_m, GM);
  const int rem = id - group * per_group;
  tm = first_m + rem % gm;
  tn = rem / gm;
}

// XCD-aware variant: the dispatcher round-robins consecutive block ids
// over the 8 XCDs, so id%8 IS the XCD. Give each XCD a contiguous,
// grouped stripe of the tile space: panel reuse then hits the XCD's own
// (non-coherent) 4 MiB L2 instead of only the die-level L3.
template <int GM>
DEV void tile_remap_xcd(int& tm, int& tn) {
  const int nbx = gridDim.x, nby = gridDim.y;
  const int T = nbx * nby;
  const int id = blockIdx.y * nbx + blockIdx.x;
  // stripe permutation is a bijection only when 8 | T (every CNN-zoo
  // grid: pow2 tiles); otherwise keep dispatch order — a clamped
  // fallback here DUPLICATED one tile and dropped another per ragged
  // stripe, silently corrupting C
  const int lid = (T & 7) == 0 ? ((id & 7) * (T >> 3) + (id >> 3)) : id;
  const int per_group = GM * nby;
  const int group = lid / per_group;
  const int first_m = group * GM;
  const int gm = min(nbx - first_m, GM);
  const int rem = lid - group * per_group;
  tm = first_m + rem % gm;
  tn = rem / gm;
}

struct WaveCoord {
  int wid, lane, wrow0, wcol0;
  DEV WaveCoord() {
    wid = threadIdx.x >> 6;
    lane = threadIdx.x & 63;
    wrow0 = (wid / WAVES_N) * WM;
    wcol0 = (wid % WAVES_N) * WN;
  }
};

// ---- async global->LDS staging for the A tile (gfx950 global_load_lds) ---
// Writes the whole [BM][BK] swizzled image with 16B-per-lane DMA: each
// wave issues (BM*BK*sizeof(T))/4096 instructions, each filling a 1 KiB
// lane-linear LDS region. The swizzle therefore moves to the SOURCE
// address (guide rule 21): each lane asks AddrFn for the element that
// belongs at its linear LDS slot. Invalid lanes (image padding, M/K
// tails) point their source at a 16-byte zero page — no exec masking, no
// pre-zeroing pass (a masked glds lane would leave stale LDS bytes).
// AddrFn: (row_in_tile, col_elem) -> const T* (16B-aligned) or zero16.
template <typename T, int ROWS, typename AddrFn>
DEV void glds_stage(T* As, const WaveCoord& w, AddrFn&& addr) {
  constexpr int ROWB = BK * (int)sizeof(T);      // bytes per image row
  constexpr int RPK = 1024 / ROWB;               // rows per 1 KiB region
  constexpr int LPR = ROWB / 16;                 // lanes per row
  constexpr int NREG = ROWS * ROWB / 1024;       // 1 KiB regions in the tile
  constexpr int NPW = NREG / 4;                  // regions per wave
#pragma unroll
  for (int i = 0; i < NPW; ++i) {
    const int j = w.wid * NPW + i;
    const int rl = RPK * j + w.lane / LPR;
    const int byte_in_row = (w.lane % LPR) * 16;
    const int kk = (byte_in_row ^ ((((rl >> 3) ^ rl) & 7) << 4)) /
                   (int)sizeof(T);
    const T* src = addr(rl, kk);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src,
        (__attribute__((address_space(3))) unsigned int*)
            &As[j * (1024 / (int)sizeof(T))],
        16, 0, 0);
  }
}

template <typename T, typename AddrFn>
DEV void glds_stage_a(T* As, const WaveCoord& w, AddrFn&& addr) {
  glds_stage<T, BM>(As, w, addr);
}

// glds instructions per wave for one ROWS-row tile (vmcnt bookkeeping)
template <typename T, int ROWS>
constexpr int glds_count() {
  return ROWS * (BK * (int)sizeof(T)) / 1024 / 4;
}

// counted s_waitcnt vmcnt(N): __syncthreads() with a glds in flight drains
// vmcnt(0) (guide: LDS-DMA is a pending LDS write on the VM counter), so
// double-buffered loops wait an explicit partial count + raw s_barrier.
template <int N>
DEV void wait_vmcnt() {
  static_assert(N == 0 || N == 4 || N == 6 || N == 8 || N == 12 ||
                N == 16 || N == 24,
                "add an asm literal for this count");
  if constexpr (N == 0) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  else if constexpr (N == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else if constexpr (N == 6) asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  else if constexpr (N == 8) asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  else if constexpr (N == 12) asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
  else if constexpr (N == 16) asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
  else if constexpr (N == 24) asm volatile("s_waitcnt vmcnt(24)" ::: "memory");
}


// ---- MFMA tile compute: acc[FM][FN] += A_tile * B_tile^T-stored -----------
DEV void mfma_compute_tile(const bf16* As, const bf16* Bs, const WaveCoord& w,
                           f32x4 acc[FM][FN]) {
  const int r = w.lane & 15;            // fragment row/col within 16
#pragma unroll
  for (int ks = 0; ks < BK / 32; ++ks) {
    const int kb = ks * 32 + (w.lane >> 4) * 8;  // 8 bf16 per lane
    bf16x8 a[FM], b[FN];
#pragma unroll
    for (int fm = 0; fm < FM; ++fm)
      a[fm] = *(const bf16x8*)&As[lds_off<bf16>(w.wrow0 + fm * 16 + r, kb)];
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
      b[fn] = *(const bf16x8*)&Bs[lds_off<bf16>(w.wcol0 + fn * 16 + r, kb)];
#pragma unroll
    for (int fm = 0; fm < FM; ++fm)
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a[fm], b[fn], acc[fm][fn], 0, 0, 0);
  }
}

DEV void mfma_compute_tile(const float* As, const float* Bs, const WaveCoord& w,
                           f32x4 acc[FM][FN]) {
  const int r = w.lane & 15;
  const int kq = w.lane >> 4;         // one f32 per lane per k-step of 4
#pragma unroll
  for (int ks = 0; ks < BK / 4; ++ks) {
    float a[FM], b[FN];
#pragma unroll
    for (int fm = 0; fm < FM; ++fm)
      a[fm] = As[lds_off<float>(w.wrow0 + fm * 16 + r, ks * 4 + kq)];
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
      b[fn] = Bs[lds_off<float>(w.wcol0 + fn * 16 + r, ks * 4 + kq)];
#pragma unroll
    for (int fm = 0; fm < FM; ++fm)
#pragma unroll
      for (int fn = 0; fn < FN; ++fn)
        acc[fm][fn] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            a[fm], b[fn], acc[fm][fn], 0, 0, 0);
  }
}

// ---- epilogue: C/D fragment map col=lane&15, row=(lane>>4)*4+j ------------
// Visits every accumulator element with its global (row, col).
template <typename F>
DEV void epilogue_visit(const WaveCoord& w, f32x4 acc[FM][FN],
                        int row0, int col0, F&& emit) {
  const int cr = (w.lane >> 4) * 4;
  const int cc = w.lane & 15;
#pragma unroll
  for (int fm = 0; fm < FM; ++fm)
#pragma unroll
    for (int fn = 0; fn < FN; ++fn)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = row0 + w.wrow0 + fm * 16 + cr + j;
        int col = col0 + w.wcol0 + fn * 16 + cc;
        emit(row, col, acc[fm][fn][j]);
      }
}

}  // namespace tile
