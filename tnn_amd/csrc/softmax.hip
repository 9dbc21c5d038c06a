// Row softmax with fused scale + causal mask, forward and gradient.
// Replaces the reference's separate causal-mask fill + cudnnSoftmax
// passes (src/nn/blocks_impl/cuda/causal_mask.cu:13, softmax.cu:47-79)
// and the softmax activation kernels
// (src/nn/activations_impl/cuda/softmax_kernels.cu:11-346).
//
// y[r][c] = exp(x[r][c]*scale - m_r) / sum_c' exp(x[r][c']*scale - m_r)
// Causal: row r belongs to query position (r % mrows) + qoff within its
// [mrows, C] score matrix; columns > that position are masked to 0
// (suffix-causal decode prefill uses qoff = kv_len - mrows).
// Gradient: dx = scale * p * (dy - dot_row(p, dy)).
//
// One 256-thread block per row; rows <= 4096 cols stage x*scale (fwd) or
// p,dy products (bwd) in LDS so the row is read from global once.

#include "common.h"
#include "kernels.h"

namespace tnn {

namespace {
constexpr int SMAX_THREADS = 256;
constexpr int SMAX_LDS_CAP = 4096;  // fp32 row cache: 16 KiB
}  // namespace

template <typename T, bool CAUSAL, bool CACHED>
__launch_bounds__(SMAX_THREADS)
__global__ void k_smax_fwd(const T* __restrict__ x, T* __restrict__ y,
                           int cols, int mrows, int qoff, float scale) {
  const int64_t row = blockIdx.x;
  const T* xr = x + row * cols;
  T* yr = y + row * cols;
  __shared__ float scratch[8];
  extern __shared__ float rowbuf[];  // CACHED only

  int limit = cols;
  if constexpr (CAUSAL) {
    int qpos = (int)(row % mrows) + qoff;
    limit = min(cols, qpos + 1);
    if (limit <= 0) {  // fully masked row: all-zero output
      for (int c = threadIdx.x; c < cols; c += SMAX_THREADS)
        yr[c] = VecIO<T>::from_f32(0.0f);
      return;
    }
  }

  float mx = -INFINITY;
  for (int c = threadIdx.x; c < limit; c += SMAX_THREADS) {
    float v = VecIO<T>::to_f32(xr[c]) * scale;
    if constexpr (CACHED) rowbuf[c] = v;
    mx = fmaxf(mx, v);
  }
  mx = block_reduce_max(mx, scratch);
  __shared__ float smax;
  if (threadIdx.x == 0) smax = mx;
  __syncthreads();
  mx = smax;

  float se = 0.0f;
  for (int c = threadIdx.x; c < limit; c += SMAX_THREADS) {
    float v = CACHED ? rowbuf[c] : VecIO<T>::to_f32(xr[c]) * scale;
    float e = __expf(v - mx);
    if constexpr (CACHED) rowbuf[c] = e;
    se += e;
  }
  __syncthreads();
  se = block_reduce_sum(se, scratch);
  __shared__ float ssum;
  if (threadIdx.x == 0) ssum = se > 0.0f ? 1.0f / se : 0.0f;
  __syncthreads();
  const float inv = ssum;

  for (int c = threadIdx.x; c < limit; c += SMAX_THREADS) {
    float e = CACHED ? rowbuf[c] : __expf(VecIO<T>::to_f32(xr[c]) * scale - mx);
    yr[c] = VecIO<T>::from_f32(e * inv);
  }
  for (int c = limit + (int)threadIdx.x; c < cols; c += SMAX_THREADS)
    yr[c] = VecIO<T>::from_f32(0.0f);
}

template <typename T, bool CACHED>
__launch_bounds__(SMAX_THREADS)
__global__ void k_smax_bwd(const T* __restrict__ p, const T* __restrict__ dy,
                           T* __restrict__ dx, int cols, float scale) {
  const int64_t row = blockIdx.x;
  const T* pr = p + row * cols;
  const T* dyr = dy + row * cols;
  T* dxr = dx + row * cols;
  __shared__ float scratch[8];
  extern __shared__ float rowbuf[];  // CACHED: [p ; dy] interleaved pairs

  float dot = 0.0f;
  for (int c = threadIdx.x; c < cols; c += SMAX_THREADS) {
    float pv = VecIO<T>::to_f32(pr[c]);
    float dv = VecIO<T>::to_f32(dyr[c]);
    if constexpr (CACHED) {
      rowbuf[2 * c] = pv;
      rowbuf[2 * c + 1] = dv;
    }
    dot += pv * dv;
  }
  dot = block_reduce_sum(dot, scratch);
  __shared__ float sdot;
  if (threadIdx.x == 0) sdot = dot;
  __syncthreads();
  dot = sdot;

  for (int c = threadIdx.x; c < cols; c += SMAX_THREADS) {
    float pv, dv;
    if constexpr (CACHED) {
      pv = rowbuf[2 * c];
      dv = rowbuf[2 * c + 1];
    } else {
      pv = VecIO<T>::to_f32(pr[c]);
      dv = VecIO<T>::to_f32(dyr[c]);
    }
    dxr[c] = VecIO<T>::from_f32(scale * pv * (dv - dot));
  }
}

void smax_fwd_launch(DT dt, const void* x, void* y, int64_t rows, int cols,
                     int mrows, int qoff, float scale, bool causal,
                     hipStream_t s) {
  const bool cached = cols <= SMAX_LDS_CAP;
  const size_t shmem = cached ? (size_t)cols * sizeof(float) : 0;
  dim3 grid(rows);
#define L(T, C, CA)                                                       \
  hipLaunchKernelGGL((k_smax_fwd<T, C, CA>), grid, dim3(SMAX_THREADS),    \
                     shmem, s, (const T*)x, (T*)y, cols, mrows, qoff, scale)
  if (dt == DT::F32) {
    if (causal) { if (cached) L(float, true, true); else L(float, true, false); }
    else { if (cached) L(float, false, true); else L(float, false, false); }
  } else {
    if (causal) { if (cached) L(bf16, true, true); else L(bf16, true, false); }
    else { if (cached) L(bf16, false, true); else L(bf16, false, false); }
  }
#undef L
}

void smax_bwd_launch(DT dt, const void* p, const void* dy, void* dx,
                     int64_t rows, int cols, float scale, hipStream_t s) {
  const bool cached = cols <= SMAX_LDS_CAP / 2;
  const size_t shmem = cached ? (size_t)cols * 2 * sizeof(float) : 0;
  dim3 grid(rows);
#define L(T, CA)                                                          \
  hipLaunchKernelGGL((k_smax_bwd<T, CA>), grid, dim3(SMAX_THREADS), shmem, \
                     s, (const T*)p, (const T*)dy, (T*)dx, cols, scale)
  if (dt == DT::F32) { if (cached) L(float, true); else L(float, false); }
  else { if (cached) L(bf16, true); else L(bf16, false); }
#undef L
}

}  // namespace tnn
