// NHWC pooling kernels (reference maxpool_ops.cu / avgpool_ops.cu:
// one-thread-per-output forward, index-replay / direct-gather backward).

#include "common.h"
#include "kernels.h"

namespace tnn {

template <typename T>
__global__ void k_maxpool_fwd(const T* __restrict__ x, T* __restrict__ y,
                              int32_t* __restrict__ idx, PoolShape ps) {
  int64_t n_out = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_out; i += stride) {
    int c = i % ps.C;
    int64_t rest = i / ps.C;
    int ow = rest % ps.OW;
    rest /= ps.OW;
    int oh = rest % ps.OH;
    int n = rest / ps.OH;
    float best = -INFINITY;
    int best_idx = -1;
    for (int kh = 0; kh < ps.KH; ++kh) {
      int ih = oh * ps.SH - ps.PH + kh;
      if (ih < 0 || ih >= ps.H) continue;
      for (int kw = 0; kw < ps.KW; ++kw) {
        int iw = ow * ps.SW - ps.PW + kw;
        if (iw < 0 || iw >= ps.W) continue;
        float v = VecIO<T>::to_f32(
            x[(((int64_t)n * ps.H + ih) * ps.W + iw) * ps.C + c]);
        if (v > best) {
          best = v;
          best_idx = ih * ps.W + iw;  // pixel within the image
        }
      }
    }
    y[i] = VecIO<T>::from_f32(best_idx < 0 ? 0.0f : best);
    idx[i] = best_idx;
  }
}

template <typename T>
__global__ void k_maxpool_bwd(const T* __restrict__ dy,
                              const int32_t* __restrict__ idx,
                              float* __restrict__ dx, PoolShape ps) {
  int64_t n_out = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n_out; i += stride) {
    int p = idx[i];
    if (p < 0) continue;
    int c = i % ps.C;
    int64_t n = i / ps.C / (ps.OW * (int64_t)ps.OH);
    atomicAdd(&dx[((int64_t)n * ps.H * ps.W + p) * ps.C + c],
              VecIO<T>::to_f32(dy[i]));
  }
}

template <typename T>
__global__ void k_avgpool_fwd(const T* __restrict__ x, T* __restrict__ y,
                              PoolShape ps) {
  int64_t n_out = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv = 1.0f / (ps.KH * ps.KW);
  for (; i < n_out; i += stride) {
    int c = i % ps.C;
    int64_t rest = i / ps.C;
    int ow = rest % ps.OW;
    rest /= ps.OW;
    int oh = rest % ps.OH;
    int n = rest / ps.OH;
    float acc = 0.0f;
    for (int kh = 0; kh < ps.KH; ++kh) {
      int ih = oh * ps.SH - ps.PH + kh;
      if (ih < 0 || ih >= ps.H) continue;
      for (int kw = 0; kw < ps.KW; ++kw) {
        int iw = ow * ps.SW - ps.PW + kw;
        if (iw < 0 || iw >= ps.W) continue;
        acc += VecIO<T>::to_f32(
            x[(((int64_t)n * ps.H + ih) * ps.W + iw) * ps.C + c]);
      }
    }
    y[i] = VecIO<T>::from_f32(acc * inv);  // count_include_pad semantics
  }
}

template <typename T>
__global__ void k_avgpool_bwd(const T* __restrict__ dy, T* __restrict__ dx,
                              PoolShape ps) {
  // direct gather: each input pixel sums dy over the windows containing it
  int64_t n_in = (int64_t)ps.N * ps.H * ps.W * ps.C;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv = 1.0f / (ps.KH * ps.KW);
  for (; i < n_in; i += stride) {
    int c = i % ps.C;
    int64_t rest = i / ps.C;
    int iw = rest % ps.W;
    rest /= ps.W;
    int ih = rest % ps.H;
    int n = rest / ps.H;
    float acc = 0.0f;
    // smallest oh with window covering ih: ceil((ih+PH-KH+1)/SH), clamped;
    // C division truncates toward zero so guard the negative numerator
    int a1 = ih + ps.PH - ps.KH + 1;
    int oh_lo = a1 > 0 ? (a1 + ps.SH - 1) / ps.SH : 0;
    int oh_hi = (ih + ps.PH) / ps.SH;
    if (oh_hi >= ps.OH) oh_hi = ps.OH - 1;
    int b1 = iw + ps.PW - ps.KW + 1;
    int ow_lo = b1 > 0 ? (b1 + ps.SW - 1) / ps.SW : 0;
    int ow_hi = (iw + ps.PW) / ps.SW;
    if (ow_hi >= ps.OW) ow_hi = ps.OW - 1;
    for (int oh = oh_lo; oh <= oh_hi; ++oh) {
      int kh = ih + ps.PH - oh * ps.SH;
      if (kh < 0 || kh >= ps.KH) continue;
      for (int ow = ow_lo; ow <= ow_hi; ++ow) {
        int kw = iw + ps.PW - ow * ps.SW;
        if (kw < 0 || kw >= ps.KW) continue;
        acc += VecIO<T>::to_f32(
            dy[(((int64_t)n * ps.OH + oh) * ps.OW + ow) * ps.C + c]);
      }
    }
    dx[i] = VecIO<T>::from_f32(acc * inv);
  }
}

static inline int pool_blocks(int64_t n) {
  int64_t b = (n + 255) / 256;
  return (int)(b < 2048 ? b : 2048);
}

void maxpool_fwd_launch(DT dt, const void* x, void* y, int32_t* idx,
                        const PoolShape& ps, hipStream_t s) {
  int64_t n = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_maxpool_fwd<float>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const float*)x, (float*)y, idx, ps);
  else
    hipLaunchKernelGGL(k_maxpool_fwd<bf16>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const bf16*)x, (bf16*)y, idx, ps);
}

void maxpool_bwd_launch(DT dt, const void* dy, const int32_t* idx,
                        float* dx_f32, const PoolShape& ps, hipStream_t s) {
  int64_t n = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_maxpool_bwd<float>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const float*)dy, idx, dx_f32, ps);
  else
    hipLaunchKernelGGL(k_maxpool_bwd<bf16>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const bf16*)dy, idx, dx_f32, ps);
}

void avgpool_fwd_launch(DT dt, const void* x, void* y, const PoolShape& ps,
                        hipStream_t s) {
  int64_t n = (int64_t)ps.N * ps.OH * ps.OW * ps.C;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_avgpool_fwd<float>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const float*)x, (float*)y, ps);
  else
    hipLaunchKernelGGL(k_avgpool_fwd<bf16>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const bf16*)x, (bf16*)y, ps);
}

void avgpool_bwd_launch(DT dt, const void* dy, void* dx, const PoolShape& ps,
                        hipStream_t s) {
  int64_t n = (int64_t)ps.N * ps.H * ps.W * ps.C;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_avgpool_bwd<float>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const float*)dy, (float*)dx, ps);
  else
    hipLaunchKernelGGL(k_avgpool_bwd<bf16>, dim3(pool_blocks(n)), dim3(256), 0,
                       s, (const bf16*)dy, (bf16*)dx, ps);
}

}  // namespace tnn
