// GroupNorm over NHWC, fwd + bwd (reference
// src/nn/layers_impl/cuda/groupnorm_ops.cu:46,102,135,171 — fused group
// stats / apply / backward-reduce / backward-apply, fp32 statistics from
// any io dtype; the BASELINE north star names GroupNorm explicitly).
//
// x [N, HW, C] with G groups of CG = C/G channels; stats are over
// (HW, CG) per (n, g). gamma/beta are fp32 (same policy as BatchNorm).

#include "common.h"
#include "kernels.h"

namespace tnn {

// one 256-thread block per (n, g): sum/sumsq reduce -> mean, invstd
template <typename T>
__launch_bounds__(256)
__global__ void k_gn_stats(const T* __restrict__ x, float* __restrict__ mean,
                           float* __restrict__ invstd, int64_t HW, int C,
                           int CG, float eps) {
  const int g = blockIdx.x % (C / CG);
  const int64_t n = blockIdx.x / (C / CG);
  const T* base = x + n * HW * C + g * CG;
  __shared__ float scratch[8];

  float s = 0.0f, ss = 0.0f;
  const int64_t cnt = HW * CG;
  for (int64_t i = threadIdx.x; i < cnt; i += 256) {
    const int64_t p = i / CG;
    const int cc = (int)(i % CG);
    const float v = VecIO<T>::to_f32(base[p * C + cc]);
    s += v;
    ss += v * v;
  }
  s = block_reduce_sum(s, scratch);
  __shared__ float sbc;
  if (threadIdx.x == 0) sbc = s;
  __syncthreads();
  s = sbc;
  __syncthreads();
  ss = block_reduce_sum(ss, scratch);
  if (threadIdx.x == 0) {
    const float m = s / (float)cnt;
    const float var = fmaxf(ss / (float)cnt - m * m, 0.0f);
    mean[blockIdx.x] = m;
    invstd[blockIdx.x] = rsqrtf(var + eps);
  }
}

// elementwise y = (x - mean) * invstd * gamma[c] + beta[c]
template <typename T>
__launch_bounds__(256)
__global__ void k_gn_apply(const T* __restrict__ x,
                           const float* __restrict__ mean,
                           const float* __restrict__ invstd,
                           const float* __restrict__ gamma,
                           const float* __restrict__ beta, T* __restrict__ y,
                           int64_t total, int64_t HWC, int C, int CG) {
  const int G = C / CG;
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (; i < total; i += stride) {
    const int64_t n = i / HWC;
    const int c = (int)(i % C);
    const int64_t ng = n * G + c / CG;
    const float v = VecIO<T>::to_f32(x[i]);
    const float xh = (v - mean[ng]) * invstd[ng];
    y[i] = VecIO<T>::from_f32(xh * gamma[c] + beta[c]);
  }
}

// per-(n,g) backward sums: s1 = sum(dy*gamma), s2 = sum(dy*gamma*xhat)
template <typename T>
__launch_bounds__(256)
__global__ void k_gn_bwd_reduce(const T* __restrict__ x,
                                const T* __restrict__ dy,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                float* __restrict__ s1, float* __restrict__ s2,
                                int64_t HW, int C, int CG) {
  const int G = C / CG;
  const int g = blockIdx.x % G;
  const int64_t n = blockIdx.x / G;
  const int64_t off = n * HW * C + g * CG;
  const T* xb = x + off;
  const T* db = dy + off;
  const float m = mean[blockIdx.x], is = invstd[blockIdx.x];
  __shared__ float scratch[8];

  float a1 = 0.0f, a2 = 0.0f;
  const int64_t cnt = HW * CG;
  for (int64_t i = threadIdx.x; i < cnt; i += 256) {
    const int64_t p = i / CG;
    const int cc = (int)(i % CG);
    const float dxh = VecIO<T>::to_f32(db[p * C + cc]) * gamma[g * CG + cc];
    const float xh = (VecIO<T>::to_f32(xb[p * C + cc]) - m) * is;
    a1 += dxh;
    a2 += dxh * xh;
  }
  a1 = block_reduce_sum(a1, scratch);
  __shared__ float sbc;
  if (threadIdx.x == 0) sbc = a1;
  __syncthreads();
  a1 = sbc;
  __syncthreads();
  a2 = block_reduce_sum(a2, scratch);
  if (threadIdx.x == 0) {
    s1[blockIdx.x] = a1;
    s2[blockIdx.x] = a2;
  }
}

// dx = invstd * (dy*gamma - (s1 + xhat*s2)/cnt)
template <typename T>
__launch_bounds__(256)
__global__ void k_gn_bwd_apply(const T* __restrict__ x,
                               const T* __restrict__ dy,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               const float* __restrict__ gamma,
                               const float* __restrict__ s1,
                               const float* __restrict__ s2,
                               T* __restrict__ dx, int64_t total, int64_t HWC,
                               int C, int CG, float inv_cnt) {
  const int G = C / CG;
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (; i < total; i += stride) {
    const int64_t n = i / HWC;
    const int c = (int)(i % C);
    const int64_t ng = n * G + c / CG;
    const float is = invstd[ng];
    const float xh = (VecIO<T>::to_f32(x[i]) - mean[ng]) * is;
    const float dxh = VecIO<T>::to_f32(dy[i]) * gamma[c];
    dx[i] = VecIO<T>::from_f32(
        is * (dxh - (s1[ng] + xh * s2[ng]) * inv_cnt));
  }
}

// dgamma[c] = sum_{n,p} dy*xhat ; dbeta[c] = sum dy — column reduce over
// [rows, C] with a 2-D row-sliced grid + one atomic per (channel, slice)
template <typename T>
__launch_bounds__(256)
__global__ void k_gn_bwd_param(const T* __restrict__ x,
                               const T* __restrict__ dy,
                               const float* __restrict__ mean,
                               const float* __restrict__ invstd,
                               float* __restrict__ dgamma,
                               float* __restrict__ dbeta, int64_t rows,
                               int64_t HW, int C, int CG) {
  const int G = C / CG;
  const int c = blockIdx.x * 256 + threadIdx.x;
  if (c >= C) return;
  const int g = c / CG;
  const int64_t r0 = rows * blockIdx.y / gridDim.y;
  const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
  float ag = 0.0f, ab = 0.0f;
  for (int64_t r = r0; r < r1; ++r) {
    const int64_t n = r / HW;
    const int64_t ng = n * G + g;
    const float d = VecIO<T>::to_f32(dy[r * C + c]);
    const float xh = (VecIO<T>::to_f32(x[r * C + c]) - mean[ng]) * invstd[ng];
    ag += d * xh;
    ab += d;
  }
  atomicAdd(&dgamma[c], ag);
  atomicAdd(&dbeta[c], ab);
}

void gn_fwd_launch(DT dt, const void* x, const float* gamma, const float* beta,
                   void* y, float* mean, float* invstd, int64_t N, int64_t HW,
                   int C, int G, float eps, hipStream_t s) {
  const int CG = C / G;
  const int64_t total = N * HW * C;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, (int64_t)4096);
#define L(T)                                                                  \
  do {                                                                        \
    hipLaunchKernelGGL(k_gn_stats<T>, dim3(N* G), dim3(256), 0, s,            \
                       (const T*)x, mean, invstd, HW, C, CG, eps);            \
    hipLaunchKernelGGL(k_gn_apply<T>, dim3(blocks), dim3(256), 0, s,          \
                       (const T*)x, mean, invstd, gamma, beta, (T*)y, total,  \
                       HW * C, C, CG);                                        \
  } while (0)
  if (dt == DT::F32) L(float);
  else L(bf16);
#undef L
}

void gn_bwd_launch(DT dt, const void* x, const void* dy, const float* mean,
                   const float* invstd, const float* gamma, float* s1,
                   float* s2, void* dx, float* dgamma, float* dbeta, int64_t N,
                   int64_t HW, int C, int G, hipStream_t s) {
  const int CG = C / G;
  const int64_t total = N * HW * C;
  const int64_t rows = N * HW;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, (int64_t)4096);
  const float inv_cnt = 1.0f / (float)(HW * CG);
  const int yslices =
      (int)std::min<int64_t>(std::max<int64_t>(rows / 256, 1), (int64_t)64);
#define L(T)                                                                  \
  do {                                                                        \
    hipLaunchKernelGGL(k_gn_bwd_reduce<T>, dim3(N* G), dim3(256), 0, s,       \
                       (const T*)x, (const T*)dy, mean, invstd, gamma, s1,    \
                       s2, HW, C, CG);                                        \
    hipLaunchKernelGGL(k_gn_bwd_apply<T>, dim3(blocks), dim3(256), 0, s,      \
                       (const T*)x, (const T*)dy, mean, invstd, gamma, s1,    \
                       s2, (T*)dx, total, HW * C, C, CG, inv_cnt);            \
    hipLaunchKernelGGL(k_gn_bwd_param<T>, dim3(ceil_div(C, 256), yslices),    \
                       dim3(256), 0, s, (const T*)x, (const T*)dy, mean,      \
                       invstd, dgamma, dbeta, rows, HW, C, CG);               \
  } while (0)
  if (dt == DT::F32) L(float);
  else L(bf16);
#undef L
}

}  // namespace tnn
