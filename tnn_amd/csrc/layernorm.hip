// LayerNorm row kernels (reference src/nn/layers_impl/cuda/layer_norm_ops.cu
// :21 fwd, :51 bwd + the cuDNN-fe path). One block per row; fp32 stats;
// dgamma/dbeta via fp32 atomics (grid is the row count).

#include "common.h"
#include "kernels.h"

namespace tnn {

template <typename T>
__global__ void k_ln_fwd(const T* __restrict__ x, const float* gamma,
                         const float* beta, T* __restrict__ y,
                         float* __restrict__ mean, float* __restrict__ invstd,
                         int cols, float eps) {
  const int64_t row = blockIdx.x;
  const T* xr = x + row * cols;
  T* yr = y + row * cols;
  __shared__ float scratch[8];
  __shared__ float s_m, s_is;

  float s = 0.0f;
  for (int c = threadIdx.x; c < cols; c += blockDim.x)
    s += VecIO<T>::to_f32(xr[c]);
  s = block_reduce_sum(s, scratch);
  if (threadIdx.x == 0) s_m = s / cols;
  __syncthreads();
  const float m = s_m;

  float v = 0.0f;
  for (int c = threadIdx.x; c < cols; c += blockDim.x) {
    float d = VecIO<T>::to_f32(xr[c]) - m;
    v += d * d;
  }
  __syncthreads();
  v = block_reduce_sum(v, scratch);
  if (threadIdx.x == 0) {
    s_is = rsqrtf(v / cols + eps);
    mean[row] = m;
    invstd[row] = s_is;
  }
  __syncthreads();
  const float is = s_is;
  for (int c = threadIdx.x; c < cols; c += blockDim.x) {
    float xhat = (VecIO<T>::to_f32(xr[c]) - m) * is;
    yr[c] = VecIO<T>::from_f32(xhat * gamma[c] + beta[c]);
  }
}

template <typename T>
__global__ void k_ln_bwd(const T* __restrict__ x, const T* __restrict__ dy,
                         const float* gamma, const float* mean,
                         const float* invstd, T* __restrict__ dx,
                         float* __restrict__ dgamma, float* __restrict__ dbeta,
                         int cols) {
  const int64_t row = blockIdx.x;
  const T* xr = x + row * cols;
  const T* dyr = dy + row * cols;
  T* dxr = dx + row * cols;
  const float m = mean[row], is = invstd[row];
  __shared__ float scratch[8];
  __shared__ float s_a, s_b;

  float sa = 0.0f, sb = 0.0f;  // sum(dy*gamma), sum(dy*gamma*xhat)
  for (int c = threadIdx.x; c < cols; c += blockDim.x) {
    float g = VecIO<T>::to_f32(dyr[c]) * gamma[c];
    float xhat = (VecIO<T>::to_f32(xr[c]) - m) * is;
    sa += g;
    sb += g * xhat;
  }
  sa = block_reduce_sum(sa, scratch);
  __syncthreads();
  sb = block_reduce_sum(sb, scratch);
  if (threadIdx.x == 0) {
    s_a = sa / cols;
    s_b = sb / cols;
  }
  __syncthreads();
  const float ma = s_a, mb = s_b;
  for (int c = threadIdx.x; c < cols; c += blockDim.x) {
    float gy = VecIO<T>::to_f32(dyr[c]);
    float xhat = (VecIO<T>::to_f32(xr[c]) - m) * is;
    dxr[c] = VecIO<T>::from_f32(is * (gy * gamma[c] - ma - xhat * mb));
    atomicAdd(&dgamma[c], gy * xhat);
    atomicAdd(&dbeta[c], gy);
  }
}

void ln_fwd_launch(DT dt, const void* x, const float* gamma, const float* beta,
                   void* y, float* mean, float* invstd, int64_t rows, int cols,
                   float eps, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ln_fwd<float>, dim3(rows), dim3(256), 0, s,
                       (const float*)x, gamma, beta, (float*)y, mean, invstd,
                       cols, eps);
  else
    hipLaunchKernelGGL(k_ln_fwd<bf16>, dim3(rows), dim3(256), 0, s,
                       (const bf16*)x, gamma, beta, (bf16*)y, mean, invstd,
                       cols, eps);
}

void ln_bwd_launch(DT dt, const void* x, const void* dy, const float* gamma,
                   const float* mean, const float* invstd, void* dx,
                   float* dgamma, float* dbeta, int64_t rows, int cols,
                   hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ln_bwd<float>, dim3(rows), dim3(256), 0, s,
                       (const float*)x, (const float*)dy, gamma, mean, invstd,
                       (float*)dx, dgamma, dbeta, cols);
  else
    hipLaunchKernelGGL(k_ln_bwd<bf16>, dim3(rows), dim3(256), 0, s,
                       (const bf16*)x, (const bf16*)dy, gamma, mean, invstd,
                       (bf16*)dx, dgamma, dbeta, cols);
}

}  // namespace tnn
