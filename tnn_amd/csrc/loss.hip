// Fused logsoftmax + cross-entropy (reference
// src/nn/loss_impl/cuda/loss_ops.cu:76 fused_logsoftmax_loss_kernel, :150
// gradient): one 256-thread block per row, wave+block max/sum reduce,
// fp32 math from any io dtype.

#include "common.h"
#include "kernels.h"

namespace tnn {

template <typename T>
__global__ void k_ce_fwd(const T* __restrict__ logits,
                         const int64_t* __restrict__ targets,
                         float* __restrict__ loss, float* __restrict__ lse,
                         int cols) {
  const int64_t row = blockIdx.x;
  const T* x = logits + row * cols;
  __shared__ float scratch[8];

  float mx = -INFINITY;
  for (int c = threadIdx.x; c < cols; c += blockDim.x)
    mx = fmaxf(mx, VecIO<T>::to_f32(x[c]));
  mx = block_reduce_max(mx, scratch);
  __shared__ float smax;
  if (threadIdx.x == 0) smax = mx;
  __syncthreads();
  mx = smax;

  float se = 0.0f;
  for (int c = threadIdx.x; c < cols; c += blockDim.x)
    se += __expf(VecIO<T>::to_f32(x[c]) - mx);
  __syncthreads();
  se = block_reduce_sum(se, scratch);
  if (threadIdx.x == 0) {
    float l = mx + __logf(se);
    lse[row] = l;
    // out-of-range target (corrupt batch / foreign ignore-index convention):
    // no OOB read; NaN loss fails loudly instead of reading garbage
    int64_t t = targets[row];
    loss[row] = (t >= 0 && t < cols) ? l - VecIO<T>::to_f32(x[t])
                                     : __int_as_float(0x7fc00000);
  }
}

template <typename T>
__global__ void k_ce_bwd(const T* __restrict__ logits,
                         const int64_t* __restrict__ targets,
                         const float* __restrict__ lse,
                         const float* __restrict__ dloss,
                         T* __restrict__ dlogits, int64_t rows, int cols) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t n = rows * cols;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t row = i / cols;
    int c = i % cols;
    float p = __expf(VecIO<T>::to_f32(logits[i]) - lse[row]);
    float g = (p - (c == (int)targets[row] ? 1.0f : 0.0f)) * dloss[row];
    dlogits[i] = VecIO<T>::from_f32(g);
  }
}

void ce_fwd_launch(DT dt, const void* logits, const int64_t* targets,
                   float* loss, float* lse, int64_t rows, int cols,
                   hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ce_fwd<float>, dim3(rows), dim3(256), 0, s,
                       (const float*)logits, targets, loss, lse, cols);
  else
    hipLaunchKernelGGL(k_ce_fwd<bf16>, dim3(rows), dim3(256), 0, s,
                       (const bf16*)logits, targets, loss, lse, cols);
}

void ce_bwd_launch(DT dt, const void* logits, const int64_t* targets,
                   const float* lse, const float* dloss, void* dlogits,
                   int64_t rows, int cols, hipStream_t s) {
  int64_t n = rows * cols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ce_bwd<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)logits, targets, lse, dloss,
                       (float*)dlogits, rows, cols);
  else
    hipLaunchKernelGGL(k_ce_bwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)logits, targets, lse, dloss,
                       (bf16*)dlogits, rows, cols);
}

}  // namespace tnn
