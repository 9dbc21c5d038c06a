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
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  const int64_t row = blockIdx.x;
  const T* x = logits + row * cols;
  __shared__ float scratch[8];
  // odd vocab sizes (50257) leave odd rows off 16-B alignment: a few
  // leading scalars re-align, then the 16-B vector body, then the tail
  const int lead =
      min(cols, (int)(((16 - ((uintptr_t)x & 15)) / sizeof(T)) & (V - 1)));
  const T* xa = x + lead;
  const int cv = (cols - lead) / V;

  float mx = -INFINITY;
  for (int c = threadIdx.x; c < lead; c += blockDim.x)
    mx = fmaxf(mx, VecIO<T>::to_f32(x[c]));
  for (int c = threadIdx.x; c < cv; c += blockDim.x) {
    P v = ((const P*)xa)[c];
#pragma unroll
    for (int j = 0; j < V; ++j) mx = fmaxf(mx, VecIO<T>::to_f32(v.e[j]));
  }
  for (int c = lead + cv * V + (int)threadIdx.x; c < cols; c += blockDim.x)
    mx = fmaxf(mx, VecIO<T>::to_f32(x[c]));
  mx = block_reduce_max(mx, scratch);
  __shared__ float smax;
  if (threadIdx.x == 0) smax = mx;
  __syncthreads();
  mx = smax;

  float se = 0.0f;
  for (int c = threadIdx.x; c < lead; c += blockDim.x)
    se += __expf(VecIO<T>::to_f32(x[c]) - mx);
  for (int c = threadIdx.x; c < cv; c += blockDim.x) {
    P v = ((const P*)xa)[c];
#pragma unroll
    for (int j = 0; j < V; ++j)
      se += __expf(VecIO<T>::to_f32(v.e[j]) - mx);
  }
  for (int c = lead + cv * V + (int)threadIdx.x; c < cols; c += blockDim.x)
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
  // one block per row: row-constant lse/dloss/target load once, 16-B
  // vector body (the flat grid-stride form did a 64-bit div/mod + two
  // scalar row loads PER ELEMENT and ran 2.5x off roofline)
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* x = logits + row * cols;
    T* dx = dlogits + row * cols;
    const float l = lse[row];
    const float dl = dloss[row];
    const int tgt = (int)targets[row];
    const int lead =
        min(cols, (int)(((16 - ((uintptr_t)x & 15)) / sizeof(T)) & (V - 1)));
    const int cv = (cols - lead) / V;
    for (int c = threadIdx.x; c < lead; c += blockDim.x) {
      float p = __expf(VecIO<T>::to_f32(x[c]) - l);
      dx[c] = VecIO<T>::from_f32((p - (c == tgt ? 1.0f : 0.0f)) * dl);
    }
    const P* xv = (const P*)(x + lead);
    P* dv = (P*)(dx + lead);
    for (int c = threadIdx.x; c < cv; c += blockDim.x) {
      P v = xv[c], o;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        int col = lead + c * V + j;
        float p = __expf(VecIO<T>::to_f32(v.e[j]) - l);
        o.e[j] = VecIO<T>::from_f32((p - (col == tgt ? 1.0f : 0.0f)) * dl);
      }
      dv[c] = o;
    }
    for (int c = lead + cv * V + (int)threadIdx.x; c < cols; c += blockDim.x) {
      float p = __expf(VecIO<T>::to_f32(x[c]) - l);
      dx[c] = VecIO<T>::from_f32((p - (c == tgt ? 1.0f : 0.0f)) * dl);
    }
  }
}

// ---------------------------------------------------------------------------
// Pointwise regression losses: MSE / MAE / Huber, mean reduction
// (reference src/nn/loss_impl/cuda/loss_ops.cu:308-390). Forward
// accumulates the scaled partial sums into one fp32 scalar; backward is
// elementwise d(mean loss)/d(pred) * upstream.
// ---------------------------------------------------------------------------

enum PtLoss : int { PT_MSE = 0, PT_MAE = 1, PT_HUBER = 2 };

template <typename T>
__launch_bounds__(256)
__global__ void k_ptloss_fwd(const T* __restrict__ pred,
                             const T* __restrict__ tgt,
                             float* __restrict__ out, int64_t n, int kind,
                             float delta, float inv_n) {
  __shared__ float scratch[8];
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * 256;
  float acc = 0.0f;
  for (; i < n; i += stride) {
    const float d = VecIO<T>::to_f32(pred[i]) - VecIO<T>::to_f32(tgt[i]);
    if (kind == PT_MSE) acc += d * d;
    else if (kind == PT_MAE) acc += fabsf(d);
    else {
      const float a = fabsf(d);
      acc += a <= delta ? 0.5f * d * d : delta * (a - 0.5f * delta);
    }
  }
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc * inv_n);
}

template <typename T>
__launch_bounds__(256)
__global__ void k_ptloss_bwd(const T* __restrict__ pred,
                             const T* __restrict__ tgt,
                             const float* __restrict__ dloss,
                             T* __restrict__ dpred, int64_t n, int kind,
                             float delta, float inv_n) {
  const float g = dloss[0] * inv_n;
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * 256;
  for (; i < n; i += stride) {
    const float d = VecIO<T>::to_f32(pred[i]) - VecIO<T>::to_f32(tgt[i]);
    float gd;
    if (kind == PT_MSE) gd = 2.0f * d;
    else if (kind == PT_MAE) gd = d > 0.0f ? 1.0f : (d < 0.0f ? -1.0f : 0.0f);
    else gd = fmaxf(-delta, fminf(d, delta));
    dpred[i] = VecIO<T>::from_f32(gd * g);
  }
}

void ptloss_fwd_launch(DT dt, const void* pred, const void* tgt, float* out,
                       int64_t n, int kind, float delta, hipStream_t s) {
  const int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  const float inv_n = 1.0f / (float)n;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ptloss_fwd<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)pred, (const float*)tgt, out, n, kind,
                       delta, inv_n);
  else
    hipLaunchKernelGGL(k_ptloss_fwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)pred, (const bf16*)tgt, out, n, kind,
                       delta, inv_n);
}

void ptloss_bwd_launch(DT dt, const void* pred, const void* tgt,
                       const float* dloss, void* dpred, int64_t n, int kind,
                       float delta, hipStream_t s) {
  const int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  const float inv_n = 1.0f / (float)n;
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_ptloss_bwd<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)pred, (const float*)tgt, dloss,
                       (float*)dpred, n, kind, delta, inv_n);
  else
    hipLaunchKernelGGL(k_ptloss_bwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)pred, (const bf16*)tgt, dloss,
                       (bf16*)dpred, n, kind, delta, inv_n);
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
  int blocks = (int)std::min<int64_t>(rows, (int64_t)4096);
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
