// Embedding gather / scatter-add (reference embedding_ops.cu:17,48).

#include "common.h"
#include "kernels.h"

namespace tnn {

template <typename T>
__global__ void k_embedding_fwd(const int64_t* __restrict__ ids,
                                const T* __restrict__ table,
                                T* __restrict__ y, int64_t n_ids, int dim) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t n = n_ids * dim;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t tok = i / dim;
    int d = i % dim;
    y[i] = table[ids[tok] * dim + d];
  }
}

template <typename T>
__global__ void k_embedding_bwd(const int64_t* __restrict__ ids,
                                const T* __restrict__ dy,
                                float* __restrict__ dtable, int64_t n_ids,
                                int dim) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t n = n_ids * dim;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t tok = i / dim;
    int d = i % dim;
    atomicAdd(&dtable[ids[tok] * dim + d], VecIO<T>::to_f32(dy[i]));
  }
}

void embedding_fwd_launch(DT dt, const int64_t* ids, const void* table,
                          void* y, int64_t n_ids, int dim, hipStream_t s) {
  int64_t n = n_ids * dim;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_embedding_fwd<float>, dim3(blocks), dim3(256), 0, s,
                       ids, (const float*)table, (float*)y, n_ids, dim);
  else
    hipLaunchKernelGGL(k_embedding_fwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       ids, (const bf16*)table, (bf16*)y, n_ids, dim);
}

void embedding_bwd_launch(DT dt, const int64_t* ids, const void* dy,
                          float* dtable_f32, int64_t n_ids, int dim,
                          hipStream_t s) {
  int64_t n = n_ids * dim;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_embedding_bwd<float>, dim3(blocks), dim3(256), 0, s,
                       ids, (const float*)dy, dtable_f32, n_ids, dim);
  else
    hipLaunchKernelGGL(k_embedding_bwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       ids, (const bf16*)dy, dtable_f32, n_ids, dim);
}

}  // namespace tnn
