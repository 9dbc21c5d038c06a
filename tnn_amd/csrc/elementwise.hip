// Elementwise kernels: activations, ReLU backward mask, dropout (Philox),
// column sums (replaces reference src/ops/cuda/kernels.cu element-wise set,
// src/nn/activations_impl/cuda/*, dropout.cu, add_bias/bgrad column reduce).
//
// All grid-stride, 256-thread workgroups (wave64 x 4), vectorized 4-wide.

#include "common.h"
#include "kernels.h"

namespace tnn {

template <typename T, typename F>
__global__ void k_ewise1(const T* __restrict__ x, T* __restrict__ y,
                         int64_t n, F f) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) y[i] = VecIO<T>::from_f32(f(VecIO<T>::to_f32(x[i])));
}

static inline int ew_blocks(int64_t n) {
  int64_t b = (n + 255) / 256;
  return (int)(b < 2048 ? b : 2048);
}

// ---- activations -----------------------------------------------------------
template <typename T>
__global__ void k_act_fwd(const T* x, T* y, int64_t n, int kind) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    y[i] = VecIO<T>::from_f32(act_apply(VecIO<T>::to_f32(x[i]), kind));
}

template <typename T>
__global__ void k_act_bwd(const T* dy, const T* x, const T* y, T* dx,
                          int64_t n, int kind) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    dx[i] = VecIO<T>::from_f32(act_grad(VecIO<T>::to_f32(dy[i]),
                                        VecIO<T>::to_f32(x[i]),
                                        VecIO<T>::to_f32(y[i]), kind));
}

// ---- fused residual add + activation (ResidualBlock / MSequential add
// join: one kernel instead of the at::native add + relu pair; backward
// is a single mask pass shared by both inputs) -------------------------------
template <typename T>
__global__ void k_add_act_fwd(const T* __restrict__ a, const T* __restrict__ b,
                              T* __restrict__ y, int64_t n, int kind) {
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  constexpr int V = 16 / sizeof(T);
  const int64_t n_v = n / V;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const P* av = (const P*)a;
  const P* bv = (const P*)b;
  P* yv = (P*)y;
  for (; i < n_v; i += stride) {
    P x = av[i], z = bv[i], o;
#pragma unroll
    for (int j = 0; j < V; ++j)
      o.e[j] = VecIO<T>::from_f32(act_apply(
          VecIO<T>::to_f32(x.e[j]) + VecIO<T>::to_f32(z.e[j]), kind));
    yv[i] = o;
  }
  for (int64_t k = n_v * V + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       k < n; k += stride)
    y[k] = VecIO<T>::from_f32(act_apply(
        VecIO<T>::to_f32(a[k]) + VecIO<T>::to_f32(b[k]), kind));
}

template <typename T>
__global__ void k_add_act_bwd(const T* __restrict__ dy, const T* __restrict__ y,
                              T* __restrict__ g, int64_t n, int kind) {
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  constexpr int V = 16 / sizeof(T);
  const int64_t n_v = n / V;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const P* dv = (const P*)dy;
  const P* yv = (const P*)y;
  P* gv = (P*)g;
  for (; i < n_v; i += stride) {
    P d = dv[i], yy = yv[i], o;
#pragma unroll
    for (int j = 0; j < V; ++j)
      o.e[j] = VecIO<T>::from_f32(
          act_grad(VecIO<T>::to_f32(d.e[j]), 0.0f,
                   VecIO<T>::to_f32(yy.e[j]), kind));
    gv[i] = o;
  }
  for (int64_t k = n_v * V + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       k < n; k += stride)
    g[k] = VecIO<T>::from_f32(act_grad(VecIO<T>::to_f32(dy[k]), 0.0f,
                                       VecIO<T>::to_f32(y[k]), kind));
}

void add_act_fwd_launch(DT dt, const void* a, const void* b, void* y,
                        int64_t n, int kind, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_add_act_fwd<float>, dim3(ew_blocks(n / 4 + 1)),
                       dim3(256), 0, s, (const float*)a, (const float*)b,
                       (float*)y, n, kind);
  else
    hipLaunchKernelGGL(k_add_act_fwd<bf16>, dim3(ew_blocks(n / 8 + 1)),
                       dim3(256), 0, s, (const bf16*)a, (const bf16*)b,
                       (bf16*)y, n, kind);
}

void add_act_bwd_launch(DT dt, const void* dy, const void* y, void* g,
                        int64_t n, int kind, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_add_act_bwd<float>, dim3(ew_blocks(n / 4 + 1)),
                       dim3(256), 0, s, (const float*)dy, (const float*)y,
                       (float*)g, n, kind);
  else
    hipLaunchKernelGGL(k_add_act_bwd<bf16>, dim3(ew_blocks(n / 8 + 1)),
                       dim3(256), 0, s, (const bf16*)dy, (const bf16*)y,
                       (bf16*)g, n, kind);
}

void act_fwd_launch(DT dt, const void* x, void* y, int64_t n, int kind,
                    hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_act_fwd<float>, dim3(ew_blocks(n)), dim3(256), 0, s,
                       (const float*)x, (float*)y, n, kind);
  else
    hipLaunchKernelGGL(k_act_fwd<bf16>, dim3(ew_blocks(n)), dim3(256), 0, s,
                       (const bf16*)x, (bf16*)y, n, kind);
}

void act_bwd_launch(DT dt, const void* dy, const void* x, const void* y,
                    void* dx, int64_t n, int kind, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_act_bwd<float>, dim3(ew_blocks(n)), dim3(256), 0, s,
                       (const float*)dy, (const float*)x, (const float*)y,
                       (float*)dx, n, kind);
  else
    hipLaunchKernelGGL(k_act_bwd<bf16>, dim3(ew_blocks(n)), dim3(256), 0, s,
                       (const bf16*)dy, (const bf16*)x, (const bf16*)y,
                       (bf16*)dx, n, kind);
}

// ---- relu backward through the saved output (mask = y > 0) -----------------
template <typename T>
__global__ void k_relu_bwd_mask(const T* dy, const T* y, T* dx, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    dx[i] = VecIO<T>::to_f32(y[i]) > 0.0f ? dy[i] : VecIO<T>::from_f32(0.0f);
}

void relu_bwd_mask_launch(DT dt, const void* dy, const void* y, void* dx,
                          int64_t n, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_relu_bwd_mask<float>, dim3(ew_blocks(n)), dim3(256), 0,
                       s, (const float*)dy, (const float*)y, (float*)dx, n);
  else
    hipLaunchKernelGGL(k_relu_bwd_mask<bf16>, dim3(ew_blocks(n)), dim3(256), 0,
                       s, (const bf16*)dy, (const bf16*)y, (bf16*)dx, n);
}

// ---- dropout (Philox 4x32; reference dropout.cu:33 vectorized form) --------
template <typename T>
__global__ void k_dropout_fwd(const T* x, T* y, uint8_t* mask, int64_t n,
                              float p, uint64_t seed,
                              const int64_t* __restrict__ ctr) {
  // hipGraph-captured step: the per-call salt is baked at capture, the
  // device counter varies per replay -> fresh Philox stream each step
  if (ctr) seed ^= (uint64_t)(*ctr) * 0x9E3779B97F4A7C15ull;
  const float scale = 1.0f / (1.0f - p);
  Philox rng(seed);
  int64_t i4 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i4 * 4 < n; i4 += stride) {
    uint4 r = rng(i4);
    unsigned int rs[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int64_t i = i4 * 4 + j;
      if (i < n) {
        bool keep = u32_to_uniform(rs[j]) > p;
        mask[i] = keep;
        y[i] = keep ? VecIO<T>::from_f32(VecIO<T>::to_f32(x[i]) * scale)
                    : VecIO<T>::from_f32(0.0f);
      }
    }
  }
}

template <typename T>
__global__ void k_dropout_bwd(const T* dy, const uint8_t* mask, T* dx,
                              int64_t n, float p) {
  const float scale = 1.0f / (1.0f - p);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    dx[i] = mask[i] ? VecIO<T>::from_f32(VecIO<T>::to_f32(dy[i]) * scale)
                    : VecIO<T>::from_f32(0.0f);
}

void dropout_fwd_launch(DT dt, const void* x, void* y, uint8_t* mask,
                        int64_t n, float p, uint64_t seed,
                        const int64_t* ctr, hipStream_t s) {
  int blocks = ew_blocks((n + 3) / 4);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_dropout_fwd<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)x, (float*)y, mask, n, p, seed, ctr);
  else
    hipLaunchKernelGGL(k_dropout_fwd<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)x, (bf16*)y, mask, n, p, seed, ctr);
}

void dropout_bwd_launch(DT dt, const void* dy, const uint8_t* mask, void* dx,
                        int64_t n, float p, hipStream_t s) {
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_dropout_bwd<float>, dim3(ew_blocks(n)), dim3(256), 0,
                       s, (const float*)dy, mask, (float*)dx, n, p);
  else
    hipLaunchKernelGGL(k_dropout_bwd<bf16>, dim3(ew_blocks(n)), dim3(256), 0,
                       s, (const bf16*)dy, mask, (bf16*)dx, n, p);
}

// ---- column sum (bias gradient; reference run_bgrad_kernel_ex) -------------
// slab finalize: sum S slabs of C2 fp32 columns (written per row-slice by
// the column-reduce kernels below — plain stores into an at::empty ws,
// replacing per-(block,col) atomics whose same-address serialization cost
// a fixed ~20-50us at 512 row slices, and the at::zeros the atomics
// needed). Block owns 32 columns; 8 thread-rows split the S range.
template <int CPB>  // columns per block; 256/CPB threads fold slices per col
__global__ void k_slab_fin(const float* __restrict__ ws,
                           float* __restrict__ out, int S, int C2) {
  constexpr int RPB = 256 / CPB;
  __shared__ float sh[256];
  const int cl = threadIdx.x % CPB;
  const int sl = threadIdx.x / CPB;
  const int c = blockIdx.x * CPB + cl;
  float a = 0.0f;
  if (c < C2)
    for (int s = sl; s < S; s += RPB) a += ws[(int64_t)s * C2 + c];
  sh[sl * CPB + cl] = a;
  __syncthreads();
  // tree-fold the slice rows (RPB is a power of two)
  for (int h = RPB / 2; h > 0; h >>= 1) {
    if (sl < h) sh[sl * CPB + cl] += sh[(sl + h) * CPB + cl];
    __syncthreads();
  }
  if (sl == 0 && c < C2) out[c] = sh[cl];
}

void slab_fin_launch(const float* ws, float* out, int slices, int c2,
                     hipStream_t s) {
  // pick columns-per-block so ~128+ blocks launch regardless of C: the
  // slice fold is latency-bound and needs the parallelism spread over
  // slices, not columns
#define SLAB_CASE(CPB)                                                       \
  hipLaunchKernelGGL(k_slab_fin<CPB>, dim3((c2 + CPB - 1) / CPB), dim3(256), \
                     0, s, ws, out, slices, c2)
  if (c2 >= 128 * 32) SLAB_CASE(32);
  else if (c2 >= 128 * 16) SLAB_CASE(16);
  else if (c2 >= 128 * 8) SLAB_CASE(8);
  else if (c2 >= 128 * 4) SLAB_CASE(4);
  else if (c2 >= 128 * 2) SLAB_CASE(2);
  else SLAB_CASE(1);
#undef SLAB_CASE
}

// x: [rows, cols] -> out_f32[cols]; each block owns a col-chunk x row-slice,
// partials combined with one atomic per (block, col).
template <typename T>
__global__ void k_colsum(const T* __restrict__ x, float* __restrict__ out,
                         int64_t rows, int cols) {
  const int col = blockIdx.x * 64 + (threadIdx.x & 63);
  const int rslice = blockIdx.y;
  const int nrs = gridDim.y;
  const int64_t r0 = rows * rslice / nrs, r1 = rows * (rslice + 1) / nrs;
  float acc = 0.0f;
  if (col < cols) {
    // 4 waves stride over the block's row slice; lanes = consecutive
    // cols; 4 independent accumulators keep 4 loads in flight per wave
    // (single-chain form measured 1.7 TB/s on the odd-vocab head bias)
    float a1 = 0.0f, a2 = 0.0f, a3 = 0.0f;
    int64_t r = r0 + (threadIdx.x >> 6);
    for (; r + 12 < r1; r += 16) {
      acc += VecIO<T>::to_f32(x[r * cols + col]);
      a1 += VecIO<T>::to_f32(x[(r + 4) * cols + col]);
      a2 += VecIO<T>::to_f32(x[(r + 8) * cols + col]);
      a3 += VecIO<T>::to_f32(x[(r + 12) * cols + col]);
    }
    for (; r < r1; r += 4) acc += VecIO<T>::to_f32(x[r * cols + col]);
    acc += (a1 + a2) + a3;
  }
  __shared__ float sh[4][64];
  sh[threadIdx.x >> 6][threadIdx.x & 63] = acc;
  __syncthreads();
  if (threadIdx.x < 64 && col < cols) {
    float v = sh[0][threadIdx.x] + sh[1][threadIdx.x] + sh[2][threadIdx.x] +
              sh[3][threadIdx.x];
    if (nrs == 1)
      out[col] = v;
    else
      atomicAdd(&out[col], v);
  }
}

// vectorized colsum: V consecutive cols per thread via one 16B load per
// iteration (the scalar kernel's 2B loads + single dependent chain leave
// ~4x bandwidth on the table); LDS combine, one atomic per col.
template <typename T>
__global__ void k_colsum_vec(const T* __restrict__ x, float* __restrict__ out,
                             int64_t rows, int cols) {
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  const int groups = cols / V;
  const int gpb = min(groups, 256);
  const int rows_per_iter = 256 / gpb;
  const int g = threadIdx.x % gpb + blockIdx.x * gpb;
  const int r_off = threadIdx.x / gpb;
  __shared__ float sh[256 * V];
  float s[V] = {};
  if (g < groups && r_off < rows_per_iter) {
    const int64_t r0 = rows * blockIdx.y / gridDim.y;
    const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
#pragma unroll 4
    for (int64_t r = r0 + r_off; r < r1; r += rows_per_iter) {
      P v = *(const P*)&x[r * cols + (int64_t)g * V];
#pragma unroll
      for (int j = 0; j < V; ++j) s[j] += VecIO<T>::to_f32(v.e[j]);
    }
  }
#pragma unroll
  for (int j = 0; j < V; ++j) sh[threadIdx.x * V + j] = s[j];
  __syncthreads();
  const int cpb_total = gpb * V;
  for (int cl = threadIdx.x; cl < cpb_total; cl += 256) {
    int gl = cl / V, j = cl % V;
    float t = 0.0f;
    for (int r = 0; r < rows_per_iter; ++r) t += sh[(r * gpb + gl) * V + j];
    int c = blockIdx.x * cpb_total + cl;
    if (c >= cols) break;  // last block may cover fewer than gpb groups
    if (gridDim.y == 1)
      out[c] = t;
    else  // per-slice slab (out = ws base); k_slab_fin sums the slices
      out[(int64_t)blockIdx.y * cols + c] = t;
  }
}

int colsum_ws_slices(DT dt, const void* x, int64_t rows, int64_t cols) {
  int V = dt == DT::F32 ? 4 : 8;
  bool vec = (((uintptr_t)x & 15) == 0) && cols % V == 0;
  if (!vec) return 0;  // atomic fallback (binding must zero the output)
  int groups = (int)(cols / V);
  int gpb = groups < 256 ? groups : 256;
  int cblocks = (groups + gpb - 1) / gpb;
  int rows_per_iter = 256 / gpb;
  int64_t iters = (rows + rows_per_iter - 1) / rows_per_iter;
  return (int)std::min<int64_t>(std::max<int64_t>(512 / cblocks, 1),
                                std::max<int64_t>(iters / 4, 1));
}

void colsum_launch(DT dt, const void* x, void* out_f32, float* ws,
                   int rslices, int64_t rows, int64_t cols, hipStream_t s) {
  if (rslices >= 1) {
    int V = dt == DT::F32 ? 4 : 8;
    int groups = (int)(cols / V);
    int gpb = groups < 256 ? groups : 256;
    int cblocks = (groups + gpb - 1) / gpb;
    dim3 grid(cblocks, rslices);
    float* target = rslices > 1 ? ws : (float*)out_f32;
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_colsum_vec<float>, grid, dim3(256), 0, s,
                         (const float*)x, target, rows, (int)cols);
    else
      hipLaunchKernelGGL(k_colsum_vec<bf16>, grid, dim3(256), 0, s,
                         (const bf16*)x, target, rows, (int)cols);
    if (rslices > 1)
      slab_fin_launch(ws, (float*)out_f32, rslices, (int)cols, s);
    return;
  }
  // fill the chip: one block per ~64-row slice, capped so atomics stay cheap
  int cblocks = (int)((cols + 63) / 64);
  int cap = std::max(1, 8192 / cblocks);
  int rsl = (int)std::min<int64_t>((rows + 63) / 64, (int64_t)cap);
  dim3 grid(cblocks, rsl);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_colsum<float>, grid, dim3(256), 0, s, (const float*)x,
                       (float*)out_f32, rows, (int)cols);
  else
    hipLaunchKernelGGL(k_colsum<bf16>, grid, dim3(256), 0, s, (const bf16*)x,
                       (float*)out_f32, rows, (int)cols);
}

// ---- f32 -> T cast ---------------------------------------------------------
template <typename T>
__global__ void k_cast(const float* x, T* y, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) y[i] = VecIO<T>::from_f32(x[i]);
}

void cast_f32_launch(DT dt_out, const float* x, void* y, int64_t n,
                     hipStream_t s) {
  if (dt_out == DT::F32)
    hipLaunchKernelGGL(k_cast<float>, dim3(ew_blocks(n)), dim3(256), 0, s, x,
                       (float*)y, n);
  else
    hipLaunchKernelGGL(k_cast<bf16>, dim3(ew_blocks(n)), dim3(256), 0, s, x,
                       (bf16*)y, n);
}

}  // namespace tnn
