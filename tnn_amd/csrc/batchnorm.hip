// Fused NHWC BatchNorm (+ReLU) kernels
// (reference src/nn/layers_impl/cuda/batchnorm_nchw_ops.cu:119-449 and the
// cuDNN-fe BN+ReLU graphs, cudnn_batchnorm_ops.cu:159).
//
// NHWC: channels are the fast dim, so thread t of a block covers channel
// (t % CPB) — global loads are perfectly coalesced across channels. All
// statistics accumulate in fp32 regardless of io dtype (SURVEY §7 hard
// part 5). Grid-level reduction: per-block partial sums combined with
// fp32 atomics, then a tiny finalize kernel.

#include "common.h"
#include "kernels.h"

namespace tnn {

// 16-byte load pack (ext_vector_type cannot hold the __hip_bfloat16 struct)
template <typename T> struct alignas(16) BnPack {
  T e[16 / sizeof(T)];
};

// pass 1a: partial sum / sumsq per channel, atomics into f32 buffers
template <typename T>
__global__ void k_bn_partial(const T* __restrict__ x, float* __restrict__ sum,
                             float* __restrict__ sumsq, int64_t rows,
                             int cols) {
  const int cpb = min(cols, 256);
  const int rows_per_iter = 256 / cpb;
  const int c = threadIdx.x % cpb + blockIdx.x * cpb;
  const int r_off = threadIdx.x / cpb;
  if (c >= cols || r_off >= rows_per_iter) return;
  const int64_t r0 = rows * blockIdx.y / gridDim.y;
  const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
  float s = 0.0f, ss = 0.0f;
  for (int64_t r = r0 + r_off; r < r1; r += rows_per_iter) {
    float v = VecIO<T>::to_f32(x[r * cols + c]);
    s += v;
    ss += v * v;
  }
  // combine the rows_per_iter partials for one channel via LDS
  __shared__ float sh_s[256], sh_ss[256];
  sh_s[threadIdx.x] = s;
  sh_ss[threadIdx.x] = ss;
  __syncthreads();
  if (r_off == 0) {
    for (int j = 1; j < rows_per_iter; ++j) {
      s += sh_s[threadIdx.x + j * cpb];
      ss += sh_ss[threadIdx.x + j * cpb];
    }
    atomicAdd(&sum[c], s);
    atomicAdd(&sumsq[c], ss);
  }
}

// Vectorized reduce variants (cols % V == 0, 16B-aligned): the scalar
// kernels issue one 2B/4B load per iteration feeding two short dependent
// FMA chains -- measured ~1.4 TB/s. 16B loads + V independent accumulator
// lanes per thread give the MLP to reach the HBM roofline. Each thread owns
// V consecutive channels; per-block partials combine through LDS and emit
// one atomicAdd per channel.
template <typename T>
__global__ void k_bn_partial_vec(const T* __restrict__ x,
                                 float* __restrict__ sum,
                                 float* __restrict__ sumsq, int64_t rows,
                                 int cols) {
  constexpr int V = 16 / sizeof(T);
  using VecT = BnPack<T>;
  const int groups = cols / V;
  const int gpb = min(groups, 256);
  const int rows_per_iter = 256 / gpb;
  const int g = threadIdx.x % gpb + blockIdx.x * gpb;
  const int r_off = threadIdx.x / gpb;
  __shared__ float sh[2 * 256 * V];
  float* sh_s = sh;
  float* sh_ss = sh + 256 * V;
  float s[V] = {}, ss[V] = {};
  if (g < groups && r_off < rows_per_iter) {
    const int64_t r0 = rows * blockIdx.y / gridDim.y;
    const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
#pragma unroll 4
    for (int64_t r = r0 + r_off; r < r1; r += rows_per_iter) {
      VecT v = *(const VecT*)&x[r * cols + (int64_t)g * V];
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float f = VecIO<T>::to_f32(v.e[j]);
        s[j] += f;
        ss[j] += f * f;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < V; ++j) {
    sh_s[threadIdx.x * V + j] = s[j];
    sh_ss[threadIdx.x * V + j] = ss[j];
  }
  __syncthreads();
  const int cpb_total = gpb * V;
  for (int cl = threadIdx.x; cl < cpb_total; cl += 256) {
    int gl = cl / V, j = cl % V;
    float ts = 0.0f, tss = 0.0f;
    for (int r = 0; r < rows_per_iter; ++r) {
      ts += sh_s[(r * gpb + gl) * V + j];
      tss += sh_ss[(r * gpb + gl) * V + j];
    }
    int c = blockIdx.x * cpb_total + cl;
    if (c >= cols) break;  // last block may cover fewer than gpb groups
    // per-slice slab [y][2][cols] (sum/sumsq are a contiguous {2,C}
    // buffer); k_slab_fin folds the slices — no atomics, no zero-init
    sum[(int64_t)blockIdx.y * 2 * cols + c] = ts;
    sum[(int64_t)blockIdx.y * 2 * cols + cols + c] = tss;
  }
}

template <typename T>
__global__ void k_bn_bwd_reduce_vec(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y_relu,
                                    const float* mean, const float* invstd,
                                    float* __restrict__ sum_dy,
                                    float* __restrict__ sum_dy_xhat,
                                    int64_t rows, int cols, float dy_scale) {
  constexpr int V = 16 / sizeof(T);
  using VecT = BnPack<T>;
  const int groups = cols / V;
  const int gpb = min(groups, 256);
  const int rows_per_iter = 256 / gpb;
  const int g = threadIdx.x % gpb + blockIdx.x * gpb;
  const int r_off = threadIdx.x / gpb;
  __shared__ float sh[2 * 256 * V];
  float* sh_s = sh;
  float* sh_sx = sh + 256 * V;
  float s[V] = {}, sx[V] = {};
  if (g < groups && r_off < rows_per_iter) {
    const int64_t r0 = rows * blockIdx.y / gridDim.y;
    const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
    float m[V], is[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      m[j] = mean[g * V + j];
      is[j] = invstd[g * V + j];
    }
#pragma unroll 2
    for (int64_t r = r0 + r_off; r < r1; r += rows_per_iter) {
      int64_t i = r * cols + (int64_t)g * V;
      VecT vg = *(const VecT*)&dy[i];
      VecT vx = *(const VecT*)&x[i];
      if (y_relu) {
        VecT vy = *(const VecT*)&y_relu[i];
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float gr = VecIO<T>::to_f32(vy.e[j]) > 0.0f
                         ? VecIO<T>::to_f32(vg.e[j]) * dy_scale : 0.0f;
          s[j] += gr;
          sx[j] += gr * (VecIO<T>::to_f32(vx.e[j]) - m[j]) * is[j];
        }
      } else {
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float gr = VecIO<T>::to_f32(vg.e[j]) * dy_scale;
          s[j] += gr;
          sx[j] += gr * (VecIO<T>::to_f32(vx.e[j]) - m[j]) * is[j];
        }
      }
    }
  }
#pragma unroll
  for (int j = 0; j < V; ++j) {
    sh_s[threadIdx.x * V + j] = s[j];
    sh_sx[threadIdx.x * V + j] = sx[j];
  }
  __syncthreads();
  const int cpb_total = gpb * V;
  for (int cl = threadIdx.x; cl < cpb_total; cl += 256) {
    int gl = cl / V, j = cl % V;
    float ts = 0.0f, tsx = 0.0f;
    for (int r = 0; r < rows_per_iter; ++r) {
      ts += sh_s[(r * gpb + gl) * V + j];
      tsx += sh_sx[(r * gpb + gl) * V + j];
    }
    int c = blockIdx.x * cpb_total + cl;
    if (c >= cols) break;  // last block may cover fewer than gpb groups
    // per-slice slab [y][2][cols] (sum_dy/sum_dy_xhat contiguous {2,C})
    sum_dy[(int64_t)blockIdx.y * 2 * cols + c] = ts;
    sum_dy[(int64_t)blockIdx.y * 2 * cols + cols + c] = tsx;
  }
}

__global__ void k_bn_finalize(float* mean, float* invstd, const float* sum,
                              const float* sumsq, float* rmean, float* rvar,
                              float momentum, int64_t rows, int cols,
                              float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= cols) return;
  float m = sum[c] / rows;
  float var = fmaxf(sumsq[c] / rows - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (rmean) {  // fused running-stat update (unbiased var, torch semantics)
    float unbiased = var * ((float)rows / (float)(rows > 1 ? rows - 1 : 1));
    rmean[c] = rmean[c] * (1.0f - momentum) + m * momentum;
    rvar[c] = rvar[c] * (1.0f - momentum) + unbiased * momentum;
  }
}

template <typename T>
__global__ void k_bn_apply(const T* __restrict__ x, const float* mean,
                           const float* invstd, const float* gamma,
                           const float* beta, T* __restrict__ y, int64_t n,
                           int cols, bool relu) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int c = i % cols;
    float v = (VecIO<T>::to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
    if (relu) v = fmaxf(v, 0.0f);
    y[i] = VecIO<T>::from_f32(v);
  }
}

// fused BN+ReLU+dropout apply (train, relu only): dropped positions write
// 0 so the saved output doubles as both the relu mask and the dropout
// mask for backward -- no mask tensor, no separate dropout pass. Kept
// positions are pre-scaled by 1/(1-p); backward just scales dy by the
// same constant wherever the saved output is > 0.
template <typename T>
__global__ void k_bn_apply_drop(const T* __restrict__ x, const float* mean,
                                const float* invstd, const float* gamma,
                                const float* beta, T* __restrict__ y,
                                int64_t n, int cols, float p, uint64_t seed,
                                const int64_t* __restrict__ ctr) {
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
        int c = i % cols;
        float v =
            (VecIO<T>::to_f32(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
        v = fmaxf(v, 0.0f);
        bool keep = u32_to_uniform(rs[j]) > p;
        y[i] = VecIO<T>::from_f32(keep ? v * scale : 0.0f);
      }
    }
  }
}

// vectorized fused BN+ReLU+dropout: one 16B load/store per thread-iter,
// two Philox draws per vector (the scalar form measured 1.65 TB/s vs ~5
// for plain bn_apply)
template <typename T>
__global__ void k_bn_apply_drop_vec(const T* __restrict__ x, const float* mean,
                                    const float* invstd, const float* gamma,
                                    const float* beta, T* __restrict__ y,
                                    int64_t nv, int groups, float p,
                                    uint64_t seed,
                                    const int64_t* __restrict__ ctr) {
  if (ctr) seed ^= (uint64_t)(*ctr) * 0x9E3779B97F4A7C15ull;
  constexpr int V = 16 / sizeof(T);
  using VecT = BnPack<T>;
  const float scale = 1.0f / (1.0f - p);
  Philox rng(seed);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nv; i += stride) {
    int c0 = (int)((uint32_t)i % (uint32_t)groups) * V;
    VecT vx = ((const VecT*)x)[i];
    uint4 r0 = rng(2 * i);
    uint4 r1 = rng(2 * i + 1);
    unsigned int rs[8] = {r0.x, r0.y, r0.z, r0.w, r1.x, r1.y, r1.z, r1.w};
    VecT vy;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      int c = c0 + j;
      float v =
          (VecIO<T>::to_f32(vx.e[j]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
      v = fmaxf(v, 0.0f);
      bool keep = u32_to_uniform(rs[j]) > p;
      vy.e[j] = VecIO<T>::from_f32(keep ? v * scale : 0.0f);
    }
    ((VecT*)y)[i] = vy;
  }
}

template <typename T>
__global__ void k_bn_infer(const T* __restrict__ x, const float* rmean,
                           const float* rvar, const float* gamma,
                           const float* beta, T* __restrict__ y, int64_t n,
                           int cols, float eps, bool relu) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int c = i % cols;
    float inv = rsqrtf(rvar[c] + eps);
    float v = (VecIO<T>::to_f32(x[i]) - rmean[c]) * inv * gamma[c] + beta[c];
    if (relu) v = fmaxf(v, 0.0f);
    y[i] = VecIO<T>::from_f32(v);
  }
}

// backward reduce: sum_dy[c], sum_dy_xhat[c] (dy masked by relu output)
template <typename T>
__global__ void k_bn_bwd_reduce(const T* __restrict__ x, const T* __restrict__ dy,
                                const T* __restrict__ y_relu, const float* mean,
                                const float* invstd, float* __restrict__ sum_dy,
                                float* __restrict__ sum_dy_xhat, int64_t rows,
                                int cols, float dy_scale) {
  const int cpb = min(cols, 256);
  const int rows_per_iter = 256 / cpb;
  const int c = threadIdx.x % cpb + blockIdx.x * cpb;
  const int r_off = threadIdx.x / cpb;
  if (c >= cols || r_off >= rows_per_iter) return;
  const int64_t r0 = rows * blockIdx.y / gridDim.y;
  const int64_t r1 = rows * (blockIdx.y + 1) / gridDim.y;
  const float m = mean[c], is = invstd[c];
  float s = 0.0f, sx = 0.0f;
  for (int64_t r = r0 + r_off; r < r1; r += rows_per_iter) {
    int64_t i = r * cols + c;
    float g = VecIO<T>::to_f32(dy[i]) * dy_scale;
    if (y_relu && VecIO<T>::to_f32(y_relu[i]) <= 0.0f) g = 0.0f;
    float xhat = (VecIO<T>::to_f32(x[i]) - m) * is;
    s += g;
    sx += g * xhat;
  }
  __shared__ float sh_s[256], sh_sx[256];
  sh_s[threadIdx.x] = s;
  sh_sx[threadIdx.x] = sx;
  __syncthreads();
  if (r_off == 0) {
    for (int j = 1; j < rows_per_iter; ++j) {
      s += sh_s[threadIdx.x + j * cpb];
      sx += sh_sx[threadIdx.x + j * cpb];
    }
    atomicAdd(&sum_dy[c], s);
    atomicAdd(&sum_dy_xhat[c], sx);
  }
}

template <typename T>
__global__ void k_bn_bwd_apply(const T* __restrict__ x, const T* __restrict__ dy,
                               const T* __restrict__ y_relu, const float* mean,
                               const float* invstd, const float* gamma,
                               const float* sum_dy, const float* sum_dy_xhat,
                               T* __restrict__ dx, const T* __restrict__ resid,
                               int64_t n, int64_t rows, int cols,
                               float dy_scale) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv_n = 1.0f / (float)rows;
  for (; i < n; i += stride) {
    int c = i % cols;
    float g = VecIO<T>::to_f32(dy[i]) * dy_scale;
    if (y_relu && VecIO<T>::to_f32(y_relu[i]) <= 0.0f) g = 0.0f;
    float xhat = (VecIO<T>::to_f32(x[i]) - mean[c]) * invstd[c];
    float v = gamma[c] * invstd[c] *
              (g - inv_n * (sum_dy[c] + xhat * sum_dy_xhat[c]));
    if (resid) v += VecIO<T>::to_f32(resid[i]);
    dx[i] = VecIO<T>::from_f32(v);
  }
}

// vectorized train-apply: V channel-contiguous elems per thread
template <typename T>
__global__ void k_bn_apply_vec(const T* __restrict__ x, const float* mean,
                               const float* invstd, const float* gamma,
                               const float* beta, T* __restrict__ y,
                               int64_t nv, int groups, bool relu) {
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nv; i += stride) {
    const int c0 = (int)(i % groups) * V;
    P vx = ((const P*)x)[i];
    P o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const int c = c0 + j;
      float v = (VecIO<T>::to_f32(vx.e[j]) - mean[c]) * invstd[c] * gamma[c] +
                beta[c];
      if (relu) v = fmaxf(v, 0.0f);
      o.e[j] = VecIO<T>::from_f32(v);
    }
    ((P*)y)[i] = o;
  }
}

// vectorized backward apply: V channel-contiguous elems per thread, one
// division per vector (the scalar form's per-element i%cols + 2B loads
// measured 2.2 TB/s on a ~5 TB/s pass)
template <typename T>
__global__ void k_bn_bwd_apply_vec(const T* __restrict__ x,
                                   const T* __restrict__ dy,
                                   const T* __restrict__ y_relu,
                                   const float* mean, const float* invstd,
                                   const float* gamma, const float* sum_dy,
                                   const float* sum_dy_xhat,
                                   T* __restrict__ dx,
                                   const T* __restrict__ resid, int64_t nv,
                                   int groups, int64_t rows, float dy_scale) {
  // resid: passthrough-residual grad added into dx (pre-activation
  // residual blocks route the junction's grad join through bn_bwd)
  constexpr int V = 16 / sizeof(T);
  struct alignas(16) P { T e[16 / sizeof(T)]; };
  const float inv_n = 1.0f / (float)rows;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nv; i += stride) {
    const int c0 = (int)(i % groups) * V;
    P vx = ((const P*)x)[i];
    P vd = ((const P*)dy)[i];
    P vr;
    if (y_relu) vr = ((const P*)y_relu)[i];
    P vres;
    if (resid) vres = ((const P*)resid)[i];
    P o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const int c = c0 + j;
      float g = VecIO<T>::to_f32(vd.e[j]) * dy_scale;
      if (y_relu && VecIO<T>::to_f32(vr.e[j]) <= 0.0f) g = 0.0f;
      const float xhat = (VecIO<T>::to_f32(vx.e[j]) - mean[c]) * invstd[c];
      float v = gamma[c] * invstd[c] *
                (g - inv_n * (sum_dy[c] + xhat * sum_dy_xhat[c]));
      if (resid) v += VecIO<T>::to_f32(vres.e[j]);
      o.e[j] = VecIO<T>::from_f32(v);
    }
    ((P*)dx)[i] = o;
  }
}

// ---------------------------------------------------------------------------
template <typename T>
static inline bool bn_vec_ok(const void* x, int cols) {
  constexpr int V = 16 / sizeof(T);
  return cols % V == 0 && (((uintptr_t)x & 15) == 0);
}

template <typename T>
static inline dim3 bn_reduce_grid_vec(int64_t rows, int cols) {
  constexpr int V = 16 / sizeof(T);
  int groups = cols / V;
  int gpb = groups < 256 ? groups : 256;
  int cblocks = (groups + gpb - 1) / gpb;
  int rows_per_iter = 256 / gpb;
  int64_t iters = (rows + rows_per_iter - 1) / rows_per_iter;
  // ~512 blocks (iters/32 capped the 128-col WRN reduces at 128 blocks,
  // 1.7 TB/s on a ~6 TB/s-roofline pass)
  int rslices = (int)std::min<int64_t>(std::max<int64_t>(512 / cblocks, 1),
                                       std::max<int64_t>(iters / 4, 1));
  return dim3(cblocks, rslices);
}

static inline dim3 bn_reduce_grid(int64_t rows, int cols) {
  int cpb = cols < 256 ? cols : 256;
  int cblocks = (cols + cpb - 1) / cpb;
  int rows_per_iter = 256 / cpb;
  // fill the chip: >=2048 blocks where the row count allows, >=16
  // iterations per block so atomics stay a rounding error
  int64_t iters = (rows + rows_per_iter - 1) / rows_per_iter;
  int rslices = (int)std::min<int64_t>(std::max<int64_t>(512 / cblocks, 1),
                                       std::max<int64_t>(iters / 4, 1));
  return dim3(cblocks, rslices);
}

int64_t bn_stats_ws_floats(DT dt, const void* x, int64_t rows, int cols) {
  // vec path: [slices][2C] slabs + a [2C] finalize region; 0 = scalar path
  const bool vec = dt == DT::F32 ? bn_vec_ok<float>((void*)x, cols)
                                 : bn_vec_ok<bf16>((void*)x, cols);
  if (!vec) return 0;
  dim3 g = dt == DT::F32 ? bn_reduce_grid_vec<float>(rows, cols)
                         : bn_reduce_grid_vec<bf16>(rows, cols);
  return ((int64_t)g.y + 1) * 2 * cols;
}

void bn_stats_launch(DT dt, const void* x, float* mean, float* invstd,
                     float* rmean, float* rvar, float momentum, float* ws,
                     int64_t rows, int cols, float eps, hipStream_t s) {
  if (ws) {
    // vec path: per-slice slabs (plain stores) + slab finalize — no
    // atomics, no zero-init
    dim3 g = dt == DT::F32 ? bn_reduce_grid_vec<float>(rows, cols)
                           : bn_reduce_grid_vec<bf16>(rows, cols);
    float* sums = ws + (int64_t)g.y * 2 * cols;
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_bn_partial_vec<float>, g, dim3(256), 0, s,
                         (const float*)x, ws, nullptr, rows, cols);
    else
      hipLaunchKernelGGL(k_bn_partial_vec<bf16>, g, dim3(256), 0, s,
                         (const bf16*)x, ws, nullptr, rows, cols);
    slab_fin_launch(ws, sums, (int)g.y, 2 * cols, s);
    hipLaunchKernelGGL(k_bn_finalize, dim3((cols + 255) / 256), dim3(256), 0,
                       s, mean, invstd, sums, sums + cols, rmean, rvar,
                       momentum, rows, cols, eps);
    return;
  }
  // scalar fallback: mean/invstd double as zeroed atomic sum/sumsq
  hipMemsetAsync(mean, 0, cols * sizeof(float), s);
  hipMemsetAsync(invstd, 0, cols * sizeof(float), s);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_bn_partial<float>, bn_reduce_grid(rows, cols),
                       dim3(256), 0, s, (const float*)x, mean, invstd, rows,
                       cols);
  else
    hipLaunchKernelGGL(k_bn_partial<bf16>, bn_reduce_grid(rows, cols),
                       dim3(256), 0, s, (const bf16*)x, mean, invstd, rows,
                       cols);
  hipLaunchKernelGGL(k_bn_finalize, dim3((cols + 255) / 256), dim3(256), 0, s,
                     mean, invstd, mean, invstd, rmean, rvar, momentum, rows,
                     cols, eps);
}

// finalize from precomputed sums (conv-epilogue fused stats): skips the
// partial-reduce pass entirely
void bn_finalize_launch(const float* sum, const float* sumsq, float* mean,
                        float* invstd, float* rmean, float* rvar,
                        float momentum, int64_t rows, int cols, float eps,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_bn_finalize, dim3((cols + 255) / 256), dim3(256), 0, s,
                     mean, invstd, sum, sumsq, rmean, rvar, momentum, rows,
                     cols, eps);
}

void bn_apply_launch(DT dt, const void* x, const float* mean,
                     const float* invstd, const float* gamma, const float* beta,
                     void* y, int64_t rows, int cols, bool relu,
                     hipStream_t s) {
  int64_t n = rows * cols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (dt == DT::F32) {
    if (bn_vec_ok<float>(x, cols)) {
      hipLaunchKernelGGL(k_bn_apply_vec<float>, dim3(blocks), dim3(256), 0, s,
                         (const float*)x, mean, invstd, gamma, beta, (float*)y,
                         n / 4, cols / 4, relu);
      return;
    }
    hipLaunchKernelGGL(k_bn_apply<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)x, mean, invstd, gamma, beta, (float*)y, n,
                       cols, relu);
  } else {
    if (bn_vec_ok<bf16>(x, cols)) {
      hipLaunchKernelGGL(k_bn_apply_vec<bf16>, dim3(blocks), dim3(256), 0, s,
                         (const bf16*)x, mean, invstd, gamma, beta, (bf16*)y,
                         n / 8, cols / 8, relu);
      return;
    }
    hipLaunchKernelGGL(k_bn_apply<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)x, mean, invstd, gamma, beta, (bf16*)y, n,
                       cols, relu);
  }
}

void bn_apply_drop_launch(DT dt, const void* x, const float* mean,
                          const float* invstd, const float* gamma,
                          const float* beta, void* y, int64_t rows, int cols,
                          float p, uint64_t seed, const int64_t* ctr,
                          hipStream_t s) {
  int64_t n = rows * cols;
  int V = dt == DT::F32 ? 4 : 8;
  if (cols % V == 0 && (((uintptr_t)x & 15) == 0) && n < (1ll << 34)) {
    int64_t nv = n / V;
    int blocks = (int)std::min<int64_t>((nv + 255) / 256, (int64_t)2048);
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_bn_apply_drop_vec<float>, dim3(blocks), dim3(256),
                         0, s, (const float*)x, mean, invstd, gamma, beta,
                         (float*)y, nv, cols / V, p, seed, ctr);
    else
      hipLaunchKernelGGL(k_bn_apply_drop_vec<bf16>, dim3(blocks), dim3(256),
                         0, s, (const bf16*)x, mean, invstd, gamma, beta,
                         (bf16*)y, nv, cols / V, p, seed, ctr);
    return;
  }
  int blocks = (int)std::min<int64_t>(((n + 3) / 4 + 255) / 256, (int64_t)2048);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_bn_apply_drop<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)x, mean, invstd, gamma, beta, (float*)y, n,
                       cols, p, seed, ctr);
  else
    hipLaunchKernelGGL(k_bn_apply_drop<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)x, mean, invstd, gamma, beta, (bf16*)y, n,
                       cols, p, seed, ctr);
}

void bn_infer_launch(DT dt, const void* x, const float* rmean,
                     const float* rvar, const float* gamma, const float* beta,
                     void* y, int64_t rows, int cols, float eps, bool relu,
                     hipStream_t s) {
  int64_t n = rows * cols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_bn_infer<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)x, rmean, rvar, gamma, beta, (float*)y, n,
                       cols, eps, relu);
  else
    hipLaunchKernelGGL(k_bn_infer<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)x, rmean, rvar, gamma, beta, (bf16*)y, n,
                       cols, eps, relu);
}

int64_t bn_bwd_ws_floats(DT dt, const void* x, const void* dy, int64_t rows,
                         int cols) {
  const bool vec = dt == DT::F32
                       ? (bn_vec_ok<float>((void*)x, cols) &&
                          bn_vec_ok<float>((void*)dy, cols))
                       : (bn_vec_ok<bf16>((void*)x, cols) &&
                          bn_vec_ok<bf16>((void*)dy, cols));
  if (!vec) return 0;
  dim3 g = dt == DT::F32 ? bn_reduce_grid_vec<float>(rows, cols)
                         : bn_reduce_grid_vec<bf16>(rows, cols);
  return (int64_t)g.y * 2 * cols;
}

void bn_bwd_reduce_launch(DT dt, const void* x, const void* dy,
                          const void* y_relu, const float* mean,
                          const float* invstd, float* sum_dy,
                          float* sum_dy_xhat, float* ws, int64_t rows,
                          int cols, float dy_scale, hipStream_t s) {
  if (ws) {
    // vec path: slabs (plain stores) + slab finalize into the contiguous
    // {2, C} sums buffer (sum_dy_xhat == sum_dy + C)
    dim3 g = dt == DT::F32 ? bn_reduce_grid_vec<float>(rows, cols)
                           : bn_reduce_grid_vec<bf16>(rows, cols);
    if (dt == DT::F32)
      hipLaunchKernelGGL(k_bn_bwd_reduce_vec<float>, g, dim3(256), 0, s,
                         (const float*)x, (const float*)dy,
                         (const float*)y_relu, mean, invstd, ws, nullptr,
                         rows, cols, dy_scale);
    else
      hipLaunchKernelGGL(k_bn_bwd_reduce_vec<bf16>, g, dim3(256), 0, s,
                         (const bf16*)x, (const bf16*)dy, (const bf16*)y_relu,
                         mean, invstd, ws, nullptr, rows, cols, dy_scale);
    slab_fin_launch(ws, sum_dy, (int)g.y, 2 * cols, s);
    return;
  }
  if (dt == DT::F32)
    hipLaunchKernelGGL(k_bn_bwd_reduce<float>, bn_reduce_grid(rows, cols),
                       dim3(256), 0, s, (const float*)x, (const float*)dy,
                       (const float*)y_relu, mean, invstd, sum_dy,
                       sum_dy_xhat, rows, cols, dy_scale);
  else
    hipLaunchKernelGGL(k_bn_bwd_reduce<bf16>, bn_reduce_grid(rows, cols),
                       dim3(256), 0, s, (const bf16*)x, (const bf16*)dy,
                       (const bf16*)y_relu, mean, invstd, sum_dy,
                       sum_dy_xhat, rows, cols, dy_scale);
}

void bn_bwd_apply_launch(DT dt, const void* x, const void* dy,
                         const void* y_relu, const float* mean,
                         const float* invstd, const float* gamma,
                         const float* sum_dy, const float* sum_dy_xhat,
                         void* dx, const void* resid, int64_t rows, int cols,
                         float dy_scale, hipStream_t s) {
  int64_t n = rows * cols;
  int blocks = (int)std::min<int64_t>((n + 255) / 256, (int64_t)2048);
  const bool vec = dt == DT::F32 ? bn_vec_ok<float>(x, cols)
                                 : bn_vec_ok<bf16>(x, cols);
  if (dt == DT::F32) {
    if (vec && ((uintptr_t)dy & 15) == 0) {
      hipLaunchKernelGGL(k_bn_bwd_apply_vec<float>, dim3(blocks), dim3(256),
                         0, s, (const float*)x, (const float*)dy,
                         (const float*)y_relu, mean, invstd, gamma, sum_dy,
                         sum_dy_xhat, (float*)dx, (const float*)resid,
                         n / 4, cols / 4, rows, dy_scale);
      return;
    }
    hipLaunchKernelGGL(k_bn_bwd_apply<float>, dim3(blocks), dim3(256), 0, s,
                       (const float*)x, (const float*)dy, (const float*)y_relu,
                       mean, invstd, gamma, sum_dy, sum_dy_xhat, (float*)dx,
                       (const float*)resid, n, rows, cols, dy_scale);
  } else {
    if (vec && ((uintptr_t)dy & 15) == 0) {
      hipLaunchKernelGGL(k_bn_bwd_apply_vec<bf16>, dim3(blocks), dim3(256),
                         0, s, (const bf16*)x, (const bf16*)dy,
                         (const bf16*)y_relu, mean, invstd, gamma, sum_dy,
                         sum_dy_xhat, (bf16*)dx, (const bf16*)resid,
                         n / 8, cols / 8, rows, dy_scale);
      return;
    }
    hipLaunchKernelGGL(k_bn_bwd_apply<bf16>, dim3(blocks), dim3(256), 0, s,
                       (const bf16*)x, (const bf16*)dy, (const bf16*)y_relu,
                       mean, invstd, gamma, sum_dy, sum_dy_xhat, (bf16*)dx,
                       (const bf16*)resid, n, rows, cols, dy_scale);
  }
}

}  // namespace tnn
