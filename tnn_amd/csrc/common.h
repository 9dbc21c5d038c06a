// Shared helpers for tnn_amd CDNA4 (gfx950) kernels.
//
// Conventions:
//  - wave = 64 lanes; workgroups are multiples of 64 threads.
//  - bf16 data is loaded as short4/short8 vectors (hipcc does not
//    auto-vectorize bf16 loads).
//  - fp32 accumulation everywhere; MFMA f32 accumulators.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define DEV __device__ __forceinline__

using bf16 = __hip_bfloat16;

// ext-vector types matching MFMA operand register counts
using f32x4 = __attribute__((ext_vector_type(4))) float;
using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;   // 4 VGPRs
using s16x8 = __attribute__((ext_vector_type(8))) short;
using int4v = __attribute__((ext_vector_type(4))) int;

constexpr int WAVE = 64;

DEV float bf2f(bf16 v) { return __bfloat162float(v); }
DEV bf16 f2bf(float v) { return __float2bfloat16(v); }

// ---- type traits -----------------------------------------------------------
template <typename T> struct VecIO;
template <> struct VecIO<float> {
  using v4 = float4;                       // 16B
  static DEV float to_f32(float x) { return x; }
  static DEV float from_f32(float x) { return x; }
};
template <> struct VecIO<bf16> {
  using v4 = short4;                       // 8B (4 elems)
  static DEV float to_f32(bf16 x) { return __bfloat162float(x); }
  static DEV bf16 from_f32(float x) { return __float2bfloat16(x); }
};

// ---- wave/block reductions -------------------------------------------------
DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  return v;
}

// Block-wide reduce into a single float (thread 0). `scratch` needs
// blockDim.x/64 floats.
DEV float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = blockDim.x >> 6;
  v = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) v = wave_reduce_sum(v);
  return v;
}

DEV float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = blockDim.x >> 6;
  v = (threadIdx.x < nw) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) v = wave_reduce_max(v);
  return v;
}

// ---- activation epilogues (kinds match tnn_amd.ops.functional.ACT_KINDS) --
enum ActKind : int {
  ACT_LINEAR = 0, ACT_RELU = 1, ACT_GELU = 2, ACT_SIGMOID = 3,
  ACT_TANH = 4, ACT_ELU = 5, ACT_LEAKY_RELU = 6, ACT_SILU = 7,
};

DEV float act_apply(float x, int kind) {
  switch (kind) {
    case ACT_RELU: return fmaxf(x, 0.0f);
    case ACT_GELU: {  // tanh approximation (reference gelu_kernels.cu)
      const float c = 0.7978845608028654f;  // sqrt(2/pi)
      float t = tanhf(c * (x + 0.044715f * x * x * x));
      return 0.5f * x * (1.0f + t);
    }
    case ACT_SIGMOID: return 1.0f / (1.0f + __expf(-x));
    case ACT_TANH: return tanhf(x);
    case ACT_ELU: return x > 0.0f ? x : __expf(x) - 1.0f;
    case ACT_LEAKY_RELU: return x > 0.0f ? x : 0.01f * x;
    case ACT_SILU: return x / (1.0f + __expf(-x));
    default: return x;
  }
}

// gradient wrt input given x (pre-act) and y (post-act)
DEV float act_grad(float dy, float x, float y, int kind) {
  switch (kind) {
    case ACT_RELU: return y > 0.0f ? dy : 0.0f;
    case ACT_GELU: {
      const float c = 0.7978845608028654f;
      float x3 = x * x * x;
      float u = c * (x + 0.044715f * x3);
      float t = tanhf(u);
      float sech2 = 1.0f - t * t;
      return dy * (0.5f * (1.0f + t) +
                   0.5f * x * sech2 * c * (1.0f + 3.0f * 0.044715f * x * x));
    }
    case ACT_SIGMOID: return dy * y * (1.0f - y);
    case ACT_TANH: return dy * (1.0f - y * y);
    case ACT_ELU: return x > 0.0f ? dy : dy * (y + 1.0f);
    case ACT_LEAKY_RELU: return x > 0.0f ? dy : 0.01f * dy;
    case ACT_SILU: {
      float s = 1.0f / (1.0f + __expf(-x));
      return dy * (s + x * s * (1.0f - s));
    }
    default: return dy;
  }
}

// ---- Philox 4x32-10 counter RNG (reference dropout.cu curand analog) -------
struct Philox {
  unsigned int k0, k1;
  DEV Philox(unsigned long long seed) {
    k0 = (unsigned int)seed;
    k1 = (unsigned int)(seed >> 32);
  }
  static DEV unsigned int mulhi(unsigned int a, unsigned int b) {
    return (unsigned int)(((unsigned long long)a * b) >> 32);
  }
  DEV uint4 operator()(unsigned long long ctr) const {
    unsigned int c0 = (unsigned int)ctr, c1 = (unsigned int)(ctr >> 32);
    unsigned int c2 = 0, c3 = 0;
    unsigned int key0 = k0, key1 = k1;
#pragma unroll
    for (int r = 0; r < 10; ++r) {
      unsigned int lo0 = 0xD2511F53u * c0, hi0 = mulhi(0xD2511F53u, c0);
      unsigned int lo1 = 0xCD9E8D57u * c2, hi1 = mulhi(0xCD9E8D57u, c2);
      unsigned int n0 = hi1 ^ c1 ^ key0, n1 = lo1;
      unsigned int n2 = hi0 ^ c3 ^ key1, n3 = lo0;
      c0 = n0; c1 = n1; c2 = n2; c3 = n3;
      key0 += 0x9E3779B9u; key1 += 0xBB67AE85u;
    }
    return make_uint4(c0, c1, c2, c3);
  }
};

DEV float u32_to_uniform(unsigned int x) {  // (0, 1]
  return (x >> 8) * (1.0f / 16777216.0f) + (1.0f / 16777216.0f);
}

#define HIP_CHECK_LAST()                                                     \
  do {                                                                       \
    hipError_t e = hipGetLastError();                                        \
    if (e != hipSuccess)                                                     \
      TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e)); \
  } while (0)

DEV int ceil_div_d(int a, int b) { return (a + b - 1) / b; }
inline int ceil_div(int a, int b) { return (a + b - 1) / b; }
