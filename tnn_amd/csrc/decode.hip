// Fused single-token decode attention: one new query row against the KV
// cache per (batch, head) — the M=1 GEMV analog of attention. Replaces
// the eager matmul → mask → softmax → matmul chain (and the reference's
// full-sequence recompute, examples/gpt2_inference.cpp:71-122) with two
// tiny kernels: split-KV partials with online softmax, then a combine.
//
// Layout: q [BH, D] bf16; K/V caches [BH, cap, D] bf16 (full fixed-size
// buffers). The live length comes either from a device int64 position
// pointer (len = pos+1; hipGraph-capturable — the grid stays fixed while
// the in-kernel range shrinks) or from a host int.
//
// Partial kernel: grid (BH, SPLITS), 256 threads. Per 256-wide KV tile:
// each thread computes one dot(q, K[i]) (q cached in LDS, K rows read as
// bf16x8 vectors), block-wide online max/sum, P staged through LDS, then
// threads re-map to (d = tid % D, sub = tid / D) and accumulate
// o[d] += p[j] * V[j][d] with consecutive-d coalesced V reads. fp32
// partial o and (m, l) go to a workspace; the combine kernel merges
// splits with the standard flash rescale.

#include "common.h"
#include "kernels.h"

namespace tnn {

template <int D>
__launch_bounds__(256)
__global__ void k_attn_decode(const bf16* __restrict__ q,
                              const bf16* __restrict__ K,
                              const bf16* __restrict__ V,
                              float* __restrict__ po, float* __restrict__ ml,
                              const int64_t* __restrict__ pos_ptr,
                              int len_static, int64_t skv, float scale) {
  constexpr int NSUB = 256 / D;   // sub-accumulators per d column
  constexpr int PER = 256 / NSUB; // tile positions per sub
  const int bh = blockIdx.x;
  const int split = blockIdx.y, NS = gridDim.y;
  const int len = pos_ptr ? (int)pos_ptr[0] + 1 : len_static;
  const int lo = (int)((int64_t)len * split / NS);
  const int hi = (int)((int64_t)len * (split + 1) / NS);

  const bf16* qr = q + (int64_t)bh * D;
  const bf16* Kb = K + (int64_t)bh * skv;
  const bf16* Vb = V + (int64_t)bh * skv;

  __shared__ float q_lds[D];
  __shared__ float p_lds[256];
  __shared__ float scratch[8];
  __shared__ float bc;  // broadcast slot

  for (int d = threadIdx.x; d < D; d += 256) q_lds[d] = bf2f(qr[d]);
  __syncthreads();

  const int d = threadIdx.x % D;
  const int sub = threadIdx.x / D;
  float m_run = -INFINITY, l_run = 0.0f, o_acc = 0.0f;

  for (int kv0 = lo; kv0 < hi; kv0 += 256) {
    const int i = kv0 + (int)threadIdx.x;
    float sv = -INFINITY;
    if (i < hi) {
      const bf16* krow = &Kb[(int64_t)i * D];
      float acc = 0.0f;
#pragma unroll
      for (int t = 0; t < D / 8; ++t) {
        bf16x8 kvv = *(const bf16x8*)&krow[t * 8];
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += q_lds[t * 8 + j] * (float)kvv[j];
      }
      sv = acc * scale;
    }
    float tmax = block_reduce_max(sv, scratch);
    if (threadIdx.x == 0) bc = tmax;
    __syncthreads();
    const float m_new = fmaxf(m_run, bc);
    const float p = (sv == -INFINITY) ? 0.0f : __expf(sv - m_new);
    p_lds[threadIdx.x] = p;
    float tsum = block_reduce_sum(p, scratch);
    if (threadIdx.x == 0) bc = tsum;
    __syncthreads();
    const float alpha = (m_run == -INFINITY) ? 0.0f : __expf(m_run - m_new);
    l_run = l_run * alpha + bc;
    o_acc *= alpha;
#pragma unroll 4
    for (int j = 0; j < PER; ++j) {
      const int t = sub * PER + j;
      if (kv0 + t < hi)
        o_acc += p_lds[t] * bf2f(Vb[(int64_t)(kv0 + t) * D + d]);
    }
    m_run = m_new;
  }

  // fold the NSUB per-d partials (all share m_run/l_run scaling)
  __syncthreads();
  p_lds[threadIdx.x] = o_acc;
  __syncthreads();
  if (sub == 0) {
    float o = 0.0f;
#pragma unroll
    for (int ss = 0; ss < NSUB; ++ss) o += p_lds[ss * D + d];
    po[((int64_t)bh * gridDim.y + split) * D + d] = o;
    if (d == 0) {
      ml[((int64_t)bh * gridDim.y + split) * 2] = m_run;
      ml[((int64_t)bh * gridDim.y + split) * 2 + 1] = l_run;
    }
  }
}

template <int D>
__launch_bounds__(128)
__global__ void k_attn_decode_fin(const float* __restrict__ po,
                                  const float* __restrict__ ml,
                                  bf16* __restrict__ out, int splits) {
  const int bh = blockIdx.x;
  const int d = threadIdx.x;
  if (d >= D) return;
  float m = -INFINITY;
#pragma unroll 4
  for (int s = 0; s < splits; ++s)
    m = fmaxf(m, ml[((int64_t)bh * splits + s) * 2]);
  float l = 0.0f, o = 0.0f;
#pragma unroll 4
  for (int s = 0; s < splits; ++s) {
    const float ms = ml[((int64_t)bh * splits + s) * 2];
    const float ls = ml[((int64_t)bh * splits + s) * 2 + 1];
    if (ls > 0.0f) {
      const float w = __expf(ms - m);
      l += ls * w;
      o += po[((int64_t)bh * splits + s) * D + d] * w;
    }
  }
  out[(int64_t)bh * D + d] = f2bf(l > 0.0f ? o / l : 0.0f);
}

void attn_decode_launch(const void* q, const void* k, const void* v, float* po,
                        float* ml, void* out, const int64_t* pos_ptr,
                        int len_static, int BH, int cap, int D, int splits,
                        float scale, hipStream_t s) {
  dim3 pg(BH, splits);
  const int64_t skv = (int64_t)cap * D;
#define L(DD)                                                                 \
  do {                                                                        \
    hipLaunchKernelGGL(k_attn_decode<DD>, pg, dim3(256), 0, s,                \
                       (const bf16*)q, (const bf16*)k, (const bf16*)v, po,    \
                       ml, pos_ptr, len_static, skv, scale);                  \
    hipLaunchKernelGGL(k_attn_decode_fin<DD>, dim3(BH), dim3(128), 0, s, po,  \
                       ml, (bf16*)out, splits);                               \
  } while (0)
  if (D == 64) L(64);
  else if (D == 128) L(128);
#undef L
}

}  // namespace tnn
