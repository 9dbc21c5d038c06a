// Probe ds_read_b64_tr_b16 semantics: fill LDS with linear u16 pattern,
// read with tr at per-lane addresses, dump what each lane receives.
#include <hip/hip_runtime.h>
#include <cstdio>
using s16x4 = __attribute__((ext_vector_type(4))) short;
typedef __attribute__((address_space(3))) s16x4* lds_v4p;

__global__ void k_probe(short* out, int mode) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 64) lds[i] = (short)i;
  __syncthreads();
  int l = threadIdx.x;
  // candidate addressing modes (element offsets)
  int off;
  switch (mode) {
    case 0: off = l * 4; break;                    // lane-linear 8B
    case 1: off = (l & 15) * 4 + (l >> 4) * 64; break;
    case 2: off = (l & 15) + (l >> 4) * 64; break; // 2B-stride (likely invalid)
    default: off = l * 4;
  }
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((lds_v4p)&lds[off]);
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
}

int main() {
  short* out;
  hipMalloc(&out, 64 * 4 * sizeof(short));
  short host[256];
  for (int mode = 0; mode < 2; ++mode) {
    hipLaunchKernelGGL(k_probe, dim3(1), dim3(64), 0, 0, out, mode);
    hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d:[%4d %4d %4d %4d] ", l, host[l*4], host[l*4+1], host[l*4+2], host[l*4+3]);
      if (l % 4 == 3) printf("\n");
    }
  }
  return 0;
}
