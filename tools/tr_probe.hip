// Standalone probe for gfx950 ds_read_b64_tr_b16 addressing semantics.
//   hipcc --offload-arch=gfx950 -O2 tools/tr_probe.hip -o /tmp/tr_probe && /tmp/tr_probe
// LDS holds a [8 rows][16 cols] bf16 image with value = row*100 + col
// (as bf16-exact small ints).  Each variant assigns per-lane addresses
// and prints what every lane's 4 result elements contain.

#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

__global__ void probe(float* out, int variant) {
  __shared__ __bf16 lds[1024];
  const int l = threadIdx.x;
  // fill [8][16]: lds[r*16+c] = r*100 + c
  if (l < 128) lds[l] = (__bf16)(float)((l / 16) * 100 + (l % 16));
  __syncthreads();
  int elem_off;
  if (variant == 0)       // uniform: every lane addr = lds[0]
    elem_off = 0;
  else if (variant == 1)  // the guide's pattern
    elem_off = (l & 15) + (l >> 4) * 64;
  else                    // 8-aligned per-lane: lane reads own 4-elem row?
    elem_off = (l & 15) * 4;
  bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4*)&lds[elem_off]);
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = (float)v[j];
}

int main() {
  float* d;
  hipMalloc(&d, 64 * 4 * sizeof(float));
  float h[256];
  for (int variant = 0; variant < 3; ++variant) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, variant);
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("== variant %d (lane: e0 e1 e2 e3)\n", variant);
    for (int l = 0; l < 20; ++l)
      printf("  l%02d: %5.0f %5.0f %5.0f %5.0f\n", l,
             h[l * 4], h[l * 4 + 1], h[l * 4 + 2], h[l * 4 + 3]);
    printf("  l16: %5.0f %5.0f %5.0f %5.0f\n",
           h[16 * 4], h[16 * 4 + 1], h[16 * 4 + 2], h[16 * 4 + 3]);
  }
  return 0;
}
