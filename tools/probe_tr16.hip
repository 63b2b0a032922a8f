// Empirical semantics probe for ds_read_b64_tr_b16 on gfx950.
// Fills LDS with identity values (element index) and prints, for two
// addressing conventions, which elements each lane receives.
//   hipcc --offload-arch=gfx950 -O3 tools/probe_tr16.hip -o /tmp/probe_tr16
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) short s16x4;

__global__ void probe(short* outA, short* outB) {
  __shared__ short lds[256];
  int t = threadIdx.x;
  for (int i = t; i < 256; i += 64) lds[i] = (short)i;
  __syncthreads();
  // A: per-lane address = base + lane_in_group*8B (+ group*128B)
  {
    auto addr = (__attribute__((address_space(3))) bf16x4*)
        &lds[(t & 15) * 4 + (t >> 4) * 64];
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(addr);
    s16x4 sv = *(s16x4*)&v;
    for (int j = 0; j < 4; ++j) outA[t * 4 + j] = sv[j];
  }
  // B: per-lane address = base + (lane&15)*2B element offset (+ group*128B)
  {
    auto addr = (__attribute__((address_space(3))) bf16x4*)
        &lds[(t & 15) + (t >> 4) * 64];
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(addr);
    s16x4 sv = *(s16x4*)&v;
    for (int j = 0; j < 4; ++j) outB[t * 4 + j] = sv[j];
  }
}

int main() {
  short *dA, *dB;
  hipMalloc(&dA, 256 * sizeof(short));
  hipMalloc(&dB, 256 * sizeof(short));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, dA, dB);
  short hA[256], hB[256];
  hipMemcpy(hA, dA, sizeof(hA), hipMemcpyDeviceToHost);
  hipMemcpy(hB, dB, sizeof(hB), hipMemcpyDeviceToHost);
  printf("variant A (lane*8B):\n");
  for (int l = 0; l < 20; ++l)
    printf("lane %2d: %3d %3d %3d %3d\n", l, hA[l * 4], hA[l * 4 + 1],
           hA[l * 4 + 2], hA[l * 4 + 3]);
  printf("variant B ((lane&15)*2B):\n");
  for (int l = 0; l < 20; ++l)
    printf("lane %2d: %3d %3d %3d %3d\n", l, hB[l * 4], hB[l * 4 + 1],
           hB[l * 4 + 2], hB[l * 4 + 3]);
  return 0;
}
