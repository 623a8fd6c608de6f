// Standalone probe for gfx950 ds_read_b64_tr_b16 semantics.
// Fills LDS with element index i (fp16 exactly represents 0..2047),
// issues one tr read per lane at addr = base + lane*8B, and dumps the
// 4 delivered elements per lane.  Guide model (cdna_hip m162): lane l,
// elem j == tile[(l>>4)*4 + j][l&15] of the [16][16] row-major fp16
// tile at base.  Build: hipcc --offload-arch=gfx950 tr_probe.hip -o tr_probe
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __fp16 f4 __attribute__((vector_size(8)));

__global__ void probe(float* out) {
  __shared__ __fp16 buf[256];  // one 16x16 tile
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    buf[i] = (__fp16)(float)i;
  __syncthreads();
  if (threadIdx.x < 64) {
    f4 v = __builtin_amdgcn_ds_read_tr16_b64_v4f16(
        (__attribute__((address_space(3))) f4*)(uintptr_t)
            &buf[threadIdx.x * 4]);
    for (int j = 0; j < 4; ++j) out[threadIdx.x * 4 + j] = (float)v[j];
  }
}

int main() {
  float* out;
  (void)hipMalloc(&out, 256 * sizeof(float));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, out);
  (void)hipDeviceSynchronize();
  float h[256];
  (void)hipMemcpy(h, out, sizeof(h), hipMemcpyDeviceToHost);
  int mism = 0;
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d:", l);
    for (int j = 0; j < 4; ++j) printf(" %4.0f", h[l * 4 + j]);
    // guide model: elem j = row ((l>>4)*4 + j), col (l&15)
    int bad = 0;
    for (int j = 0; j < 4; ++j) {
      float expect = (float)(((l >> 4) * 4 + j) * 16 + (l & 15));
      if (h[l * 4 + j] != expect) bad = 1;
    }
    printf(bad ? "  <-- MISMATCH\n" : "\n");
    mism += bad;
  }
  printf(mism ? "MODEL MISMATCH in %d lanes\n" : "MODEL OK\n", mism);
  return 0;
}
