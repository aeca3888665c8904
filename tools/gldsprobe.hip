// Probe: gfx950 global_load_lds landing pattern (per-lane global source,
// wave-uniform LDS base; expect data at base + lane*size).
#include <hip/hip_runtime.h>
#include <stdio.h>
typedef __attribute__((ext_vector_type(4))) float f32x4p;
__global__ void k(const float* g, float* out) {
  __shared__ float lds[2048];
  int lane = threadIdx.x & 63;
  int wid = threadIdx.x >> 6;
  // wave w loads 64x16B from a PERMUTED global source into base w*256
  const float* src = g + ((lane * 7) % 64) * 4 + wid * 256;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)src,
      (__attribute__((address_space(3))) unsigned int*)(lds + wid * 256), 16,
      0, 0);
  __syncthreads();
  for (int i = threadIdx.x; i < 512; i += blockDim.x) out[i] = lds[i];
}
int main() {
  float *g, *o;
  hipMalloc(&g, 2048 * 4); hipMalloc(&o, 512 * 4);
  float h[2048]; for (int i = 0; i < 2048; ++i) h[i] = (float)i;
  hipMemcpy(g, h, sizeof(h), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(k, dim3(1), dim3(128), 0, 0, g, o);
  float r[512]; hipMemcpy(r, o, sizeof(r), hipMemcpyDeviceToHost);
  int ok = 1;
  for (int w = 0; w < 2; ++w)
    for (int l = 0; l < 64; ++l)
      for (int j = 0; j < 4; ++j) {
        float want = (float)(((l * 7) % 64) * 4 + w * 256 + j);
        float got = r[w * 256 + l * 4 + j];
        if (want != got) { if (ok) printf("MISMATCH w%d l%d j%d: got %.0f want %.0f\n", w, l, j, got, want); ok = 0; }
      }
  printf(ok ? "glds: lane-ordered landing CONFIRMED\n" : "glds: pattern differs\n");
  for (int i = 0; i < 8; ++i) printf("%0.f ", r[i]);
  printf("\n");
  return 0;
}
