#include <hip/hip_runtime.h>
#include <stdio.h>
typedef __attribute__((ext_vector_type(4))) short s16x4t;

template <int MODE>
__global__ void probe_ds(short* out) {
  __shared__ short lds[4096];
  int tid = threadIdx.x;
  for (int i = tid; i < 4096; i += 64) lds[i] = (short)i;
  __syncthreads();
  int addr = (MODE == 0) ? tid * 4 : (MODE == 1 ? (tid & 15) * 4 + (tid >> 4) * 256 : (tid & 15) + (tid >> 4) * 64);
  s16x4t v = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) s16x4t*)(lds + addr));
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = v[j];
}

int main() {
  short* o; hipMalloc(&o, 64*4*2);
  short r[256];
  const char* names[3] = {"addr=lane*4", "addr=(l15)*4+(lg)*256", "addr=(l15)+(lg)*64"};
  for (int m = 0; m < 3; ++m) {
    if (m==0) hipLaunchKernelGGL((probe_ds<0>), dim3(1), dim3(64), 0, 0, o);
    if (m==1) hipLaunchKernelGGL((probe_ds<1>), dim3(1), dim3(64), 0, 0, o);
    if (m==2) hipLaunchKernelGGL((probe_ds<2>), dim3(1), dim3(64), 0, 0, o);
    hipMemcpy(r, o, sizeof(r), hipMemcpyDeviceToHost);
    printf("== ds_read_tr16_b64 %s ==\n", names[m]);
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d:", l);
      for (int j = 0; j < 4; ++j) printf(" %4d", r[l*4+j]);
      printf("\n");
    }
  }
  return 0;
}
