// Fused multi-tensor AdamW for gfx950 (SURVEY.md §2b "Fused AdamW";
// reference uses torch.optim.AdamW(fused=True) at 01:73, 02:88, 04:113,
// 05:197, 06:151, 07:152).
//
// One launch updates every parameter chunk: the host packs chunk descriptors
// (pointers + length + dtypes) into an int64 device tensor, one 256-thread
// block per chunk, grid-stride within the chunk. Moments are fp32; params
// bf16 or fp32; grads bf16 or fp32 (fp32 under FSDP's reduce_dtype=fp32).
// Decoupled weight decay (AdamW): p *= (1 - lr*wd) before the Adam update.
#include "common.h"

#define ADAMW_CHUNK 262144  // elements per descriptor chunk

struct AdamWDesc {
  // int64 fields, laid out to match the host-side [n, 7] int64 tensor
  int64_t p, g, m, v;  // raw device pointers
  int64_t n;           // elements in this chunk
  int64_t p_is_bf16;
  int64_t g_is_bf16;
};

__global__ void __launch_bounds__(256) adamw_kernel(
    const AdamWDesc* __restrict__ descs, int nchunks, float lr, float beta1,
    float beta2, float eps, float wd, float inv_bc1, float inv_bc2) {
  for (int c = blockIdx.x; c < nchunks; c += gridDim.x) {
    AdamWDesc d = descs[c];
    float* m = (float*)d.m;
    float* v = (float*)d.v;
    const bool pbf = d.p_is_bf16 != 0;
    const bool gbf = d.g_is_bf16 != 0;
    const float decay = 1.0f - lr * wd;
    for (int64_t i = threadIdx.x; i < d.n; i += blockDim.x) {
      float g = gbf ? bf2f(((const short*)d.g)[i]) : ((const float*)d.g)[i];
      float p = pbf ? bf2f(((const short*)d.p)[i]) : ((const float*)d.p)[i];
      float mi = m[i] = beta1 * m[i] + (1.0f - beta1) * g;
      float vi = v[i] = beta2 * v[i] + (1.0f - beta2) * g * g;
      float mhat = mi * inv_bc1;
      float vhat = vi * inv_bc2;
      p = p * decay - lr * mhat / (sqrtf(vhat) + eps);
      if (pbf) ((short*)d.p)[i] = f2bf(p);
      else ((float*)d.p)[i] = p;
    }
  }
}

extern "C" {
void adamw_launch(const void* descs, int nchunks, float lr, float beta1,
                  float beta2, float eps, float wd, float inv_bc1,
                  float inv_bc2, hipStream_t s) {
  int grid = nchunks < 16384 ? nchunks : 16384;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(adamw_kernel, dim3(grid), dim3(256), 0, s,
                     (const AdamWDesc*)descs, nchunks, lr, beta1, beta2, eps,
                     wd, inv_bc1, inv_bc2);
}
}
