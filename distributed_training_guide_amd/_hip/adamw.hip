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

typedef __attribute__((ext_vector_type(4))) short s16x4v;

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
    // vectorized main body: 8 elements per thread per iteration (2x 16 B
    // f32 moment streams in flight — memory-bound, G13 vectorization +
    // deeper MLP; update via rsqrt: p -= lr*m̂ * rsqrt-style reciprocal)
    const int64_t n8 = d.n / 8;
    for (int64_t qq = threadIdx.x; qq < n8; qq += blockDim.x) {
      int64_t i = qq * 8;
      f32x4 g4[2], p4[2], m4[2], v4[2];
      if (gbf) {
        s16x8 gv = *reinterpret_cast<const s16x8*>((const short*)d.g + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) g4[j >> 2][j & 3] = bf2f(gv[j]);
      } else {
        g4[0] = *reinterpret_cast<const f32x4*>((const float*)d.g + i);
        g4[1] = *reinterpret_cast<const f32x4*>((const float*)d.g + i + 4);
      }
      if (pbf) {
        s16x8 pv = *reinterpret_cast<const s16x8*>((const short*)d.p + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) p4[j >> 2][j & 3] = bf2f(pv[j]);
      } else {
        p4[0] = *reinterpret_cast<const f32x4*>((const float*)d.p + i);
        p4[1] = *reinterpret_cast<const f32x4*>((const float*)d.p + i + 4);
      }
      m4[0] = *reinterpret_cast<const f32x4*>(m + i);
      m4[1] = *reinterpret_cast<const f32x4*>(m + i + 4);
      v4[0] = *reinterpret_cast<const f32x4*>(v + i);
      v4[1] = *reinterpret_cast<const f32x4*>(v + i + 4);
#pragma unroll
      for (int h = 0; h < 2; ++h)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          m4[h][j] = beta1 * m4[h][j] + (1.0f - beta1) * g4[h][j];
          v4[h][j] = beta2 * v4[h][j] + (1.0f - beta2) * g4[h][j] * g4[h][j];
          p4[h][j] = p4[h][j] * decay -
                     lr * (m4[h][j] * inv_bc1) /
                         (sqrtf(v4[h][j] * inv_bc2) + eps);
        }
      *reinterpret_cast<f32x4*>(m + i) = m4[0];
      *reinterpret_cast<f32x4*>(m + i + 4) = m4[1];
      *reinterpret_cast<f32x4*>(v + i) = v4[0];
      *reinterpret_cast<f32x4*>(v + i + 4) = v4[1];
      if (pbf) {
        s16x8 pv;
#pragma unroll
        for (int j = 0; j < 8; ++j) pv[j] = f2bf(p4[j >> 2][j & 3]);
        *reinterpret_cast<s16x8*>((short*)d.p + i) = pv;
      } else {
        *reinterpret_cast<f32x4*>((float*)d.p + i) = p4[0];
        *reinterpret_cast<f32x4*>((float*)d.p + i + 4) = p4[1];
      }
    }
    // scalar tail
    for (int64_t i = n8 * 8 + threadIdx.x; i < d.n; i += blockDim.x) {
      float g = gbf ? bf2f(((const short*)d.g)[i]) : ((const float*)d.g)[i];
      float p = pbf ? bf2f(((const short*)d.p)[i]) : ((const float*)d.p)[i];
      float mi = m[i] = beta1 * m[i] + (1.0f - beta1) * g;
      float vi = v[i] = beta2 * v[i] + (1.0f - beta2) * g * g;
      p = p * decay - lr * (mi * inv_bc1) / (sqrtf(vi * inv_bc2) + eps);
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
