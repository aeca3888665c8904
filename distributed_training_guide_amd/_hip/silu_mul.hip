// Fused SwiGLU gate: y = silu(g) * u, with g/u packed as gu = [..., 2*I]
// (g = gu[..., :I], u = gu[..., I:]) — the single fused epilogue after the
// gate_up GEMM of a Llama MLP (SURVEY.md §2b "SiLU-gated MLP").
//
// Packing gate+up into ONE GEMM output halves the GEMM launch count and lets
// this kernel read both halves of each row in one pass. Memory-bound:
// bf16x8 vector loads (G13), grid-stride, f32 math.
//
//   silu(x) = x * sigmoid(x);  d/dx silu = sigmoid(x) * (1 + x * (1 - sigmoid(x)))
#include "common.h"

__global__ void __launch_bounds__(256) silu_mul_fwd_kernel(
    const short* __restrict__ gu, short* __restrict__ y, int64_t nrows, int I) {
  for (int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       idx < nrows * (int64_t)I; idx += (int64_t)gridDim.x * blockDim.x * 8) {
    int64_t row = idx / I;
    int col = (int)(idx % I);
    const short* g = gu + row * (2 * (int64_t)I) + col;
    const short* u = g + I;
    s16x8 gv = *reinterpret_cast<const s16x8*>(g);
    s16x8 uv = *reinterpret_cast<const s16x8*>(u);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = bf2f(gv[j]);
      float sig = 1.0f / (1.0f + __expf(-x));
      o[j] = f2bf(x * sig * bf2f(uv[j]));
    }
    *reinterpret_cast<s16x8*>(y + idx) = o;
  }
}

__global__ void __launch_bounds__(256) silu_mul_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ gu,
    short* __restrict__ dgu, int64_t nrows, int I) {
  for (int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       idx < nrows * (int64_t)I; idx += (int64_t)gridDim.x * blockDim.x * 8) {
    int64_t row = idx / I;
    int col = (int)(idx % I);
    int64_t base = row * (2 * (int64_t)I) + col;
    s16x8 gv = *reinterpret_cast<const s16x8*>(gu + base);
    s16x8 uv = *reinterpret_cast<const s16x8*>(gu + base + I);
    s16x8 dv = *reinterpret_cast<const s16x8*>(dy + idx);
    s16x8 dg, du;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = bf2f(gv[j]);
      float u = bf2f(uv[j]);
      float d = bf2f(dv[j]);
      float sig = 1.0f / (1.0f + __expf(-x));
      float silu = x * sig;
      dg[j] = f2bf(d * u * sig * (1.0f + x * (1.0f - sig)));
      du[j] = f2bf(d * silu);
    }
    *reinterpret_cast<s16x8*>(dgu + base) = dg;
    *reinterpret_cast<s16x8*>(dgu + base + I) = du;
  }
}

extern "C" {
void silu_mul_fwd_launch(const void* gu, void* y, int64_t nrows, int I,
                         hipStream_t s) {
  int64_t want = (nrows * (int64_t)I + 8 * 256 - 1) / (8 * 256);
  int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  hipLaunchKernelGGL(silu_mul_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)gu, (short*)y, nrows, I);
}
void silu_mul_bwd_launch(const void* dy, const void* gu, void* dgu,
                         int64_t nrows, int I, hipStream_t s) {
  int64_t want = (nrows * (int64_t)I + 8 * 256 - 1) / (8 * 256);
  int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  hipLaunchKernelGGL(silu_mul_bwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)dy, (const short*)gu, (short*)dgu, nrows, I);
}
}
