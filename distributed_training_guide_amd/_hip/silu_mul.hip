// Fused SwiGLU gate: y = silu(g) * u, with g/u packed as gu = [..., 2*I]
// (g = gu[..., :I], u = gu[..., I:]) — the single fused epilogue after the
// gate_up GEMM of a Llama MLP (SURVEY.md §2b "SiLU-gated MLP").
//
// Packing gate+up into ONE GEMM output halves the GEMM launch count and
// lets this kernel read both halves of each row in one pass.  Wave-per-row
// iteration: row/col come from the wave id and lane (the flat-index
// variant paid a 64-bit integer divide — a libcall on GCN — per 8
// elements).  Memory-bound: bf16x8 vector loads (G13), f32 math.
//
//   silu(x) = x * sigmoid(x);  d/dx silu = sigmoid(x) * (1 + x * (1 - sigmoid(x)))
#include "common.h"

__global__ void __launch_bounds__(256) silu_mul_fwd_kernel(
    const short* __restrict__ gu, short* __restrict__ y, int64_t nrows,
    int I) {
  const int gwave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int nwaves = gridDim.x * blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  for (int64_t row = gwave; row < nrows; row += nwaves) {
    const short* g = gu + row * (2 * (int64_t)I);
    const short* u = g + I;
    short* yr = y + row * (int64_t)I;
    for (int i = lane * 8; i < I; i += WAVE * 8) {
      s16x8 gv = *reinterpret_cast<const s16x8*>(g + i);
      s16x8 uv = *reinterpret_cast<const s16x8*>(u + i);
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = bf2f(gv[j]);
        float sig = 1.0f / (1.0f + __expf(-x));
        o[j] = f2bf(x * sig * bf2f(uv[j]));
      }
      *reinterpret_cast<s16x8*>(yr + i) = o;
    }
  }
}

__global__ void __launch_bounds__(256) silu_mul_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ gu,
    short* __restrict__ dgu, int64_t nrows, int I) {
  const int gwave = (blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const int nwaves = gridDim.x * blockDim.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  for (int64_t row = gwave; row < nrows; row += nwaves) {
    const short* dyr = dy + row * (int64_t)I;
    const short* g = gu + row * (2 * (int64_t)I);
    const short* u = g + I;
    short* dg = dgu + row * (2 * (int64_t)I);
    short* du = dg + I;
    for (int i = lane * 8; i < I; i += WAVE * 8) {
      s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
      s16x8 gv = *reinterpret_cast<const s16x8*>(g + i);
      s16x8 uv = *reinterpret_cast<const s16x8*>(u + i);
      s16x8 og, ou;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float x = bf2f(gv[j]);
        float sig = 1.0f / (1.0f + __expf(-x));
        float sx = x * sig;
        float d = bf2f(dv[j]);
        ou[j] = f2bf(d * sx);
        og[j] = f2bf(d * bf2f(uv[j]) * sig * (1.0f + x * (1.0f - sig)));
      }
      *reinterpret_cast<s16x8*>(dg + i) = og;
      *reinterpret_cast<s16x8*>(du + i) = ou;
    }
  }
}

extern "C" {
void silu_mul_fwd_launch(const void* gu, void* y, int64_t nrows, int I,
                         hipStream_t s) {
  int64_t want = (nrows + 3) / 4;  // 4 waves per block
  int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  hipLaunchKernelGGL(silu_mul_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)gu, (short*)y, nrows, I);
}
void silu_mul_bwd_launch(const void* dy, const void* gu, void* dgu,
                         int64_t nrows, int I, hipStream_t s) {
  int64_t want = (nrows + 3) / 4;
  int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  hipLaunchKernelGGL(silu_mul_bwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)dy, (const short*)gu, (short*)dgu, nrows,
                     I);
}
}
