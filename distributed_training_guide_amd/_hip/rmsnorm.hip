// Fused RMSNorm forward/backward for gfx950.
//
// Replaces the reference's LlamaRMSNorm (transformers; invoked from every
// Llama forward — see SURVEY.md §2b "RMSNorm"). One workgroup per row,
// bf16x8 vectorized loads (G13), f32 accumulation, wave+block shuffle
// reductions. Forward saves rstd (f32 per row) for the backward.
//
//   y = x * rsqrt(mean(x^2) + eps) * w
//   dx = rstd * (g - xhat * mean(g * xhat)),  g = dy * w,  xhat = x * rstd
//   dw = sum_rows dy * xhat   (two-stage: per-block partials, then reduce)
#include "common.h"
#include <stdlib.h>

// ---------------- forward ----------------
__global__ void __launch_bounds__(256) rmsnorm_fwd_kernel(
    const short* __restrict__ x, const short* __restrict__ w,
    short* __restrict__ y, float* __restrict__ rstd_out,
    int64_t nrows, int H, float eps) {
  __shared__ float scratch[8];
  const float invH = 1.0f / (float)H;
  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * H;
    short* yr = y + row * H;
    float ss = 0.0f;
    if ((H & 7) == 0) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) { float f = bf2f(v[j]); ss += f * f; }
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        float f = bf2f(xr[i]); ss += f * f;
      }
    }
    ss = block4_sum(ss, scratch);
    const float rstd = rsqrtf(ss * invH + eps);
    if (threadIdx.x == 0 && rstd_out != nullptr) rstd_out[row] = rstd;
    if ((H & 7) == 0) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(v[j]) * rstd * bf2f(wv[j]));
        *reinterpret_cast<s16x8*>(yr + i) = o;
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x)
        yr[i] = f2bf(bf2f(xr[i]) * rstd * bf2f(w[i]));
    }
    __syncthreads();
  }
}

// ---------------- backward: dx + per-block dw partials ----------------
// dw partials accumulate in REGISTERS across this block's rows (each
// thread owns fixed columns) and spill to dw_partial ONCE at the end;
// dy/x rows are staged in dynamic LDS on pass 1 so pass 2 never re-reads
// global. Traffic/row: read dy+x, write dx (the memory-bound floor).
__global__ void __launch_bounds__(256) rmsnorm_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ rstd,
    short* __restrict__ dx, float* __restrict__ dw_partial,
    int64_t nrows, int H) {
  __shared__ float scratch[8];
  extern __shared__ short rowbuf[];  // [H] dy then [H] x
  short* dy_l = rowbuf;
  short* x_l = rowbuf + H;
  const float invH = 1.0f / (float)H;
  const bool vec = (H & 7) == 0 && H <= 2048 * 8;
  // per-thread dw accumulators: chunk c covers column i = c*2048 + tid*8
  const int nch = (H + 2047) / 2048;
  float dwacc[8][8];
#pragma unroll
  for (int c = 0; c < 8; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) dwacc[c][j] = 0.0f;

  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* dyr = dy + row * H;
    const short* xr = x + row * H;
    short* dxr = dx + row * H;
    const float rs = rstd[row];
    float dot = 0.0f;
    if (vec) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
        s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        *reinterpret_cast<s16x8*>(dy_l + i) = dv;
        *reinterpret_cast<s16x8*>(x_l + i) = xv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          dot += bf2f(dv[j]) * bf2f(wv[j]) * bf2f(xv[j]) * rs;
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        dy_l[i] = dyr[i];
        x_l[i] = xr[i];
        dot += bf2f(dyr[i]) * bf2f(w[i]) * bf2f(xr[i]) * rs;
      }
    }
    dot = block4_sum(dot, scratch) * invH;
    if (vec) {
      int c = 0;
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8, ++c) {
        s16x8 dv = *reinterpret_cast<const s16x8*>(dy_l + i);
        s16x8 xv = *reinterpret_cast<const s16x8*>(x_l + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        s16x8 o;
        float* acc = dwacc[c];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = bf2f(dv[j]) * bf2f(wv[j]);
          float xhat = bf2f(xv[j]) * rs;
          o[j] = f2bf(rs * (g - xhat * dot));
          acc[j] += bf2f(dv[j]) * xhat;
        }
        *reinterpret_cast<s16x8*>(dxr + i) = o;
      }
    } else {
      // scalar fallback: accumulate straight to the partial row (rare)
      float* dwp = dw_partial + (int64_t)blockIdx.x * H;
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        float g = bf2f(dy_l[i]) * bf2f(w[i]);
        float xhat = bf2f(x_l[i]) * rs;
        dxr[i] = f2bf(rs * (g - xhat * dot));
        dwp[i] += bf2f(dy_l[i]) * xhat;
      }
    }
    __syncthreads();
  }
  // spill register accumulators once
  float* dwp = dw_partial + (int64_t)blockIdx.x * H;
  if (vec) {
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      if (c >= nch) break;
      int i = c * 2048 + threadIdx.x * 8;
      if (i < H) {
#pragma unroll
        for (int j = 0; j < 8; ++j) dwp[i + j] = dwacc[c][j];
      }
    }
  }
}

// ---------------- backward, wave-per-row (H % 8 == 0, H <= 4096) -----
// One 64-lane wave owns a whole row: dy/x staged in REGISTERS between the
// dot pass and the dx pass (no LDS round trip), w cached in registers
// across rows, the row reduction is 6 wave shuffles instead of a
// block_reduce's LDS+barrier ladder.  The block's 4 waves are fully
// independent — zero __syncthreads in the loop.  dw register partials
// spill once per WAVE (dw_partial row = blockIdx*4 + wave).
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

__global__ void __launch_bounds__(256) rmsnorm_bwd_wave_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ rstd,
    short* __restrict__ dx, float* __restrict__ dw_partial,
    int64_t nrows, int H) {
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int nch = H / (WAVE * 8);  // <= 8
  s16x8 wv[8];
#pragma unroll
  for (int c = 0; c < 8; ++c)
    if (c < nch)
      wv[c] = *reinterpret_cast<const s16x8*>(w + c * WAVE * 8 + lane * 8);
  float dwacc[8][8];
#pragma unroll
  for (int c = 0; c < 8; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) dwacc[c][j] = 0.0f;

  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < nrows;
       row += (int64_t)gridDim.x * 4) {
    const short* dyr = dy + row * H;
    const short* xr = x + row * H;
    short* dxr = dx + row * H;
    const float rs = rstd[row];
    s16x8 dv[8], xv[8];
    float dot = 0.0f;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      if (c >= nch) break;
      const int i = c * WAVE * 8 + lane * 8;
      dv[c] = *reinterpret_cast<const s16x8*>(dyr + i);
      xv[c] = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(dv[c][j]) * bf2f(wv[c][j]) * bf2f(xv[c][j]);
    }
    dot = wave_sum(dot) * rs / (float)H;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      if (c >= nch) break;
      const int i = c * WAVE * 8 + lane * 8;
      s16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf2f(dv[c][j]) * bf2f(wv[c][j]);
        float xhat = bf2f(xv[c][j]) * rs;
        o[j] = f2bf(rs * (g - xhat * dot));
        dwacc[c][j] += bf2f(dv[c][j]) * xhat;
      }
      *reinterpret_cast<s16x8*>(dxr + i) = o;
    }
  }
  float* dwp = dw_partial + ((int64_t)blockIdx.x * 4 + wave) * H;
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    if (c >= nch) break;
    const int i = c * WAVE * 8 + lane * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) dwp[i + j] = dwacc[c][j];
  }
}

// reduce dw partials: [nblocks, H] f32 -> dw [H] bf16.  Two stages so the
// read of the (up to 2048 x H) partial buffer parallelizes over enough
// blocks to reach memory bandwidth (a single H/256-block pass is
// CU-bound); stage 1 folds rows into a [16][H] buffer, stage 2 finishes.
#define DW_RY 16
__global__ void __launch_bounds__(256) rmsnorm_dw_stage1_kernel(
    const float* __restrict__ dw_partial, float* __restrict__ partial2,
    int nblocks, int H) {
  const int y = blockIdx.y;
  const int r0 = (int)(((int64_t)nblocks * y) / DW_RY);
  const int r1 = (int)(((int64_t)nblocks * (y + 1)) / DW_RY);
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < H;
       i += gridDim.x * blockDim.x) {
    float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int b = r0;
    for (; b + 8 <= r1; b += 8)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s[j] += dw_partial[(int64_t)(b + j) * H + i];
    for (; b < r1; ++b) s[0] += dw_partial[(int64_t)b * H + i];
    partial2[(int64_t)y * H + i] = ((s[0] + s[1]) + (s[2] + s[3])) +
                                   ((s[4] + s[5]) + (s[6] + s[7]));
  }
}

__global__ void __launch_bounds__(256) rmsnorm_dw_stage2_kernel(
    const float* __restrict__ partial2, short* __restrict__ dw, int H) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < H;
       i += gridDim.x * blockDim.x) {
    float s = 0.f;
#pragma unroll
    for (int y = 0; y < DW_RY; ++y) s += partial2[(int64_t)y * H + i];
    dw[i] = f2bf(s);
  }
}

extern "C" {
void rmsnorm_dw_reduce_launch(const float*, void*, int, int,
                              hipStream_t);
void rmsnorm_fwd_launch(const void* x, const void* w, void* y, void* rstd,
                        int64_t nrows, int H, float eps, hipStream_t s) {
  int grid = (int)(nrows < 2048 ? nrows : 2048);
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)x, (const short*)w, (short*)y, (float*)rstd,
                     nrows, H, eps);
}
void rmsnorm_bwd_launch(const void* dy, const void* x, const void* w,
                        const void* rstd, void* dx, float* dw_partial,
                        void* dw, int nblocks, int64_t nrows, int H,
                        hipStream_t s) {
  static int wave_ok = -1;
  if (wave_ok < 0) {  // measured: block-per-row wins at bs24/H=4096 (0.281
    // vs 0.319 ms — the wave variant's ~200 VGPRs cap occupancy at
    // 2 waves/SIMD); DTGA_NORM_WAVE=1 selects the wave path for re-eval
    const char* e = getenv("DTGA_NORM_WAVE");
    wave_ok = (e && e[0] == '1');
  }
  if (wave_ok && (H & (64 * 8 - 1)) == 0 && H <= 4096 && nblocks >= 4) {
    // wave-per-row fast path: 4 independent waves per block, partial
    // rows = 4 * grid (fits the caller's [nblocks, H] allocation)
    int grid = nblocks / 4;
    int64_t need = (nrows + 3) / 4;
    if (need < grid) grid = (int)(need < 1 ? 1 : need);
    hipLaunchKernelGGL(rmsnorm_bwd_wave_kernel, dim3(grid), dim3(256), 0, s,
                       (const short*)dy, (const short*)x, (const short*)w,
                       (const float*)rstd, (short*)dx, dw_partial, nrows, H);
    rmsnorm_dw_reduce_launch(dw_partial, dw, grid * 4, H, s);
    return;
  }
  // scalar fallback (H not a multiple of 8) accumulates into dw_partial
  // directly, so it must start zeroed; the vectorized path overwrites.
  if ((H & 7) || H > 2048 * 8)
    hipMemsetAsync(dw_partial, 0, (size_t)nblocks * H * sizeof(float), s);
  size_t shmem = 2 * (size_t)H * sizeof(short);
  hipLaunchKernelGGL(rmsnorm_bwd_kernel, dim3(nblocks), dim3(256), shmem, s,
                     (const short*)dy, (const short*)x, (const short*)w,
                     (const float*)rstd, (short*)dx, dw_partial, nrows, H);
  rmsnorm_dw_reduce_launch(dw_partial, dw, nblocks, H, s);
}
void rmsnorm_dw_reduce_launch(const float* dw_partial, void* dw, int nblocks,
                              int H, hipStream_t s) {
  // scratch for stage 1 lives at the END of dw_partial's allocation?  No —
  // callers allocate exactly [nblocks, H]; reuse its first DW_RY rows is
  // unsafe (still being read).  Grab a cached block via hipMallocAsync.
  float* p2 = nullptr;
  (void)hipMallocAsync((void**)&p2, (size_t)DW_RY * H * sizeof(float), s);
  int rgrid = (H + 255) / 256;
  hipLaunchKernelGGL(rmsnorm_dw_stage1_kernel, dim3(rgrid, DW_RY), dim3(256),
                     0, s, dw_partial, p2, nblocks, H);
  hipLaunchKernelGGL(rmsnorm_dw_stage2_kernel, dim3(rgrid), dim3(256), 0, s,
                     p2, (short*)dw, H);
  (void)hipFreeAsync(p2, s);
}
}
