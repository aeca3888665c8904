// Fused LayerNorm fwd/bwd + tanh-GELU fwd/bwd for gfx950 — the GPT-2
// block ops (reference chapter-1 smoke model trains HF gpt2, whose
// LayerNorm/GELU come from transformers; SURVEY.md §2b "torch.compile
// fusions").  Same machinery as rmsnorm.hip: one workgroup per row,
// bf16x8 vector loads, f32 accumulation, register dw/db accumulators
// spilled once per block; the [nblocks,H] partials are reduced by the
// rmsnorm two-stage reducer.
//
//   y  = (x - mu) * rstd * w + b,   rstd = rsqrt(var + eps)
//   dx = rstd * (g - mean(g) - xhat * mean(g * xhat)),  g = dy * w
//   dw = sum_rows dy * xhat;  db = sum_rows dy
//
//   gelu(x) = 0.5 x (1 + tanh(k (x + 0.044715 x^3))), k = sqrt(2/pi)
#include "common.h"

#define GELU_K 0.7978845608028654f
#define GELU_C 0.044715f

// ---------------- layernorm forward ----------------
__global__ void __launch_bounds__(256) ln_fwd_kernel(
    const short* __restrict__ x, const short* __restrict__ w,
    const short* __restrict__ b, short* __restrict__ y,
    float* __restrict__ mu_out, float* __restrict__ rstd_out,
    int64_t nrows, int H, float eps) {
  __shared__ float scratch[8];
  __shared__ float mu_sh;
  const float invH = 1.0f / (float)H;
  const bool vec = (H & 7) == 0;
  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* xr = x + row * H;
    short* yr = y + row * H;
    float s = 0.0f, ss = 0.0f;
    if (vec) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf2f(v[j]);
          s += f;
          ss += f * f;
        }
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        float f = bf2f(xr[i]);
        s += f;
        ss += f * f;
      }
    }
    s = block4_sum(s, scratch);
    if (threadIdx.x == 0) mu_sh = s * invH;
    __syncthreads();
    const float mu = mu_sh;
    ss = block4_sum(ss, scratch);
    const float var = ss * invH - mu * mu;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (mu_out != nullptr) mu_out[row] = mu;
      if (rstd_out != nullptr) rstd_out[row] = rstd;
    }
    if (vec) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 v = *reinterpret_cast<const s16x8*>(xr + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        s16x8 bv = *reinterpret_cast<const s16x8*>(b + i);
        s16x8 o;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = f2bf((bf2f(v[j]) - mu) * rstd * bf2f(wv[j]) + bf2f(bv[j]));
        *reinterpret_cast<s16x8*>(yr + i) = o;
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x)
        yr[i] = f2bf((bf2f(xr[i]) - mu) * rstd * bf2f(w[i]) + bf2f(b[i]));
    }
    __syncthreads();
  }
}

// ---------------- layernorm backward ----------------
// register accumulators cover H <= 4*2048 (gpt2 768 .. gpt2-xl 1600);
// larger H takes the scalar fallback accumulating into the partials.
__global__ void __launch_bounds__(256) ln_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ mu,
    const float* __restrict__ rstd, short* __restrict__ dx,
    float* __restrict__ dw_partial, float* __restrict__ db_partial,
    int64_t nrows, int H) {
  __shared__ float scratch[8];
  __shared__ float sh_mg;
  extern __shared__ short rowbuf[];  // [H] dy then [H] x
  short* dy_l = rowbuf;
  short* x_l = rowbuf + H;
  const float invH = 1.0f / (float)H;
  const bool vec = (H & 7) == 0 && H <= 2048 * 4;
  const int nch = (H + 2047) / 2048;
  float dwacc[4][8], dbacc[4][8];
#pragma unroll
  for (int c = 0; c < 4; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) dwacc[c][j] = dbacc[c][j] = 0.0f;

  for (int64_t row = blockIdx.x; row < nrows; row += gridDim.x) {
    const short* dyr = dy + row * H;
    const short* xr = x + row * H;
    short* dxr = dx + row * H;
    const float m = mu[row];
    const float rs = rstd[row];
    float sg = 0.0f, sgx = 0.0f;
    if (vec) {
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8) {
        s16x8 dv = *reinterpret_cast<const s16x8*>(dyr + i);
        s16x8 xv = *reinterpret_cast<const s16x8*>(xr + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        *reinterpret_cast<s16x8*>(dy_l + i) = dv;
        *reinterpret_cast<s16x8*>(x_l + i) = xv;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float g = bf2f(dv[j]) * bf2f(wv[j]);
          float xhat = (bf2f(xv[j]) - m) * rs;
          sg += g;
          sgx += g * xhat;
        }
      }
    } else {
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        dy_l[i] = dyr[i];
        x_l[i] = xr[i];
        float g = bf2f(dyr[i]) * bf2f(w[i]);
        sg += g * 1.0f;
        sgx += g * (bf2f(xr[i]) - m) * rs;
      }
    }
    sg = block4_sum(sg, scratch);
    if (threadIdx.x == 0) sh_mg = sg * invH;
    __syncthreads();
    const float mg = sh_mg;
    sgx = block4_sum(sgx, scratch);
    const float mgx = sgx * invH;
    if (vec) {
      int c = 0;
      for (int i = threadIdx.x * 8; i < H; i += blockDim.x * 8, ++c) {
        s16x8 dv = *reinterpret_cast<const s16x8*>(dy_l + i);
        s16x8 xv = *reinterpret_cast<const s16x8*>(x_l + i);
        s16x8 wv = *reinterpret_cast<const s16x8*>(w + i);
        s16x8 o;
        float* wa = dwacc[c];
        float* ba = dbacc[c];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float d = bf2f(dv[j]);
          float g = d * bf2f(wv[j]);
          float xhat = (bf2f(xv[j]) - m) * rs;
          o[j] = f2bf(rs * (g - mg - xhat * mgx));
          wa[j] += d * xhat;
          ba[j] += d;
        }
        *reinterpret_cast<s16x8*>(dxr + i) = o;
      }
    } else {
      float* dwp = dw_partial + (int64_t)blockIdx.x * H;
      float* dbp = db_partial + (int64_t)blockIdx.x * H;
      for (int i = threadIdx.x; i < H; i += blockDim.x) {
        float d = bf2f(dy_l[i]);
        float g = d * bf2f(w[i]);
        float xhat = (bf2f(x_l[i]) - m) * rs;
        dxr[i] = f2bf(rs * (g - mg - xhat * mgx));
        dwp[i] += d * xhat;
        dbp[i] += d;
      }
    }
    __syncthreads();
  }
  if (vec) {
    float* dwp = dw_partial + (int64_t)blockIdx.x * H;
    float* dbp = db_partial + (int64_t)blockIdx.x * H;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c >= nch) break;
      int i = c * 2048 + threadIdx.x * 8;
      if (i < H) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dwp[i + j] = dwacc[c][j];
          dbp[i + j] = dbacc[c][j];
        }
      }
    }
  }
}

// ---------------- tanh-GELU ----------------
__global__ void __launch_bounds__(256) gelu_fwd_kernel(
    const short* __restrict__ x, short* __restrict__ y, int64_t n) {
  for (int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       idx < n; idx += (int64_t)gridDim.x * blockDim.x * 8) {
    s16x8 v = *reinterpret_cast<const s16x8*>(x + idx);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      float t = tanhf(GELU_K * (f + GELU_C * f * f * f));
      o[j] = f2bf(0.5f * f * (1.0f + t));
    }
    *reinterpret_cast<s16x8*>(y + idx) = o;
  }
}

__global__ void __launch_bounds__(256) gelu_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    short* __restrict__ dx, int64_t n) {
  for (int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       idx < n; idx += (int64_t)gridDim.x * blockDim.x * 8) {
    s16x8 dv = *reinterpret_cast<const s16x8*>(dy + idx);
    s16x8 v = *reinterpret_cast<const s16x8*>(x + idx);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      float t = tanhf(GELU_K * (f + GELU_C * f * f * f));
      float du = GELU_K * (1.0f + 3.0f * GELU_C * f * f);
      float d = 0.5f * (1.0f + t) + 0.5f * f * (1.0f - t * t) * du;
      o[j] = f2bf(bf2f(dv[j]) * d);
    }
    *reinterpret_cast<s16x8*>(dx + idx) = o;
  }
}

extern "C" {
void rmsnorm_dw_reduce_launch(const float*, void*, int, int, hipStream_t);

void ln_fwd_launch(const void* x, const void* w, const void* b, void* y,
                   float* mu, float* rstd, int64_t nrows, int H, float eps,
                   hipStream_t s) {
  int grid = (int)(nrows < 2048 ? (nrows < 1 ? 1 : nrows) : 2048);
  hipLaunchKernelGGL(ln_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)x, (const short*)w, (const short*)b,
                     (short*)y, mu, rstd, nrows, H, eps);
}

void ln_bwd_launch(const void* dy, const void* x, const void* w,
                   const float* mu, const float* rstd, void* dx,
                   float* dw_partial, float* db_partial, void* dw, void* db,
                   int nblocks, int64_t nrows, int H, hipStream_t s) {
  if ((H & 7) || H > 2048 * 4) {
    hipMemsetAsync(dw_partial, 0, (size_t)nblocks * H * sizeof(float), s);
    hipMemsetAsync(db_partial, 0, (size_t)nblocks * H * sizeof(float), s);
  }
  size_t shmem = 2 * (size_t)H * sizeof(short);
  hipLaunchKernelGGL(ln_bwd_kernel, dim3(nblocks), dim3(256), shmem, s,
                     (const short*)dy, (const short*)x, (const short*)w, mu,
                     rstd, (short*)dx, dw_partial, db_partial, nrows, H);
  rmsnorm_dw_reduce_launch(dw_partial, dw, nblocks, H, s);
  rmsnorm_dw_reduce_launch(db_partial, db, nblocks, H, s);
}

void gelu_fwd_launch(const void* x, void* y, int64_t n, hipStream_t s) {
  int64_t blocks = (n / 8 + 255) / 256;
  int grid = (int)(blocks < 4096 ? (blocks < 1 ? 1 : blocks) : 4096);
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)x, (short*)y, n);
}

void gelu_bwd_launch(const void* dy, const void* x, void* dx, int64_t n,
                     hipStream_t s) {
  int64_t blocks = (n / 8 + 255) / 256;
  int grid = (int)(blocks < 4096 ? (blocks < 1 ? 1 : blocks) : 4096);
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(grid), dim3(256), 0, s,
                     (const short*)dy, (const short*)x, (short*)dx, n);
}
}
