// Common device helpers for the MI355X (gfx950 / CDNA4) kernels.
//
// Everything here is CDNA4-native: 64-wide wavefronts, bf16 vector loads as
// short4/short8 reinterprets (hipcc does not auto-vectorize bf16 loads),
// f32 accumulation, wave shuffle reductions over width 64.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64

// Debug-build device assertions (SURVEY.md §5 "race detection /
// sanitizers": the compute-sanitizer analogue for this kernel set).
// Enable with DTGA_HIP_FLAGS="-DDTGA_DEBUG" and a rebuild (touch the
// sources or rm build/); zero cost in release builds.  A failed assert
// prints its site and traps the wavefront, which surfaces as a HIP
// error on the next synchronize instead of silent corruption.
#ifdef DTGA_DEBUG
#define DTGA_KERNEL_ASSERT(cond)                                         \
  do {                                                                   \
    if (!(cond)) {                                                       \
      printf("DTGA_KERNEL_ASSERT failed: %s at %s:%d (block %d tid %d)\n", \
             #cond, __FILE__, __LINE__, (int)blockIdx.x,                 \
             (int)threadIdx.x);                                          \
      __builtin_trap();                                                  \
    }                                                                    \
  } while (0)
#else
#define DTGA_KERNEL_ASSERT(cond) \
  do {                           \
  } while (0)
#endif

using bf16 = __hip_bfloat16;
typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;

__device__ __forceinline__ float bf2f(short u) {
  union { float f; uint32_t i; } c;
  c.i = ((uint32_t)(uint16_t)u) << 16;
  return c.f;
}

__device__ __forceinline__ short f2bf(float f) {
  // round-to-nearest-even bf16 conversion
  union { float f; uint32_t i; } c;
  c.f = f;
  uint32_t lsb = (c.i >> 16) & 1u;
  uint32_t rounded = c.i + 0x7fffu + lsb;
  if ((c.i & 0x7f800000u) == 0x7f800000u) rounded = c.i;  // inf/nan: truncate
  return (short)(rounded >> 16);
}

// ---- wave reductions (64 lanes) ----
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// ---- block reductions (block size a multiple of 64, <= 1024) ----
// `scratch` must hold >= blockDim.x/64 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) {
    // reduce across the (<=16) wave partials inside wave 0
    for (int off = 1; off < nwaves; off <<= 1) r += __shfl_xor(r, off, WAVE);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
    for (int off = 1; off < nwaves; off <<= 1) r = fmaxf(r, __shfl_xor(r, off, WAVE));
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  float out = scratch[0];
  __syncthreads();
  return out;
}


// 4-wave block sum with ONE barrier (blockDim == 256): wave partials to
// scratch, one sync, every thread folds the 4 partials. The caller's
// loop-end __syncthreads() protects scratch reuse across iterations.
__device__ __forceinline__ float block4_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  return (scratch[0] + scratch[1]) + (scratch[2] + scratch[3]);
}

#define HIP_CHECK_DEV(expr)                                                  \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) return _e;                                         \
  } while (0)

// ---- attention kernel shared idioms (fwd v3 / bwd v2) ----
// pack two f32 into one reg of 2 bf16 (no builtin on gfx950)
__device__ __forceinline__ uint32_t cvt_pk_bf16(float lo, float hi) {
  uint32_t r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// MFMA C-fragment "already-in-B/A-operand-order" row permutation: feeding
// A-operand rows of tile mt in order perm16(mt, l15) makes C position
// (mt, lg, r) hold source row (mt>>1)*32 + lg*8 + (mt&1)*4 + r, i.e. the
// packed C rows are directly the k-dim layout (kc*32 + lg*8 + j) that the
// next MFMA's A/B operand wants.
__device__ __forceinline__ int perm16(int mt, int l15) {
  return (mt >> 1) * 32 + (l15 >> 2) * 8 + (mt & 1) * 4 + (l15 & 3);
}
__device__ __forceinline__ int cpos16(int mt, int lg) {
  return (mt >> 1) * 32 + lg * 8 + (mt & 1) * 4;
}

// ---- ds_read_tr16_b64 (gfx950 hardware 4x4 transpose read) ----
// Probed semantics (profiles/tr16_probe_semantics.log): within each 16-lane
// group, output lane i = 4a+b receives element b of input lanes
// {a, a+4, a+8, a+12}.  With input lane L reading 8 B at
// (row0 + (L>>2), col0 + (L&3)*4) of a row-major [R][C] image, output lane
// i gets rows row0..row0+3 at column col0+i — a [4 row][16 col] block
// delivered column-major (guide T10).
typedef __attribute__((ext_vector_type(4))) short s16x4tr;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
#define LDS_AS __attribute__((address_space(3)))


// 16-col-subtile image ([64 rows][16 cols] blocks, row stride 32 B): the
// guide's conflict-free layout for ds_read_b64_tr_b16, still b128-readable
// per row (8-col runs at col%16 in {0,8}). Pairs with the glds source
// mapping in the attention kernels (chunk ch lands rows (ch&1)*32 + l/2,
// cols (ch>>1)*16 + (l&1)*8).
__device__ __forceinline__ int st_idx(int row, int col) {
  return ((col >> 4) << 10) + (row << 4) + (col & 15);
}

__device__ __forceinline__ s16x8 tr16_frag_st(const short* lds_base, int row0,
                                              int col0, int l15) {
  const int r = row0 + (l15 >> 2);
  const int c = col0 + (l15 & 3) * 4;
  s16x4tr lo = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (LDS_AS s16x4tr*)(lds_base + st_idx(r, c)));
  s16x4tr hi = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (LDS_AS s16x4tr*)(lds_base + st_idx(r + 4, c)));
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    out[j] = lo[j];
    out[4 + j] = hi[j];
  }
  return out;
}
