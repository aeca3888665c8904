// Fused qkv-split + RoPE for gfx950: one pass over the packed qkv
// projection output produces rotated contiguous q and k plus contiguous v
// (forward), and one pass re-packs dq/dk/dv into dqkv with the inverse
// rotation (backward).  Replaces the split -> 2x rope (+contiguous copies)
// -> v.contiguous chain and the backward grad-cat (SURVEY.md §2b "RoPE";
// rotate-half convention as rope.hip).
//
// Layouts: qkv [B,S,W] bf16, W = (Hq+2*Hkv)*D; q [B,S,Hq,D];
// k/v [B,S,Hkv,D]; cos/sin f32 [max_pos, D/2]; positions int32 [S] or null.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4v;

// vectorized over 8 consecutive rotation pairs per thread: bf16x8 loads of
// the (d2, d2+halfD) halves + two f32x4 table loads each — the scalar-pair
// version ran at 4x the traffic floor (requires D % 16 == 0, true for all
// registry models)
__global__ void __launch_bounds__(256) qkv_rope_fwd_kernel(
    const short* __restrict__ qkv, short* __restrict__ oq,
    short* __restrict__ ok, short* __restrict__ ov,
    const float* __restrict__ cs, const float* __restrict__ sn,
    const int* __restrict__ positions, int64_t BS, int S, int Hq, int Hkv,
    int D) {
  const int halfD = D / 2;
  const int HA = Hq + 2 * Hkv;
  const int64_t W = (int64_t)HA * D;
  const int nb = halfD / 8;  // 8-pair blocks per head row
  int64_t nelem = BS * HA * nb;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < nelem; idx += (int64_t)gridDim.x * blockDim.x) {
    const int d2 = (int)(idx % nb) * 8;
    const int64_t rowa = idx / nb;
    const int ha = (int)(rowa % HA);
    const int64_t bs = rowa / HA;
    const int s = (int)(bs % S);
    const short* src = qkv + bs * W + (int64_t)ha * D;
    s16x8 v0 = *reinterpret_cast<const s16x8*>(src + d2);
    s16x8 v1 = *reinterpret_cast<const s16x8*>(src + d2 + halfD);
    if (ha < Hq + Hkv) {  // q or k: rotate
      const int pos = positions ? positions[s] : s;
      const float* cp = cs + (int64_t)pos * halfD + d2;
      const float* sp = sn + (int64_t)pos * halfD + d2;
      f32x4v c0 = *reinterpret_cast<const f32x4v*>(cp);
      f32x4v c1 = *reinterpret_cast<const f32x4v*>(cp + 4);
      f32x4v s0 = *reinterpret_cast<const f32x4v*>(sp);
      f32x4v s1 = *reinterpret_cast<const f32x4v*>(sp + 4);
      s16x8 o0, o1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float c = j < 4 ? c0[j & 3] : c1[j & 3];
        const float sv = j < 4 ? s0[j & 3] : s1[j & 3];
        const float x0 = bf2f(v0[j]);
        const float x1 = bf2f(v1[j]);
        o0[j] = f2bf(x0 * c - x1 * sv);
        o1[j] = f2bf(x0 * sv + x1 * c);
      }
      short* dst = (ha < Hq)
                       ? oq + (bs * Hq + ha) * (int64_t)D
                       : ok + (bs * Hkv + (ha - Hq)) * (int64_t)D;
      *reinterpret_cast<s16x8*>(dst + d2) = o0;
      *reinterpret_cast<s16x8*>(dst + d2 + halfD) = o1;
    } else {  // v: plain copy
      short* dst = ov + (bs * Hkv + (ha - Hq - Hkv)) * (int64_t)D;
      *reinterpret_cast<s16x8*>(dst + d2) = v0;
      *reinterpret_cast<s16x8*>(dst + d2 + halfD) = v1;
    }
  }
}

__global__ void __launch_bounds__(256) qkv_rope_bwd_kernel(
    const short* __restrict__ dq, const short* __restrict__ dk,
    const short* __restrict__ dv, short* __restrict__ dqkv,
    const float* __restrict__ cs, const float* __restrict__ sn,
    const int* __restrict__ positions, int64_t BS, int S, int Hq, int Hkv,
    int D) {
  const int halfD = D / 2;
  const int HA = Hq + 2 * Hkv;
  const int64_t W = (int64_t)HA * D;
  const int nb = halfD / 8;
  int64_t nelem = BS * HA * nb;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < nelem; idx += (int64_t)gridDim.x * blockDim.x) {
    const int d2 = (int)(idx % nb) * 8;
    const int64_t rowa = idx / nb;
    const int ha = (int)(rowa % HA);
    const int64_t bs = rowa / HA;
    const int s = (int)(bs % S);
    short* dst = dqkv + bs * W + (int64_t)ha * D;
    if (ha < Hq + Hkv) {  // dq or dk: inverse rotation
      const short* src = (ha < Hq)
                             ? dq + (bs * Hq + ha) * (int64_t)D
                             : dk + (bs * Hkv + (ha - Hq)) * (int64_t)D;
      s16x8 v0 = *reinterpret_cast<const s16x8*>(src + d2);
      s16x8 v1 = *reinterpret_cast<const s16x8*>(src + d2 + halfD);
      const int pos = positions ? positions[s] : s;
      const float* cp = cs + (int64_t)pos * halfD + d2;
      const float* sp = sn + (int64_t)pos * halfD + d2;
      f32x4v c0 = *reinterpret_cast<const f32x4v*>(cp);
      f32x4v c1 = *reinterpret_cast<const f32x4v*>(cp + 4);
      f32x4v s0 = *reinterpret_cast<const f32x4v*>(sp);
      f32x4v s1 = *reinterpret_cast<const f32x4v*>(sp + 4);
      s16x8 o0, o1;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float c = j < 4 ? c0[j & 3] : c1[j & 3];
        const float sv = j < 4 ? s0[j & 3] : s1[j & 3];
        const float g0 = bf2f(v0[j]);
        const float g1 = bf2f(v1[j]);
        o0[j] = f2bf(g0 * c + g1 * sv);
        o1[j] = f2bf(-g0 * sv + g1 * c);
      }
      *reinterpret_cast<s16x8*>(dst + d2) = o0;
      *reinterpret_cast<s16x8*>(dst + d2 + halfD) = o1;
    } else {
      const short* src = dv + (bs * Hkv + (ha - Hq - Hkv)) * (int64_t)D;
      *reinterpret_cast<s16x8*>(dst + d2) =
          *reinterpret_cast<const s16x8*>(src + d2);
      *reinterpret_cast<s16x8*>(dst + d2 + halfD) =
          *reinterpret_cast<const s16x8*>(src + d2 + halfD);
    }
  }
}

extern "C" {
void qkv_rope_fwd_launch(const void* qkv, void* q, void* k, void* v,
                         const float* cs, const float* sn,
                         const int* positions, int64_t BS, int S, int Hq,
                         int Hkv, int D, hipStream_t stream) {
  int64_t nelem = BS * (Hq + 2 * Hkv) * (D / 16);
  int64_t want = (nelem + 255) / 256;
  int grid = (int)(want < 4096 ? (want < 1 ? 1 : want) : 4096);
  hipLaunchKernelGGL(qkv_rope_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)qkv, (short*)q, (short*)k, (short*)v, cs,
                     sn, positions, BS, S, Hq, Hkv, D);
}
void qkv_rope_bwd_launch(const void* dq, const void* dk, const void* dv,
                         void* dqkv, const float* cs, const float* sn,
                         const int* positions, int64_t BS, int S, int Hq,
                         int Hkv, int D, hipStream_t stream) {
  int64_t nelem = BS * (Hq + 2 * Hkv) * (D / 16);
  int64_t want = (nelem + 255) / 256;
  int grid = (int)(want < 4096 ? (want < 1 ? 1 : want) : 4096);
  hipLaunchKernelGGL(qkv_rope_bwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)dq, (const short*)dk, (const short*)dv,
                     (short*)dqkv, cs, sn, positions, BS, S, Hq, Hkv, D);
}
}
