// Causal flash-attention forward for gfx950 (MFMA 16x16x32 bf16, online
// softmax). Replaces the reference's flash-attn-2 dependency
// (attn_implementation="flash_attention_2", 05:93 / 06:73 / 07:71 —
// SURVEY.md §2b "Flash attention 2").
//
// Layout: q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 contiguous (BSHD — matches the
// projection output so the model never transposes). GQA: Hq % Hkv == 0.
// Causal mask always (training a causal LM). Saves lse = m + log(l)
// ([B,Hq,S] f32) for the backward's recompute.
//
// Structure (correctness-first v1):
//   grid = B * Hq * ceil(S/64); block = 256 threads = 4 waves.
//   Each wave owns 16 query rows; KV tiles of 32 keys stream past.
//   Per KV tile: S = scale*Q.K^T (8 MFMA), online softmax in the C-fragment
//   layout (row reduce = shfl over the 16-lane group), P staged through a
//   1 KiB LDS tile per wave to re-enter MFMA as the A operand, O += P.V
//   (8 MFMA) with the standard rescale.
//
// MFMA fragment maps used here (gfx950 v_mfma_f32_16x16x32_bf16):
//   A[m][k]: lane l holds m = l&15, k = (l>>4)*8 + j (j=0..7)
//   B[k][n]: lane l holds n = l&15, k = (l>>4)*8 + j
//   C/D     : lane l holds n = l&15, m = (l>>4)*4 + r (r=0..3)
// Any consistent bijective k-placement cancels between A and B; the C/D map
// matches the guide (§3 Fragment layout).
#include "common.h"

using bf16x8 = s16x8;  // 8 bf16 in 4 VGPRs: one A/B fragment
using f32x4v = f32x4;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// row-reduce max/sum across the 16 lanes that share a C-fragment row
__device__ __forceinline__ float group16_max(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__global__ void __launch_bounds__(256) attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, int D,
    float scale) {
  // LDS: per-wave [16][32] bf16 P-staging tile
  __shared__ short p_lds[4][16 * 32];

  const int ntiles_q = (S + 63) / 64;
  const int bid = blockIdx.x;
  const int qtile = bid % ntiles_q;
  const int h = (bid / ntiles_q) % Hq;
  const int b = bid / (ntiles_q * Hq);
  const int hkv = h / (Hq / Hkv);

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;  // 0..3: k-group (A/B) or row-group (C)

  const int q0 = qtile * 64 + wid * 16;  // first query row of this wave
  // NOTE: waves whose rows fall past S still run (rows clamped, stores
  // guarded) so every wave executes the same barrier sequence.
  const int nd = D / 16;                 // # of 16-wide d tiles (4 or 8)
  const int ndk = D / 32;                // # of 32-deep k chunks (2 or 4)

  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* qb = q + ((int64_t)b * S * Hq + h) * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;

  // ---- load Q fragments (row = l15, 8 contiguous d at lg*8 per chunk) ----
  bf16x8 qf[4];  // up to D=128 -> 4 chunks of 32
  {
    int qrow = q0 + l15;
    if (qrow >= S) qrow = S - 1;  // clamped; stores are guarded later
    const short* qp = qb + (int64_t)qrow * strideS_q;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < ndk)
        qf[c] = *reinterpret_cast<const bf16x8*>(qp + c * 32 + lg * 8);
  }

  // ---- running state ----
  f32x4 oacc[8];  // [nd] d-tiles, C layout
#pragma unroll
  for (int i = 0; i < 8; ++i) oacc[i] = {0.f, 0.f, 0.f, 0.f};
  float mrow[4], lrow[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { mrow[r] = -INFINITY; lrow[r] = 0.f; }

  // causal loop bound, uniform across the block's 4 waves so the barriers
  // match (waves left of the diagonal see fully-masked tiles -> P == 0).
  const int kv_end = min(S, qtile * 64 + 64);
  for (int kv0 = 0; kv0 < kv_end; kv0 += 32) {
    // ---- S = scale * Q K^T for keys [kv0, kv0+32) : 2 C-frags ----
    f32x4 sfrag[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      int key = kv0 + half * 16 + l15;
      int keyc = key < S ? key : S - 1;
      const short* kp = kb + (int64_t)keyc * strideS_kv;
#pragma unroll
      for (int c = 0; c < 4; ++c)
        if (c < ndk) {
          bf16x8 kf = *reinterpret_cast<const bf16x8*>(kp + c * 32 + lg * 8);
          acc = mfma16(qf[c], kf, acc);
        }
      sfrag[half] = acc;
    }
    // ---- causal mask + online softmax (C layout) ----
    float tile_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = -INFINITY;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int key = kv0 + half * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qrow = q0 + lg * 4 + r;
        float s = sfrag[half][r] * scale;
        if (key > qrow || key >= S) s = -INFINITY;
        sfrag[half][r] = s;
        tile_max[r] = fmaxf(tile_max[r], s);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = group16_max(tile_max[r]);

    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float mnew = fmaxf(mrow[r], tile_max[r]);
      alpha[r] = (mrow[r] == -INFINITY) ? 0.0f : __expf(mrow[r] - mnew);
      mrow[r] = mnew;
    }
    // P = exp(s - m); row-sum
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = (sfrag[half][r] == -INFINITY)
                      ? 0.0f
                      : __expf(sfrag[half][r] - mrow[r]);
        sfrag[half][r] = p;
        psum[r] += p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      psum[r] = group16_sum(psum[r]);
      lrow[r] = lrow[r] * alpha[r] + psum[r];
    }
    // ---- stage P to LDS (C layout -> A layout) ----
    short* pl = p_lds[wid];
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pl[(lg * 4 + r) * 32 + half * 16 + l15] = f2bf(sfrag[half][r]);
    __builtin_amdgcn_s_barrier();  // wave-local LDS tile; barrier syncs block
    bf16x8 pf = *reinterpret_cast<const bf16x8*>(pl + l15 * 32 + lg * 8);

    // ---- O = O*alpha + P V ----
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      // B fragment of V: n = d tile col l15, k = key kv0 + lg*8 + j
      bf16x8 vf;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int key = kv0 + lg * 8 + j;
        int keyc = key < S ? key : S - 1;
        vf[j] = vb[(int64_t)keyc * strideS_kv + dt * 16 + l15];
      }
      f32x4 acc = oacc[dt];
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[r] *= alpha[r];
      oacc[dt] = mfma16(pf, vf, acc);
    }
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue: O /= l, store O and lse ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int qrow = q0 + lg * 4 + r;
    if (qrow >= S) continue;
    float invl = (lrow[r] > 0.f) ? 1.0f / lrow[r] : 0.0f;
    short* op = o + ((int64_t)b * S * Hq + (int64_t)qrow * Hq + h) * D;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      op[dt * 16 + l15] = f2bf(oacc[dt][r] * invl);
    }
    if (l15 == 0)
      lse_out[((int64_t)b * Hq + h) * S + qrow] =
          mrow[r] + __logf(fmaxf(lrow[r], 1e-30f));
  }
}

extern "C" {
void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int S, int Hq, int Hkv, int D,
                     float scale, hipStream_t stream) {
  int ntiles = (S + 63) / 64;
  int64_t grid = (int64_t)B * Hq * ntiles;
  hipLaunchKernelGGL(attn_fwd_kernel, dim3((uint32_t)grid), dim3(256), 0,
                     stream, (const short*)q, (const short*)k, (const short*)v,
                     (short*)o, lse, B, S, Hq, Hkv, D, scale);
}
}
