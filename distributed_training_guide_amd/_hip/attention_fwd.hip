// Causal flash-attention forward for gfx950 (MFMA 16x16x32 bf16, online
// softmax) — v2: LDS-staged K/V tiles with XOR swizzle.
//
// Replaces the reference's flash-attn-2 dependency
// (attn_implementation="flash_attention_2", 05:93 / 06:73 / 07:71 —
// SURVEY.md §2b "Flash attention 2").
//
// Layout: q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 contiguous (BSHD). GQA via
// Hq % Hkv == 0. Causal always. Saves lse [B,Hq,S] f32 for the backward.
//
// Structure:
//   grid = B * Hq * ceil(S/128); block = 256 threads = 4 waves.
//   Each wave owns 32 query rows (Q fragments live in registers for the
//   whole kernel); KV tiles of 64 keys are cooperatively staged into LDS
//   (vectorized 16 B global loads, byte ^= (row&7)<<4 slot swizzle so the
//   per-lane-row ds_read_b128 fragment reads are conflict-free — guide §6
//   G4) and shared by all 4 waves. Per tile: S = scale*QK^T (32 MFMA),
//   online softmax in the C-fragment layout, P staged through a per-wave
//   swizzled LDS tile to re-enter MFMA as the A operand, O += P.V
//   (32 MFMA) with the standard rescale.
//
// MFMA fragment maps (gfx950 v_mfma_f32_16x16x32_bf16):
//   A[m][k]: lane l holds m = l&15, k = (l>>4)*8 + j (j=0..7)
//   B[k][n]: lane l holds n = l&15, k = (l>>4)*8 + j
//   C/D     : lane l holds n = l&15, m = (l>>4)*4 + r (r=0..3)
#include "common.h"

using bf16x8 = s16x8;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

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

// swizzled element index into a [rows][D] bf16 LDS tile (16B slot XOR)
__device__ __forceinline__ int swz(int row, int d, int D) {
  return row * D + (((d >> 3) ^ (row & 7)) << 3) + (d & 7);
}

#define QBLK 128  // q rows per block (32 per wave)
#define KVBLK 64  // keys per LDS tile

__global__ void __launch_bounds__(256) attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, int D,
    float scale) {
  __shared__ short k_lds[KVBLK * 128];
  __shared__ short v_lds[KVBLK * 128];
  __shared__ short p_lds[4][32 * KVBLK];  // per-wave [32 q][64 k], swizzled

  const int ntiles_q = (S + QBLK - 1) / QBLK;
  const int bid = blockIdx.x;
  const int qtile = bid % ntiles_q;
  const int h = (bid / ntiles_q) % Hq;
  const int b = bid / (ntiles_q * Hq);
  const int hkv = h / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int q0 = qtile * QBLK + wid * 32;  // this wave's 32 q rows
  const int nd = D / 16, ndk = D / 32;
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* qb = q + ((int64_t)b * S * Hq + h) * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;

  // ---- Q fragments: [2 mh][ndk chunks], rows q0 + mh*16 + l15 ----
  bf16x8 qf[2][4];
#pragma unroll
  for (int mh = 0; mh < 2; ++mh) {
    int qrow = q0 + mh * 16 + l15;
    if (qrow >= S) qrow = S - 1;
    const short* qp = qb + (int64_t)qrow * strideS_q;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < ndk)
        qf[mh][c] = *reinterpret_cast<const bf16x8*>(qp + c * 32 + lg * 8);
  }

  f32x4 oacc[2][8];
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int i = 0; i < 8; ++i) oacc[mh][i] = {0.f, 0.f, 0.f, 0.f};
  float mrow[2][4], lrow[2][4];
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      mrow[mh][r] = -INFINITY;
      lrow[mh][r] = 0.f;
    }

  const int kv_end = min(S, qtile * QBLK + QBLK);  // block-uniform causal
  const int nvec = KVBLK * D / 8 / 256;            // 16B vectors per thread

  for (int kv0 = 0; kv0 < kv_end; kv0 += KVBLK) {
    // ---- cooperative staging: K and V tiles, swizzled ----
#pragma unroll
    for (int vv = 0; vv < 4; ++vv) {
      if (vv >= nvec) break;
      int vecid = vv * 256 + tid;
      int key = vecid / (D / 8);
      int slot = vecid % (D / 8);
      int keyg = kv0 + key;
      if (keyg >= S) keyg = S - 1;
      const short* kp = kb + (int64_t)keyg * strideS_kv + slot * 8;
      const short* vp = vb + (int64_t)keyg * strideS_kv + slot * 8;
      int dst = key * D + (((slot ^ (key & 7)) & (D / 8 - 1)) << 3);
      *reinterpret_cast<bf16x8*>(k_lds + dst) =
          *reinterpret_cast<const bf16x8*>(kp);
      *reinterpret_cast<bf16x8*>(v_lds + dst) =
          *reinterpret_cast<const bf16x8*>(vp);
    }
    __builtin_amdgcn_s_barrier();

    // waves fully left of the diagonal skip compute (barriers stay uniform)
    const bool active = (kv0 <= q0 + 31);

    // ---- S = scale * Q K^T : [2 mh][4 ntiles] C-frags ----
    f32x4 sfrag[2][4];
    float alpha[2][4];
    if (active) {
#pragma unroll
    for (int mh = 0; mh < 2; ++mh)
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        int key = nt * 16 + l15;
#pragma unroll
        for (int c = 0; c < 4; ++c)
          if (c < ndk) {
            int d = c * 32 + lg * 8;
            bf16x8 kf = *reinterpret_cast<const bf16x8*>(
                k_lds + swz(key, d, D));
            acc = mfma16(qf[mh][c], kf, acc);
          }
        sfrag[mh][nt] = acc;
      }

    // ---- causal mask + online softmax (C layout) ----
#pragma unroll
    for (int mh = 0; mh < 2; ++mh) {
      float tmax[4] = {-INFINITY, -INFINITY, -INFINITY, -INFINITY};
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int key = kv0 + nt * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int qrow = q0 + mh * 16 + lg * 4 + r;
          float s = sfrag[mh][nt][r] * scale;
          if (key > qrow || key >= S) s = -INFINITY;
          sfrag[mh][nt][r] = s;
          tmax[r] = fmaxf(tmax[r], s);
        }
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        tmax[r] = group16_max(tmax[r]);
        float mnew = fmaxf(mrow[mh][r], tmax[r]);
        alpha[mh][r] =
            (mrow[mh][r] == -INFINITY) ? 0.0f : __expf(mrow[mh][r] - mnew);
        mrow[mh][r] = mnew;
      }
      float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float p = (sfrag[mh][nt][r] == -INFINITY)
                        ? 0.0f
                        : __expf(sfrag[mh][nt][r] - mrow[mh][r]);
          sfrag[mh][nt][r] = p;
          psum[r] += p;
        }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        psum[r] = group16_sum(psum[r]);
        lrow[mh][r] = lrow[mh][r] * alpha[mh][r] + psum[r];
      }
      // stage P (swizzled [32 q][64 k] per-wave tile)
      short* pl = p_lds[wid];
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        int col = nt * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int row = mh * 16 + lg * 4 + r;
          pl[swz(row, col, KVBLK)] = f2bf(sfrag[mh][nt][r]);
        }
      }
    }
    }  // active
    __builtin_amdgcn_s_barrier();

    // ---- P A-frags + O accumulate ----
    if (active) {
    bf16x8 pa[2][2];
#pragma unroll
    for (int mh = 0; mh < 2; ++mh)
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        int row = mh * 16 + l15;
        int kk = kc * 32 + lg * 8;
        pa[mh][kc] = *reinterpret_cast<const bf16x8*>(
            p_lds[wid] + swz(row, kk, KVBLK));
      }
    bool rescale_done[2] = {false, false};
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      bf16x8 vf[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        int d = dt * 16 + l15;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int key = kc * 32 + lg * 8 + j;
          vf[kc][j] = v_lds[swz(key, d, D)];
        }
      }
#pragma unroll
      for (int mh = 0; mh < 2; ++mh) {
        f32x4 acc = oacc[mh][dt];
#pragma unroll
        for (int r = 0; r < 4; ++r) acc[r] *= alpha[mh][r];
        acc = mfma16(pa[mh][0], vf[0], acc);
        acc = mfma16(pa[mh][1], vf[1], acc);
        oacc[mh][dt] = acc;
      }
    }
    (void)rescale_done;
    }  // active
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue ----
#pragma unroll
  for (int mh = 0; mh < 2; ++mh)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qrow = q0 + mh * 16 + lg * 4 + r;
      if (qrow >= S) continue;
      float invl = (lrow[mh][r] > 0.f) ? 1.0f / lrow[mh][r] : 0.0f;
      short* op = o + ((int64_t)b * S * Hq + (int64_t)qrow * Hq + h) * D;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        if (dt >= nd) break;
        op[dt * 16 + l15] = f2bf(oacc[mh][dt][r] * invl);
      }
      if (l15 == 0)
        lse_out[((int64_t)b * Hq + h) * S + qrow] =
            mrow[mh][r] + __logf(fmaxf(lrow[mh][r], 1e-30f));
    }
}

extern "C" {
void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, int B, int S, int Hq, int Hkv, int D,
                     float scale, hipStream_t stream) {
  int ntiles = (S + QBLK - 1) / QBLK;
  int64_t grid = (int64_t)B * Hq * ntiles;
  hipLaunchKernelGGL(attn_fwd_kernel, dim3((uint32_t)grid), dim3(256), 0,
                     stream, (const short*)q, (const short*)k, (const short*)v,
                     (short*)o, lse, B, S, Hq, Hkv, D, scale);
}
}
