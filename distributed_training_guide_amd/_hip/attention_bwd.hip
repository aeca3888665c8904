// Causal flash-attention backward for gfx950 (MFMA 16x16x32 bf16).
// Counterpart of attention_fwd.hip; replaces flash-attn-2's backward
// (SURVEY.md §2b "Flash attention 2", "bwd recompute variant").
//
// Standard flash-2 backward with lse recompute, split into three atomic-free
// kernels (the loop order that would fuse them needs cross-block atomics on
// dq — prohibitive write amplification at training shapes):
//   1. delta[b,h,s] = rowsum(dO * O)
//   2. dk/dv: block owns 64 keys of one (b, hkv) (one wave = 16 keys);
//      loops q tiles of 32 rows and the GQA query-head group, accumulating
//      dK/dV in registers; one exclusive store.
//   3. dq: block owns 64 q rows of one (b, hq) (one wave = 16 rows); loops
//      kv tiles of 32 keys accumulating dQ in registers; one store.
// Math (S_raw = Q.K^T, P = exp(scale*S_raw - lse)):
//   dV += P^T dO
//   dP = dO V^T;  dS_raw = scale * P * (dP - delta)
//   dK += dS_raw^T Q;  dQ += dS_raw K
//
// Fragment maps as in attention_fwd.hip:
//   A[m][k]: lane l -> m = l&15, k = (l>>4)*8 + j   (k spans 0..31)
//   B[k][n]: lane l -> n = l&15, k = (l>>4)*8 + j
//   C/D    : lane l -> n = l&15, m = (l>>4)*4 + r
#include "common.h"

using bf16x8 = s16x8;

__device__ __forceinline__ f32x4 mfma16b(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// ---------------- delta = rowsum(dO * O) ----------------
__global__ void __launch_bounds__(256) attn_delta_kernel(
    const short* __restrict__ dout, const short* __restrict__ o,
    float* __restrict__ delta, int B, int S, int Hq, int D) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  int64_t nrows = (int64_t)B * S * Hq;
  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < nrows;
       row += (int64_t)gridDim.x * 4) {
    const short* dp = dout + row * D;
    const short* op = o + row * D;
    float s = 0.f;
    for (int i = lane; i < D; i += WAVE) s += bf2f(dp[i]) * bf2f(op[i]);
    s = wave_reduce_sum(s);
    if (lane == 0) {
      // row = (b*S + s_pos)*Hq + h  ->  delta is [B, Hq, S]
      int h = (int)(row % Hq);
      int64_t bs = row / Hq;
      int sp = (int)(bs % S);
      int b = (int)(bs / S);
      delta[((int64_t)b * Hq + h) * S + sp] = s;
    }
  }
}

// ---------------- dk / dv ----------------
__global__ void __launch_bounds__(256) attn_dkdv_kernel(
    const short* __restrict__ dout, const short* __restrict__ q,
    const short* __restrict__ k, const short* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int B, int S, int Hq,
    int Hkv, int D, float scale) {
  // per-wave LDS: [32 q][16 k] bf16 staging tiles for P and dS
  __shared__ short p_lds[4][32 * 16];
  __shared__ short ds_lds[4][32 * 16];

  const int nkt = (S + 63) / 64;
  const int kt = blockIdx.x % nkt;
  const int hkv = (blockIdx.x / nkt) % Hkv;
  const int b = blockIdx.x / (nkt * Hkv);
  const int group = Hq / Hkv;

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int kv0 = kt * 64 + wid * 16;  // this wave's 16 keys
  const int nd = D / 16, ndk = D / 32;
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;

  // K/V B-fragments (n = key l15, k = d lg*8+j — contiguous 16 B loads)
  bf16x8 kf[4], vf[4];
  {
    int key = kv0 + l15;
    int keyc = key < S ? key : S - 1;
    const short* kp = kb + (int64_t)keyc * strideS_kv;
    const short* vp = vb + (int64_t)keyc * strideS_kv;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < ndk) {
        kf[c] = *reinterpret_cast<const bf16x8*>(kp + c * 32 + lg * 8);
        vf[c] = *reinterpret_cast<const bf16x8*>(vp + c * 32 + lg * 8);
      }
  }

  // dK/dV accumulators: [16 keys][D] as C-frags over nd d-tiles
  f32x4 dkacc[8], dvacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    dkacc[i] = {0.f, 0.f, 0.f, 0.f};
    dvacc[i] = {0.f, 0.f, 0.f, 0.f};
  }

  const int qstart = kt * 64;  // block-uniform causal start
  for (int h = hkv * group; h < (hkv + 1) * group; ++h) {
    const short* qb = q + ((int64_t)b * S * Hq + h) * D;
    const short* dob = dout + ((int64_t)b * S * Hq + h) * D;
    const float* lseb = lse + ((int64_t)b * Hq + h) * S;
    const float* delb = delta + ((int64_t)b * Hq + h) * S;
    for (int qt = qstart; qt < S; qt += 32) {
      // ---- S_raw[32q][16k], dP[32q][16k] as two 16-row C-frags each ----
      f32x4 sfrag[2], dpfrag[2];
#pragma unroll
      for (int mh = 0; mh < 2; ++mh) {
        f32x4 sa = {0.f, 0.f, 0.f, 0.f}, da = {0.f, 0.f, 0.f, 0.f};
        int qrow = qt + mh * 16 + l15;
        int qrc = qrow < S ? qrow : S - 1;
        const short* qp = qb + (int64_t)qrc * strideS_q;
        const short* dop = dob + (int64_t)qrc * strideS_q;
#pragma unroll
        for (int c = 0; c < 4; ++c)
          if (c < ndk) {
            bf16x8 qa = *reinterpret_cast<const bf16x8*>(qp + c * 32 + lg * 8);
            bf16x8 doa =
                *reinterpret_cast<const bf16x8*>(dop + c * 32 + lg * 8);
            sa = mfma16b(qa, kf[c], sa);
            da = mfma16b(doa, vf[c], da);
          }
        sfrag[mh] = sa;
        dpfrag[mh] = da;
      }
      // ---- P, dS (C layout: col=key kv0+l15, row=qt+mh*16+lg*4+r) ----
      {
        int key = kv0 + l15;
        short* pl = p_lds[wid];
        short* dsl = ds_lds[wid];
#pragma unroll
        for (int mh = 0; mh < 2; ++mh) {
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int qrow = qt + mh * 16 + lg * 4 + r;
            float p = 0.f, ds = 0.f;
            if (key <= qrow && key < S && qrow < S) {
              float pexp = __expf(scale * sfrag[mh][r] - lseb[qrow]);
              p = pexp;
              ds = scale * pexp * (dpfrag[mh][r] - delb[qrow]);
            }
            int qloc = mh * 16 + lg * 4 + r;
            pl[qloc * 16 + l15] = f2bf(p);
            dsl[qloc * 16 + l15] = f2bf(ds);
          }
        }
      }
      __builtin_amdgcn_s_barrier();
      // ---- A-frags of P^T / dS^T: m = key(l15), k = q(lg*8+j spans 32) ----
      bf16x8 pt, dst;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int qloc = lg * 8 + j;
        pt[j] = p_lds[wid][qloc * 16 + l15];
        dst[j] = ds_lds[wid][qloc * 16 + l15];
      }
      __builtin_amdgcn_s_barrier();
      // ---- dV += P^T dO ; dK += dS^T Q (B strided over 32 q rows) ----
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        if (dt >= nd) break;
        bf16x8 dof, qf2;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int qrow = qt + lg * 8 + j;
          int qrc = qrow < S ? qrow : S - 1;
          dof[j] = dob[(int64_t)qrc * strideS_q + dt * 16 + l15];
          qf2[j] = qb[(int64_t)qrc * strideS_q + dt * 16 + l15];
        }
        dvacc[dt] = mfma16b(pt, dof, dvacc[dt]);
        dkacc[dt] = mfma16b(dst, qf2, dkacc[dt]);
      }
    }
  }

  // ---- store dK/dV (exclusive: one block per (b,hkv,key)) ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int key = kv0 + lg * 4 + r;
    if (key >= S) continue;
    short* dkp = dk + ((int64_t)b * S * Hkv + (int64_t)key * Hkv + hkv) * D;
    short* dvp = dv + ((int64_t)b * S * Hkv + (int64_t)key * Hkv + hkv) * D;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      dkp[dt * 16 + l15] = f2bf(dkacc[dt][r]);
      dvp[dt * 16 + l15] = f2bf(dvacc[dt][r]);
    }
  }
}

// ---------------- dq ----------------
__global__ void __launch_bounds__(256) attn_dq_kernel(
    const short* __restrict__ dout, const short* __restrict__ q,
    const short* __restrict__ k, const short* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int B, int S, int Hq, int Hkv, int D,
    float scale) {
  // per-wave LDS: [16 q][32 k] bf16 staging for dS
  __shared__ short ds_lds[4][16 * 32];

  const int nqt = (S + 63) / 64;
  const int qtile = blockIdx.x % nqt;
  const int h = (blockIdx.x / nqt) % Hq;
  const int b = blockIdx.x / (nqt * Hq);
  const int hkv = h / (Hq / Hkv);

  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int q0 = qtile * 64 + wid * 16;
  const int nd = D / 16, ndk = D / 32;
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* qb = q + ((int64_t)b * S * Hq + h) * D;
  const short* dob = dout + ((int64_t)b * S * Hq + h) * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;
  const float* lseb = lse + ((int64_t)b * Hq + h) * S;
  const float* delb = delta + ((int64_t)b * Hq + h) * S;

  // Q / dO A-fragments for this wave's 16 rows
  bf16x8 qf[4], dof[4];
  {
    int qrow = q0 + l15;
    int qrc = qrow < S ? qrow : S - 1;
    const short* qp = qb + (int64_t)qrc * strideS_q;
    const short* dop = dob + (int64_t)qrc * strideS_q;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < ndk) {
        qf[c] = *reinterpret_cast<const bf16x8*>(qp + c * 32 + lg * 8);
        dof[c] = *reinterpret_cast<const bf16x8*>(dop + c * 32 + lg * 8);
      }
  }

  // dQ accumulator: [16 q][D] C-frags
  f32x4 dqacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) dqacc[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = min(S, qtile * 64 + 64);  // block-uniform
  for (int kv0 = 0; kv0 < kv_end; kv0 += 32) {
    // ---- S_raw[16q][32k], dP[16q][32k] (two key-halves of C-frags) ----
    f32x4 sfrag[2], dpfrag[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      f32x4 sa = {0.f, 0.f, 0.f, 0.f}, da = {0.f, 0.f, 0.f, 0.f};
      int key = kv0 + half * 16 + l15;
      int keyc = key < S ? key : S - 1;
      const short* kp = kb + (int64_t)keyc * strideS_kv;
      const short* vp = vb + (int64_t)keyc * strideS_kv;
#pragma unroll
      for (int c = 0; c < 4; ++c)
        if (c < ndk) {
          bf16x8 kfr = *reinterpret_cast<const bf16x8*>(kp + c * 32 + lg * 8);
          bf16x8 vfr = *reinterpret_cast<const bf16x8*>(vp + c * 32 + lg * 8);
          sa = mfma16b(qf[c], kfr, sa);
          da = mfma16b(dof[c], vfr, da);
        }
      sfrag[half] = sa;
      dpfrag[half] = da;
    }
    // ---- dS (C layout) -> LDS [16 q][32 k] ----
    short* dsl = ds_lds[wid];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      int key = kv0 + half * 16 + l15;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int qrow = q0 + lg * 4 + r;
        float ds = 0.f;
        if (key <= qrow && key < S && qrow < S) {
          float pexp = __expf(scale * sfrag[half][r] - lseb[qrow]);
          ds = scale * pexp * (dpfrag[half][r] - delb[qrow]);
        }
        dsl[(lg * 4 + r) * 32 + half * 16 + l15] = f2bf(ds);
      }
    }
    __builtin_amdgcn_s_barrier();
    // A-frag of dS: m = q(l15), k = key(lg*8+j)
    bf16x8 dsa = *reinterpret_cast<const bf16x8*>(dsl + l15 * 32 + lg * 8);
    __builtin_amdgcn_s_barrier();
    // ---- dQ += dS K : B[k=key][n=d] strided over keys ----
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      bf16x8 kfr;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int key = kv0 + lg * 8 + j;
        int keyc = key < S ? key : S - 1;
        kfr[j] = kb[(int64_t)keyc * strideS_kv + dt * 16 + l15];
      }
      dqacc[dt] = mfma16b(dsa, kfr, dqacc[dt]);
    }
  }

  // ---- store dQ ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int qrow = q0 + lg * 4 + r;
    if (qrow >= S) continue;
    short* dqp = dq + ((int64_t)b * S * Hq + (int64_t)qrow * Hq + h) * D;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd) break;
      dqp[dt * 16 + l15] = f2bf(dqacc[dt][r]);
    }
  }
}

extern "C" {
void attn_bwd_launch(const void* dout, const void* q, const void* k,
                     const void* v, const void* o, const float* lse,
                     float* delta, void* dq, void* dk, void* dv, int B, int S,
                     int Hq, int Hkv, int D, float scale, hipStream_t stream) {
  int64_t nrows = (int64_t)B * S * Hq;
  int64_t dwant = (nrows + 3) / 4;
  int dgrid = (int)(dwant < 2048 ? (dwant < 1 ? 1 : dwant) : 2048);
  hipLaunchKernelGGL(attn_delta_kernel, dim3(dgrid), dim3(256), 0, stream,
                     (const short*)dout, (const short*)o, delta, B, S, Hq, D);
  int nkt = (S + 63) / 64;
  hipLaunchKernelGGL(attn_dkdv_kernel, dim3((uint32_t)((int64_t)B * Hkv * nkt)),
                     dim3(256), 0, stream, (const short*)dout, (const short*)q,
                     (const short*)k, (const short*)v, lse, delta, (short*)dk,
                     (short*)dv, B, S, Hq, Hkv, D, scale);
  hipLaunchKernelGGL(attn_dq_kernel, dim3((uint32_t)((int64_t)B * Hq * nkt)),
                     dim3(256), 0, stream, (const short*)dout, (const short*)q,
                     (const short*)k, (const short*)v, lse, delta, (short*)dq,
                     B, S, Hq, Hkv, D, scale);
}
}
