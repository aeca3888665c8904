// Causal flash-attention backward for gfx950 — v2: same idiom set as
// attention_fwd.hip v3 (swapped/permuted MFMA fragments with in-register
// P/dS, direct-from-L2 A/B operand loads, transposed double-buffered LDS
// for the k-dim-transposed operands).
// Counterpart of attention_fwd.hip; replaces flash-attn-2's backward
// (SURVEY.md §2b "Flash attention 2", "bwd recompute variant").
//
// Standard flash-2 backward with lse recompute, split into three atomic-free
// kernels (the loop order that would fuse them needs cross-block atomics on
// dq — prohibitive write amplification at training shapes):
//   1. delta[b,h,s] = rowsum(dO * O)
//   2. dk/dv: block owns 64 keys of one (b, hkv) (one wave = 16 keys);
//      loops (gqa-head, 64-q-row tile), accumulating dK/dV in registers.
//      Q^T / dO^T tiles staged transposed+double-buffered in LDS; Q/dO
//      A-fragments read straight from global (L2-resident with the
//      XCD-aware remap); P^T/dS^T stay in registers via the perm16 trick.
//   3. dq: block owns 128 q rows of one (b, hq) (one wave = 32); loops
//      64-key tiles; the fwd-v3 structure with K^T staged transposed and
//      dS packed in registers as the dQ MFMA's A operand.
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
#define LOG2E 1.4426950408889634f

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
    for (int i = lane * 8; i < D; i += WAVE * 8) {
      bf16x8 dv = *reinterpret_cast<const bf16x8*>(dp + i);
      bf16x8 ov = *reinterpret_cast<const bf16x8*>(op + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bf2f(dv[j]) * bf2f(ov[j]);
    }
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
__global__ void __launch_bounds__(256, 2) attn_dkdv_kernel(
    const short* __restrict__ dout, const short* __restrict__ q,
    const short* __restrict__ k, const short* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int B, int S, int Hq,
    int Hkv, int D, float scale) {
  // transposed double-buffered staging of the 64-q-row tile:
  // q^T[d][qrow], dO^T[d][qrow]; plus the tile's lse/delta rows
  __shared__ short qt_lds[2][128 * 64];
  __shared__ short dot_lds[2][128 * 64];
  __shared__ float lse_lds[2][64];
  __shared__ float del_lds[2][64];

  const int nkt = (S + 63) / 64;
  const int group = Hq / Hkv;

  // XCD-aware remap: blocks of one (b,hkv) on one XCD (shared Q/dO/K/V L2)
  int b, hkv, kt;
  {
    const int NG = B * Hkv;
    int bid = blockIdx.x;
    int gid, within;
    if ((NG & 7) == 0) {
      gid = (bid >> 3) / nkt * 8 + (bid & 7);
      within = (bid >> 3) % nkt;
    } else {
      gid = bid / nkt;
      within = bid % nkt;
    }
    b = gid / Hkv;
    hkv = gid % Hkv;
    kt = within;
  }

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  const int kv0 = kt * 64 + wid * 16;  // this wave's 16 keys
  const int nd16 = D >> 4, nkc = D >> 5;
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;

  // K/V B-fragments (n = key l15, k = d lg*8+j — contiguous 16 B loads)
  bf16x8 kf[4], vf[4];
  {
    int key = kv0 + l15;
    int keyc = key < S ? key : S - 1;
    const short* kp = kb + (int64_t)keyc * strideS_kv + lg * 8;
    const short* vp = vb + (int64_t)keyc * strideS_kv + lg * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < nkc) {
        kf[c] = *reinterpret_cast<const bf16x8*>(kp + c * 32);
        vf[c] = *reinterpret_cast<const bf16x8*>(vp + c * 32);
      }
  }

  // dK/dV accumulators: C[m=key][n=d] frags per d-tile
  // (lane: d = l15, key = kv0 + lg*4 + r)
  f32x4 dkacc[8], dvacc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    dkacc[i] = {0.f, 0.f, 0.f, 0.f};
    dvacc[i] = {0.f, 0.f, 0.f, 0.f};
  }

  // iteration space: (h in gqa group) x (64-row q tiles >= causal start)
  const int qstart = kt * 64;
  const int nqt = (S - qstart + 63) / 64;
  const int niter = group * nqt;

  // staging: thread stages one q row (qrow = qt + (tid&63)), 16B of d per
  // slot; wave w covers d-slots {2w, 2w+1, 8+2w, 8+2w+1}
  const int srow = tid & 63;
  const int sslot0 = wid * 2;
  auto stage = [&](int it, int buf) {
    const int hh = hkv * group + it / nqt;
    const int qt = qstart + (it % nqt) * 64;
    int qrow = qt + srow;
    if (qrow >= S) qrow = S - 1;
    const short* qp = q + ((int64_t)b * S * Hq + hh) * D + qrow * strideS_q;
    const short* dop =
        dout + ((int64_t)b * S * Hq + hh) * D + qrow * strideS_q;
    short* qdst = qt_lds[buf];
    short* ddst = dot_lds[buf];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        int slot = half * 8 + sslot0 + ss;
        if (slot * 8 >= D) break;
        bf16x8 qv = *reinterpret_cast<const bf16x8*>(qp + slot * 8);
        bf16x8 dv = *reinterpret_cast<const bf16x8*>(dop + slot * 8);
        int d0 = slot * 8;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          qdst[tr_idx(d0 + i, srow)] = qv[i];
          ddst[tr_idx(d0 + i, srow)] = dv[i];
        }
      }
    }
    if (tid < 64) {
      const float* lseb = lse + ((int64_t)b * Hq + hh) * S;
      int rr = qt + tid < S ? qt + tid : S - 1;
      lse_lds[buf][tid] = lseb[rr];
    } else if (tid < 128) {
      const float* delb = delta + ((int64_t)b * Hq + hh) * S;
      int rr = qt + tid - 64 < S ? qt + tid - 64 : S - 1;
      del_lds[buf][tid - 64] = delb[rr];
    }
  };

  stage(0, 0);
  __syncthreads();

  for (int it = 0; it < niter; ++it) {
    const int buf = it & 1;
    if (it + 1 < niter) stage(it + 1, buf ^ 1);

    const int hh = hkv * group + it / nqt;
    const int qt = qstart + (it % nqt) * 64;
    const short* qb = q + ((int64_t)b * S * Hq + hh) * D;
    const short* dob = dout + ((int64_t)b * S * Hq + hh) * D;

    // ---- S[mt], dP[mt]: C[m=qrow(perm)][n=key]; A = Q/dO rows fed in
    // perm16 order (direct global loads), B = kf/vf ----
    f32x4 sfrag[4], dpfrag[4];
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      int qrow = qt + perm16(mt, l15);
      int qrc = qrow < S ? qrow : S - 1;
      const short* qp = qb + (int64_t)qrc * strideS_q + lg * 8;
      const short* dop = dob + (int64_t)qrc * strideS_q + lg * 8;
      f32x4 sa = {0.f, 0.f, 0.f, 0.f}, da = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int c = 0; c < 4; ++c)
        if (c < nkc) {
          bf16x8 qa = *reinterpret_cast<const bf16x8*>(qp + c * 32);
          bf16x8 doa = *reinterpret_cast<const bf16x8*>(dop + c * 32);
          sa = mfma16b(qa, kf[c], sa);
          da = mfma16b(doa, vf[c], da);
        }
      sfrag[mt] = sa;
      dpfrag[mt] = da;
    }

    // ---- P = exp(scale*S - lse), dS = scale*P*(dP - delta); pack into
    // A-operand order (qrow k-dim = kc*32+lg*8+j via perm16) ----
    const int key = kv0 + l15;  // this lane's key (C n-position)
    const bool diag = (qt < kt * 64 + 64) || (qt + 63 >= S);
    uint32_t pk_p[2][4], pk_ds[2][4];
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      const int qoff = cpos16(mt, lg);
      float p[4], ds[4];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qloc = qoff + r;
        const int qrow = qt + qloc;
        float e = (scale * sfrag[mt][r] - lse_lds[buf][qloc]) * LOG2E;
        if (diag && (key > qrow || qrow >= S || key >= S)) e = -INFINITY;
        p[r] = exp2f(e);
        ds[r] = scale * p[r] * (dpfrag[mt][r] - del_lds[buf][qloc]);
      }
      const int kc = mt >> 1, rp = (mt & 1) * 2;
      pk_p[kc][rp + 0] = cvt_pk_bf16(p[0], p[1]);
      pk_p[kc][rp + 1] = cvt_pk_bf16(p[2], p[3]);
      pk_ds[kc][rp + 0] = cvt_pk_bf16(ds[0], ds[1]);
      pk_ds[kc][rp + 1] = cvt_pk_bf16(ds[2], ds[3]);
    }

    // ---- dV += P^T dO ; dK += dS^T Q (B operands from transposed LDS) ----
    const short* qtl = qt_lds[buf];
    const short* dtl = dot_lds[buf];
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd16) break;
      const int d = dt * 16 + l15;
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8 dofr = *reinterpret_cast<const bf16x8*>(
            dtl + tr_idx(d, kc * 32 + lg * 8));
        bf16x8 qfr = *reinterpret_cast<const bf16x8*>(
            qtl + tr_idx(d, kc * 32 + lg * 8));
        dvacc[dt] = mfma16b(*reinterpret_cast<const bf16x8*>(&pk_p[kc][0]),
                            dofr, dvacc[dt]);
        dkacc[dt] = mfma16b(*reinterpret_cast<const bf16x8*>(&pk_ds[kc][0]),
                            qfr, dkacc[dt]);
      }
    }
    __syncthreads();
  }

  // ---- store dK/dV (exclusive: one block per (b,hkv,key)) ----
  // C layout: lane d = l15 (per dt), key = kv0 + lg*4 + r
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int keyw = kv0 + lg * 4 + r;
    if (keyw >= S) continue;
    short* dkp = dk + ((int64_t)b * S * Hkv + (int64_t)keyw * Hkv + hkv) * D;
    short* dvp = dv + ((int64_t)b * S * Hkv + (int64_t)keyw * Hkv + hkv) * D;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd16) break;
      dkp[dt * 16 + l15] = f2bf(dkacc[dt][r]);
      dvp[dt * 16 + l15] = f2bf(dvacc[dt][r]);
    }
  }
}

// ---------------- dq ----------------
__global__ void __launch_bounds__(256, 2) attn_dq_kernel(
    const short* __restrict__ dout, const short* __restrict__ q,
    const short* __restrict__ k, const short* __restrict__ v,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int B, int S, int Hq, int Hkv, int D,
    float scale) {
  // transposed double-buffered K^T[d][key] staging
  __shared__ short kt_lds[2][128 * 64];

  const int ntq = (S + 127) / 128;
  const int gqa = Hq / Hkv;
  const int bpg = gqa * ntq;

  int b, h, qtile;
  {
    const int NG = B * Hkv;
    int bid = blockIdx.x;
    int gid, within;
    if ((NG & 7) == 0) {
      gid = (bid >> 3) / bpg * 8 + (bid & 7);
      within = (bid >> 3) % bpg;
    } else {
      gid = bid / bpg;
      within = bid % bpg;
    }
    b = gid / Hkv;
    const int hkv0 = gid % Hkv;
    h = hkv0 * gqa + within / ntq;
    qtile = within % ntq;
  }
  const int hkv = h / gqa;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  // this wave's 32 q rows as two 16-row groups interleaved across the
  // block (rows w*16 and w*16+64) — causal activity is then uniform
  // across waves (no per-tile barrier idling); see attention_fwd.hip
  const int rowb[2] = {qtile * 128 + wid * 16, qtile * 128 + 64 + wid * 16};
  const int nd16 = D >> 4, nkc = D >> 5;
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* qb = q + ((int64_t)b * S * Hq + h) * D;
  const short* dob = dout + ((int64_t)b * S * Hq + h) * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;
  const float* lseb = lse + ((int64_t)b * Hq + h) * S;
  const float* delb = delta + ((int64_t)b * Hq + h) * S;

  // Q / dO B-fragments (n = qrow = l15 per nq tile) + per-lane lse/delta
  bf16x8 qf[2][4], dof[2][4];
  float lse_r[2], del_r[2];
#pragma unroll
  for (int nq = 0; nq < 2; ++nq) {
    int qrow = rowb[nq] + l15;
    int qrc = qrow < S ? qrow : S - 1;
    const short* qp = qb + (int64_t)qrc * strideS_q + lg * 8;
    const short* dop = dob + (int64_t)qrc * strideS_q + lg * 8;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      if (c < nkc) {
        qf[nq][c] = *reinterpret_cast<const bf16x8*>(qp + c * 32);
        dof[nq][c] = *reinterpret_cast<const bf16x8*>(dop + c * 32);
      }
    lse_r[nq] = lseb[qrc];
    del_r[nq] = delb[qrc];
  }

  // dQ accumulators: C[m=qrow][n=d] frags (lane: d = l15, qrow = lg*4+r)
  f32x4 dqacc[8][2];
#pragma unroll
  for (int dt = 0; dt < 8; ++dt)
#pragma unroll
    for (int nq = 0; nq < 2; ++nq) dqacc[dt][nq] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = min(S, qtile * 128 + 128);
  const int ntiles = (kv_end + 63) / 64;

  const int srow = tid & 63;
  const int sslot0 = wid * 2;
  auto stage_k = [&](int kv0s, int buf) {
    int keyg = kv0s + srow;
    if (keyg >= S) keyg = S - 1;
    const short* kp = kb + (int64_t)keyg * strideS_kv;
    short* dst = kt_lds[buf];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
#pragma unroll
      for (int ss = 0; ss < 2; ++ss) {
        int slot = half * 8 + sslot0 + ss;
        if (slot * 8 >= D) break;
        bf16x8 vec = *reinterpret_cast<const bf16x8*>(kp + slot * 8);
        int d0 = slot * 8;
#pragma unroll
        for (int i = 0; i < 8; ++i) dst[tr_idx(d0 + i, srow)] = vec[i];
      }
    }
  };

  stage_k(0, 0);
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * 64;
    const int buf = t & 1;
    if (t + 1 < ntiles) stage_k((t + 1) * 64, buf ^ 1);

    // group 0 (lower rows) drops out at the final diagonal tiles; group 1
    // is live for every tile — uniform across waves (no barrier idling)
    const bool act0 = kv0 <= rowb[0] + 15;
    {
      // ---- per mt: S^T = mfma(K_perm, Q), dP^T = mfma(V_perm, dO), then
      // immediately exp/pack dS into the dQ MFMA's A operand (keys in
      // kc*32+lg*8+j order via perm16) — keeps only one mt of S/dP live ----
      const bool smask = (kv_end < kv0 + 64);
      uint32_t pk_ds[2][2][4];  // [nq][kc][4]
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        int keyg = kv0 + perm16(mt, l15);
        if (keyg >= S) keyg = S - 1;
        const short* kp = kb + (int64_t)keyg * strideS_kv + lg * 8;
        const short* vp = vb + (int64_t)keyg * strideS_kv + lg * 8;
        f32x4 s0 = {0.f, 0.f, 0.f, 0.f}, s1 = {0.f, 0.f, 0.f, 0.f};
        f32x4 d0 = {0.f, 0.f, 0.f, 0.f}, d1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int c = 0; c < 4; ++c)
          if (c < nkc) {
            bf16x8 ka = *reinterpret_cast<const bf16x8*>(kp + c * 32);
            bf16x8 va = *reinterpret_cast<const bf16x8*>(vp + c * 32);
            if (act0) {
              s0 = mfma16b(ka, qf[0][c], s0);
              d0 = mfma16b(va, dof[0][c], d0);
            }
            s1 = mfma16b(ka, qf[1][c], s1);
            d1 = mfma16b(va, dof[1][c], d1);
          }
        const int koff = cpos16(mt, lg);
        const int kc = mt >> 1, rp = (mt & 1) * 2;
#pragma unroll
        for (int nq = 0; nq < 2; ++nq) {
          if (nq == 0 && !act0) continue;
          const int qrow = rowb[nq] + l15;
          const bool diag = (kv0 + 63 > rowb[nq]) || smask;
          const f32x4 sv = nq ? s1 : s0;
          const f32x4 dv = nq ? d1 : d0;
          float ds[4];
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int keyr = kv0 + koff + r;
            float e = (scale * sv[r] - lse_r[nq]) * LOG2E;
            if (diag && (keyr > qrow || keyr >= S || qrow >= S))
              e = -INFINITY;
            float p = exp2f(e);
            ds[r] = scale * p * (dv[r] - del_r[nq]);
          }
          pk_ds[nq][kc][rp + 0] = cvt_pk_bf16(ds[0], ds[1]);
          pk_ds[nq][kc][rp + 1] = cvt_pk_bf16(ds[2], ds[3]);
        }
      }

      // ---- dQ += dS K (B = K^T-staged rows back in [key][d]... B operand
      // [k=key][n=d] read from kt_lds transposed image) ----
      const short* ktl = kt_lds[buf];
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        if (dt >= nd16) break;
        const int d = dt * 16 + l15;
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          bf16x8 kfr = *reinterpret_cast<const bf16x8*>(
              ktl + tr_idx(d, kc * 32 + lg * 8));
          if (act0)
            dqacc[dt][0] = mfma16b(
                *reinterpret_cast<const bf16x8*>(&pk_ds[0][kc][0]), kfr,
                dqacc[dt][0]);
          dqacc[dt][1] = mfma16b(
              *reinterpret_cast<const bf16x8*>(&pk_ds[1][kc][0]), kfr,
              dqacc[dt][1]);
        }
      }
    }
    __syncthreads();
  }

  // ---- store dQ: lane d = l15 (per dt), qrow = q0 + nq*16 + lg*4 + r ----
#pragma unroll
  for (int nq = 0; nq < 2; ++nq) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int qrow = rowb[nq] + lg * 4 + r;
      if (qrow >= S) continue;
      short* dqp = dq + ((int64_t)b * S * Hq + (int64_t)qrow * Hq + h) * D;
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        if (dt >= nd16) break;
        dqp[dt * 16 + l15] = f2bf(dqacc[dt][nq][r]);
      }
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
  int ntq = (S + 127) / 128;
  hipLaunchKernelGGL(attn_dq_kernel, dim3((uint32_t)((int64_t)B * Hq * ntq)),
                     dim3(256), 0, stream, (const short*)dout, (const short*)q,
                     (const short*)k, (const short*)v, lse, delta, (short*)dq,
                     B, S, Hq, Hkv, D, scale);
}
}
