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
// one row per 16-lane group (16 lanes x 8 bf16 = 128 elems = a whole
// D=128 row per pass) — a row-per-wave version left 48/64 lanes idle at
// D<=128 (guide §5 mistake 6)
__global__ void __launch_bounds__(256) attn_delta_kernel(
    const short* __restrict__ dout, const short* __restrict__ o,
    float* __restrict__ delta, int B, int S, int Hq, int D) {
  const int grp = threadIdx.x >> 4;       // 16 groups per block
  const int gl = threadIdx.x & 15;
  int64_t nrows = (int64_t)B * S * Hq;
  for (int64_t row = (int64_t)blockIdx.x * 16 + grp; row < nrows;
       row += (int64_t)gridDim.x * 16) {
    const short* dp = dout + row * D;
    const short* op = o + row * D;
    float s = 0.f;
    for (int i = gl * 8; i < D; i += 16 * 8) {
      bf16x8 dv = *reinterpret_cast<const bf16x8*>(dp + i);
      bf16x8 ov = *reinterpret_cast<const bf16x8*>(op + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bf2f(dv[j]) * bf2f(ov[j]);
    }
#pragma unroll
    for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, WAVE);
    if (gl == 0) {
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
  // double-buffered ROW-major staging of the 64-q-row tile (q[qrow][d],
  // dO[qrow][d], st_idx subtile image): serves BOTH the Q/dO A-fragments
  // (b128 row reads, replacing per-tile global gathers) and the
  // dV/dK B-fragments (tr16 hardware-transpose reads); plus lse/delta rows
  __shared__ short q_lds[2][64 * 128];
  __shared__ short do_lds[2][64 * 128];
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

  // staging via LDS-DMA (st_idx subtile image;
  // conflict-free), one (h, q-tile) AHEAD so global latency hides behind
  // a full tile of MFMA
  // LDS-DMA staging (global_load_lds; see attention_fwd.hip): the next
  // (h, q-tile)'s Q/dO tiles stream into the other buffer with no staging
  // registers and no vmcnt park; the per-lane global slot is pre-XOR'd so
  // the contiguous landing equals the st_idx subtile image.
  const int nslot = D >> 3;
  const int ncw = nslot >> 2;
  const int srow_half = lane >> 1;       // subtile-half row within chunk
  const int scol_half = (lane & 1) * 8;  // 8-col half within subtile
  auto stage_qdo = [&](int it, int buf) {
    const int hh = hkv * group + it / nqt;
    const int qt = qstart + (it % nqt) * 64;
    const short* qbse = q + ((int64_t)b * S * Hq + hh) * D;
    const short* dbse = dout + ((int64_t)b * S * Hq + hh) * D;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i >= ncw) break;
      const int ch = wid * ncw + i;
      const int row = (ch & 1) * 32 + srow_half;
      int r = qt + row;
      if (r >= S) r = S - 1;
      const int64_t goff =
          (int64_t)r * strideS_q + (ch >> 1) * 16 + scol_half;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(qbse +
                                                                  goff),
          (__attribute__((address_space(3))) unsigned int*)(q_lds[buf] +
                                                            ch * 512),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(dbse +
                                                                  goff),
          (__attribute__((address_space(3))) unsigned int*)(do_lds[buf] +
                                                            ch * 512),
          16, 0, 0);
    }
  };
  auto stage_lse = [&](int it, int buf) {
    const int hh = hkv * group + it / nqt;
    const int qt = qstart + (it % nqt) * 64;
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

  stage_qdo(0, 0);
  stage_lse(0, 0);
  __syncthreads();

  for (int it = 0; it < niter; ++it) {
    const int buf = it & 1;
    if (it + 1 < niter) {
      stage_qdo(it + 1, buf ^ 1);
      stage_lse(it + 1, buf ^ 1);
    }

    const int qt = qstart + (it % nqt) * 64;
    const short* ql = q_lds[buf];
    const short* dol = do_lds[buf];

    // ---- per mt: S/dP MFMAs (C[m=qrow(perm)][n=key]; A = Q/dO rows in
    // perm16 order from LDS, B = kf/vf), then immediately
    // P = exp(scale*S - lse), dS = scale*P*(dP - delta), packed into
    // A-operand order — only one mt of S/dP state stays live ----
    const int key = kv0 + l15;  // this lane's key (C n-position)
    const bool diag = (qt < kt * 64 + 64) || (qt + 63 >= S);
    u32x4 pk_p[2], pk_ds[2];
    __builtin_amdgcn_s_setprio(1);  // T5: prioritize the MFMA stream
#pragma unroll
    for (int mth = 0; mth < 2; ++mth) {
      // two q-row tiles at once -> 4 independent MFMA accumulator chains
      // (dependent-accumulator latency > issue rate with only 2)
      const int qra = perm16(2 * mth, l15), qrb = perm16(2 * mth + 1, l15);
      f32x4 sa0 = {0.f, 0.f, 0.f, 0.f}, da0 = {0.f, 0.f, 0.f, 0.f};
      f32x4 sa1 = {0.f, 0.f, 0.f, 0.f}, da1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int c = 0; c < 4; ++c)
        if (c < nkc) {
          bf16x8 qa0 = *reinterpret_cast<const bf16x8*>(
              ql + st_idx(qra, c * 32 + lg * 8));
          bf16x8 qa1 = *reinterpret_cast<const bf16x8*>(
              ql + st_idx(qrb, c * 32 + lg * 8));
          bf16x8 do0 = *reinterpret_cast<const bf16x8*>(
              dol + st_idx(qra, c * 32 + lg * 8));
          bf16x8 do1 = *reinterpret_cast<const bf16x8*>(
              dol + st_idx(qrb, c * 32 + lg * 8));
          sa0 = mfma16b(qa0, kf[c], sa0);
          sa1 = mfma16b(qa1, kf[c], sa1);
          da0 = mfma16b(do0, vf[c], da0);
          da1 = mfma16b(do1, vf[c], da1);
        }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int mt = 2 * mth + half;
        const f32x4 sa = half ? sa1 : sa0;
        const f32x4 da = half ? da1 : da0;
        const int qoff = cpos16(mt, lg);
        float p[4], ds[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int qloc = qoff + r;
          const int qrow = qt + qloc;
          float e = (scale * sa[r] - lse_lds[buf][qloc]) * LOG2E;
          if (diag && (key > qrow || qrow >= S || key >= S)) e = -INFINITY;
          p[r] = exp2f(e);
          ds[r] = scale * p[r] * (da[r] - del_lds[buf][qloc]);
        }
        const int kc = mt >> 1, rp = (mt & 1) * 2;
        pk_p[kc][rp + 0] = cvt_pk_bf16(p[0], p[1]);
        pk_p[kc][rp + 1] = cvt_pk_bf16(p[2], p[3]);
        pk_ds[kc][rp + 0] = cvt_pk_bf16(ds[0], ds[1]);
        pk_ds[kc][rp + 1] = cvt_pk_bf16(ds[2], ds[3]);
      }
    }

    // ---- dV += P^T dO ; dK += dS^T Q (B operands via tr16 transpose
    // reads from the same row-major tiles) ----
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd16) break;
#pragma unroll
      for (int kc = 0; kc < 2; ++kc) {
        bf16x8 dofr = tr16_frag_st(dol, kc * 32 + lg * 8, dt * 16, l15);
        bf16x8 qfr = tr16_frag_st(ql, kc * 32 + lg * 8, dt * 16, l15);
        dvacc[dt] = mfma16b(__builtin_bit_cast(bf16x8, pk_p[kc]), dofr,
                            dvacc[dt]);
        dkacc[dt] = mfma16b(__builtin_bit_cast(bf16x8, pk_ds[kc]), qfr,
                            dkacc[dt]);
      }
    }
    __builtin_amdgcn_s_setprio(0);
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
  // double-buffered K and V tiles (st_idx subtile image): K serves
  // the S^T A-fragments (b128 row reads) AND the dQ B-fragments (tr16
  // transpose reads); V serves the dP^T A-fragments
  __shared__ short k_lds[2][64 * 128];
  __shared__ short v_lds[2][64 * 128];

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

  // LDS-DMA staging (global_load_lds; see attention_fwd.hip)
  const int nslot = D >> 3;
  const int ncw = nslot >> 2;
  const int srow_half = lane >> 1;       // subtile-half row within chunk
  const int scol_half = (lane & 1) * 8;  // 8-col half within subtile
  auto stage_k = [&](int kv0s, int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i >= ncw) break;
      const int ch = wid * ncw + i;
      const int key = (ch & 1) * 32 + srow_half;
      int kg = kv0s + key;
      if (kg >= S) kg = S - 1;
      const int64_t goff =
          (int64_t)kg * strideS_kv + (ch >> 1) * 16 + scol_half;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(kb + goff),
          (__attribute__((address_space(3))) unsigned int*)(k_lds[buf] +
                                                            ch * 512),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)(vb + goff),
          (__attribute__((address_space(3))) unsigned int*)(v_lds[buf] +
                                                            ch * 512),
          16, 0, 0);
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
    // NOTE: an `if (act0)`-guarded MFMA accumulation here (skipping the
    // lower 16-row group on its fully-masked diagonal tiles) miscompiled —
    // sparse wrong dqacc[0][0] values on waves 2/3 (hipcc/ROCm 7.2, wave-
    // uniform condition derived from wid). Both groups are computed
    // unconditionally; the causal mask already zeroes dS of masked tiles.
    {
      // ---- per mt: S^T = mfma(K_perm, Q), dP^T = mfma(V_perm, dO), then
      // immediately exp/pack dS into the dQ MFMA's A operand (keys in
      // kc*32+lg*8+j order via perm16) — keeps only one mt of S/dP live ----
      const bool smask = (kv_end < kv0 + 64);
      u32x4 pk_ds[2][2];  // [nq][kc]
      const short* kl = k_lds[buf];
      const short* vl = v_lds[buf];
      __builtin_amdgcn_s_setprio(1);  // T5: prioritize the MFMA stream
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        const int kr = perm16(mt, l15);  // local key row
        f32x4 s0 = {0.f, 0.f, 0.f, 0.f}, s1 = {0.f, 0.f, 0.f, 0.f};
        f32x4 d0 = {0.f, 0.f, 0.f, 0.f}, d1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int c = 0; c < 4; ++c)
          if (c < nkc) {
            bf16x8 ka = *reinterpret_cast<const bf16x8*>(
                kl + st_idx(kr, c * 32 + lg * 8));
            bf16x8 va = *reinterpret_cast<const bf16x8*>(
                vl + st_idx(kr, c * 32 + lg * 8));
            s0 = mfma16b(ka, qf[0][c], s0);
            d0 = mfma16b(va, dof[0][c], d0);
            s1 = mfma16b(ka, qf[1][c], s1);
            d1 = mfma16b(va, dof[1][c], d1);
          }
        const int koff = cpos16(mt, lg);
        const int kc = mt >> 1, rp = (mt & 1) * 2;
#pragma unroll
        for (int nq = 0; nq < 2; ++nq) {
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

      // ---- dQ += dS K: B operand [k=key][n=d] via tr16 transpose reads
      // from the row-major K tile ----
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        if (dt >= nd16) break;
#pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
#ifdef DQ_SCALAR_B
          bf16x8 kfr;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            kfr[j] = kl[st_idx(kc * 32 + lg * 8 + j, dt * 16 + l15)];
#else
          bf16x8 kfr = tr16_frag_st(kl, kc * 32 + lg * 8, dt * 16, l15);
#endif
          dqacc[dt][0] = mfma16b(__builtin_bit_cast(bf16x8, pk_ds[0][kc]),
                                 kfr, dqacc[dt][0]);
          dqacc[dt][1] = mfma16b(__builtin_bit_cast(bf16x8, pk_ds[1][kc]),
                                 kfr, dqacc[dt][1]);
        }
      }
      __builtin_amdgcn_s_setprio(0);
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
  int64_t dwant = (nrows + 15) / 16;
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
