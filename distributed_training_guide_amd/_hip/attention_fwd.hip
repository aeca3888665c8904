// Causal flash-attention forward for gfx950 — v5: swapped-operand S^T MFMA
// with in-register P, double-buffered LDS K/V tiles prefetched a full tile
// ahead (T14), one barrier per tile.
//
// Replaces the reference's flash-attn-2 dependency
// (attn_implementation="flash_attention_2", 05:93 / 06:73 / 07:71 —
// SURVEY.md §2b "Flash attention 2").
//
// Layout: q [B,S,Hq,D], k/v [B,S,Hkv,D] bf16 contiguous (BSHD). GQA via
// Hq % Hkv == 0. Causal always. Saves lse [B,Hq,S] f32 for the backward.
//
// v3 design (guide §5.5 / §5 "common mistakes"):
//   * S^T = mfma(A=K, B=Q): the C fragment then holds qrow = lane&15 —
//     softmax rows are (mostly) lane-local: 16 values in registers + two
//     shfl_xor steps, instead of 16-lane tree reductions per row.
//   * Key-permuted A fragments: K rows are fed to the QK^T MFMA in an
//     order chosen so the C-fragment's per-lane key positions equal the
//     B-operand layout that P^T needs for the P·V MFMA. P therefore never
//     leaves registers: exp -> v_cvt_pk_bf16_f32 -> B fragment. No P LDS
//     round-trip, no block barrier between QK^T and P·V.
//   * K and V are staged into double-buffered LDS, with the global loads
//     for tile t+1 issued at the TOP of tile t's compute: the ~200-900cy
//     global latency lands behind a full tile of MFMA (T14), and the
//     A-fragment reads become ~50cy ds_read_b128 the compiler interleaves
//     with the MFMA stream (PMC before this change: 67% SQ_WAIT_ANY on the
//     direct-global K path). K image row-major 16B-slot-swizzled
//     (conflict-free writes AND reads); the XCD-aware remap keeps a
//     (b,kv-head) group's K/V on one XCD's L2.
//   * V is staged TRANSPOSED into double-buffered LDS (v^T[d][key], key
//     index XOR-swizzled): the P·V MFMA's A operand needs 8 consecutive
//     keys per lane, which only a transposed image can serve as
//     ds_read_b128. One barrier per KV tile (ping-pong buffers).
//   * O^T accumulates in C fragments (qrow = lane&15 throughout), scale
//     folded into exp2f, epilogue writes 4 consecutive d per lane (b64).
//
// MFMA fragment maps (gfx950 v_mfma_f32_16x16x32_bf16):
//   A[m][k]: lane l holds m = l&15, k = (l>>4)*8 + j (j=0..7)
//   B[k][n]: lane l holds n = l&15, k = (l>>4)*8 + j
//   C/D     : lane l holds n = l&15, m = (l>>4)*4 + r (r=0..3)
#include "common.h"

using bf16x8 = s16x8;
typedef __attribute__((ext_vector_type(2))) short s16x2;
typedef __attribute__((ext_vector_type(4))) short s16x4v;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

#define QBLK 128   // q rows per block (32 per wave)
#define KVBLK 64   // keys per LDS tile
#define LOG2E 1.4426950408889634f

__global__ void __launch_bounds__(256, 2) attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int B, int S, int Hq, int Hkv, int D,
    float scale) {
  // both tiles in the 16-col-subtile image (st_idx; conflict-free tr16
  // reads + b128 row reads), double-buffered
  __shared__ short k_lds[2][KVBLK * 128];
  __shared__ short v_lds[2][KVBLK * 128];

  const int ntq = (S + QBLK - 1) / QBLK;
  const int gqa = Hq / Hkv;
  const int bpg = gqa * ntq;  // blocks per (b, hkv) group

  // XCD-aware remap (guide T1): all blocks of one (b,hkv) group land on one
  // XCD so its K/V stay in that XCD's L2. Dispatcher places block i on XCD
  // i%8 (performance only, never correctness).
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
    const int hkv = gid % Hkv;
    h = hkv * gqa + within / ntq;
    qtile = within % ntq;
  }
  const int hkv = h / gqa;

  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int lane = tid & (WAVE - 1);
  const int l15 = lane & 15;
  const int lg = lane >> 4;

  // this wave's 32 q rows, as two 16-row groups INTERLEAVED across the
  // block (rows w*16 and w*16+64): every wave then touches both halves of
  // the q range, so causal tile-skipping is uniform across waves and no
  // wave idles at the per-tile barrier (guide: wave load balance)
  const int rowb[2] = {qtile * QBLK + wid * 16, qtile * QBLK + 64 + wid * 16};
  const int nd16 = D >> 4;                 // 16-d tiles (8 for D=128)
  const int nkc = D >> 5;                  // 32-d MFMA K chunks
  const int64_t strideS_q = (int64_t)Hq * D;
  const int64_t strideS_kv = (int64_t)Hkv * D;
  const short* qb = q + ((int64_t)b * S * Hq + h) * D;
  const short* kb = k + ((int64_t)b * S * Hkv + hkv) * D;
  const short* vb = v + ((int64_t)b * S * Hkv + hkv) * D;

  // ---- Q B-fragments: [nq 2][kc], row q0 + nq*16 + l15, d = kc*32+lg*8 --
  bf16x8 qf[2][4];
#pragma unroll
  for (int nq = 0; nq < 2; ++nq) {
    int qrow = rowb[nq] + l15;
    if (qrow >= S) qrow = S - 1;
    const short* qp = qb + (int64_t)qrow * strideS_q + lg * 8;
#pragma unroll
    for (int kc = 0; kc < 4; ++kc)
      if (kc < nkc)
        qf[nq][kc] = *reinterpret_cast<const bf16x8*>(qp + kc * 32);
  }

  // key permutation: C-fragment m-position p (= mt*16 + lg*4 + r) must hold
  // key (mt>>1)*32 + lg*8 + (mt&1)*4 + r, so the packed P rows are already
  // the B-operand key order (kc*32 + lg*8 + j). A-operand lane position is
  // l15 = lg_c*4 + r of the C tile.
  int kperm[4];  // key offset of A-fragment row l15 in tile mt
#pragma unroll
  for (int mt = 0; mt < 4; ++mt)
    kperm[mt] = (mt >> 1) * 32 + (l15 >> 2) * 8 + (mt & 1) * 4 + (l15 & 3);
  // per-lane key offsets of the C fragment (for causal masking):
  // key(mt, r) = (mt>>1)*32 + lg*8 + (mt&1)*4 + r
  int ckey[4];
#pragma unroll
  for (int mt = 0; mt < 4; ++mt)
    ckey[mt] = (mt >> 1) * 32 + lg * 8 + (mt & 1) * 4;

  // O^T accumulators: [d tile][nq] C frags; lane: qrow=l15, d=lg*4+r
  f32x4 oacc[8][2];
#pragma unroll
  for (int dt = 0; dt < 8; ++dt)
#pragma unroll
    for (int nq = 0; nq < 2; ++nq) oacc[dt][nq] = {0.f, 0.f, 0.f, 0.f};
  float mrow[2] = {-INFINITY, -INFINITY};
  float lrow[2] = {0.f, 0.f};

  const int kv_end = min(S, qtile * QBLK + QBLK);
  const int ntiles = (kv_end + KVBLK - 1) / KVBLK;
  const float c = scale * LOG2E;
  (void)hkv;

  // ---- staging via LDS-DMA (global_load_lds, probed: data lands at
  // lds_base + lane*16 with per-lane global sources — tools/gldsprobe.hip):
  // no staging registers, no ds_write pass, no vmcnt park.  The per-lane
  // global slot is pre-XOR'd so the contiguous landing equals the
  // st_idx subtile image; tile t+1's DMA issues at the top of tile t and
  // completes under a full tile of MFMA before the end-of-tile barrier.
  const int nslot = D >> 3;
  const int ncw = nslot >> 2;     // chunks per wave per tensor
  const int srow_half = lane >> 1;       // subtile-half row within chunk
  const int scol_half = (lane & 1) * 8;  // 8-col half within subtile
  auto stage_kv = [&](int kv0, int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i >= ncw) break;
      const int ch = wid * ncw + i;
      const int key = (ch & 1) * 32 + srow_half;
      int kg = kv0 + key;
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

  stage_kv(0, 0);
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    const int kv0 = t * KVBLK;
    const int buf = t & 1;
    // stage next tile into the other buffer (global loads issue before the
    // MFMA stream — T14 spirit; ds_writes don't touch the compute buffer)
    if (t + 1 < ntiles) stage_kv((t + 1) * KVBLK, buf ^ 1);

    // Both 16-row groups are computed unconditionally on every tile: an
    // `if (act0)`-guarded MFMA accumulation (skipping group 0 on its
    // fully-masked diagonal tiles) miscompiled in the dq kernel of
    // attention_bwd.hip (sparse wrong accumulator values; hipcc/ROCm 7.2)
    // — the causal mask yields p=0 / alpha=1 there, so skipping is a
    // ~3% optimization not worth the hazard.

    // ---- S^T = mfma(K, Q): [mt 4][nq 2] C frags; K A-frags from the
    // row-major staged tile (ds_read_b128, ~50cyc, hidden by MFMA) ----
    const short* kl = k_lds[buf];
    f32x4 sfrag[4][2];
    __builtin_amdgcn_s_setprio(1);  // T5: prioritize the QK^T MFMA stream
#pragma unroll
    for (int mth = 0; mth < 2; ++mth) {
      // two key-tiles at once -> 4 independent accumulator chains (the
      // dependent-accumulator MFMA latency exceeds the issue rate; a
      // 2-chain version left the pipe half idle)
      const int kra = kperm[2 * mth], krb = kperm[2 * mth + 1];
      f32x4 a0 = {0.f, 0.f, 0.f, 0.f}, a1 = {0.f, 0.f, 0.f, 0.f};
      f32x4 b0 = {0.f, 0.f, 0.f, 0.f}, b1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < 4; ++kc)
        if (kc < nkc) {
          bf16x8 kfa = *reinterpret_cast<const bf16x8*>(
              kl + st_idx(kra, kc * 32 + lg * 8));
          bf16x8 kfb = *reinterpret_cast<const bf16x8*>(
              kl + st_idx(krb, kc * 32 + lg * 8));
          a0 = mfma16(kfa, qf[0][kc], a0);
          b0 = mfma16(kfb, qf[0][kc], b0);
          a1 = mfma16(kfa, qf[1][kc], a1);
          b1 = mfma16(kfb, qf[1][kc], b1);
        }
      sfrag[2 * mth][0] = a0;
      sfrag[2 * mth][1] = a1;
      sfrag[2 * mth + 1][0] = b0;
      sfrag[2 * mth + 1][1] = b1;
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- causal mask + online softmax (rows lane-local) ----
    u32x4 pk[2][2];  // [nq][kc]: 4 regs of 2 bf16 (keys j=0..7)
    float alpha[2];
#pragma unroll
    for (int nq = 0; nq < 2; ++nq) {
      const int qrow = rowb[nq] + l15;
      const bool need_mask =
          (kv0 + KVBLK - 1) > rowb[nq] || kv_end < kv0 + KVBLK;
      if (need_mask) {
#pragma unroll
        for (int mt = 0; mt < 4; ++mt)
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            int key = kv0 + ckey[mt] + r;
            if (key > qrow || key >= S) sfrag[mt][nq][r] = -INFINITY;
          }
      }
      float tmax = -INFINITY;
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        f32x4 s = sfrag[mt][nq];
        tmax = fmaxf(tmax, fmaxf(fmaxf(s[0], s[1]), fmaxf(s[2], s[3])));
      }
      // row spread over lanes l15, l15+16, l15+32, l15+48
      tmax = fmaxf(tmax, __shfl_xor(tmax, 16, WAVE));
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
      float mnew = fmaxf(mrow[nq], tmax);
      alpha[nq] =
          (mrow[nq] == -INFINITY) ? 0.0f : exp2f((mrow[nq] - mnew) * c);
      mrow[nq] = mnew;
      float psum = 0.f;
#pragma unroll
      for (int mt = 0; mt < 4; ++mt) {
        f32x4 s = sfrag[mt][nq];
        f32x4 p;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          p[r] = (s[r] == -INFINITY) ? 0.0f : exp2f((s[r] - mnew) * c);
          psum += p[r];
        }
        // keys of this C tile are already in B-operand order: mt -> (kc =
        // mt>>1, reg pair = (mt&1)*2 + {0,1})
        pk[nq][mt >> 1][(mt & 1) * 2 + 0] = cvt_pk_bf16(p[0], p[1]);
        pk[nq][mt >> 1][(mt & 1) * 2 + 1] = cvt_pk_bf16(p[2], p[3]);
      }
      psum += __shfl_xor(psum, 16, WAVE);
      psum += __shfl_xor(psum, 32, WAVE);
      lrow[nq] = lrow[nq] * alpha[nq] + psum;
    }

    // ---- O^T += mfma(V^T, P^T): V^T A-frags via tr16 hardware-transpose
    // reads from the row-major V tile (no transposed copy) ----
    const short* vt = v_lds[buf];
    __builtin_amdgcn_s_setprio(1);  // T5: prioritize the P.V MFMA stream
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd16) break;
      bf16x8 va[2];
#pragma unroll
      for (int kc = 0; kc < 2; ++kc)
        va[kc] = tr16_frag_st(vt, kc * 32 + lg * 8, dt * 16, l15);
#pragma unroll
      for (int nq = 0; nq < 2; ++nq) {
        f32x4 acc = oacc[dt][nq];
#pragma unroll
        for (int r = 0; r < 4; ++r) acc[r] *= alpha[nq];
        acc = mfma16(va[0], __builtin_bit_cast(bf16x8, pk[nq][0]), acc);
        acc = mfma16(va[1], __builtin_bit_cast(bf16x8, pk[nq][1]), acc);
        oacc[dt][nq] = acc;
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // ---- epilogue: lane holds qrow = l15, d = dt*16 + lg*4 + r ----
#pragma unroll
  for (int nq = 0; nq < 2; ++nq) {
    const int qrow = rowb[nq] + l15;
    if (qrow >= S) continue;
    const float invl = (lrow[nq] > 0.f) ? 1.0f / lrow[nq] : 0.0f;
    short* op = o + ((int64_t)b * S * Hq + (int64_t)qrow * Hq + h) * D;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt) {
      if (dt >= nd16) break;
      s16x4v outv;
#pragma unroll
      for (int r = 0; r < 4; ++r) outv[r] = f2bf(oacc[dt][nq][r] * invl);
      *reinterpret_cast<s16x4v*>(op + dt * 16 + lg * 4) = outv;
    }
    if (lg == 0)
      lse_out[((int64_t)b * Hq + h) * S + qrow] =
          mrow[nq] * scale + __logf(fmaxf(lrow[nq], 1e-30f));
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
