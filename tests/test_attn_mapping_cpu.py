"""Lane-exact CPU simulation of attn_dq_kernel's fragment/index math
(perm16, cpos16, rm_idx swizzle, tr16 transpose-read mapping, causal
masking, interleaved row groups) against a numpy reference — guards the
HIP kernels' mapping logic without a GPU (the tr16 lane semantics were
verified on hardware: profiles/tr16_probe_semantics.log)."""
import numpy as np

np.random.seed(0)
S, D = 128, 64
scale = D ** -0.5
Q = np.random.randn(S, D).astype(np.float32)
K = np.random.randn(S, D).astype(np.float32)
V = np.random.randn(S, D).astype(np.float32)
dO = np.random.randn(S, D).astype(np.float32)

# reference
Sr = Q @ K.T * scale
mask = np.tril(np.ones((S, S), bool))
Sr[~mask] = -np.inf
P = np.exp(Sr - Sr.max(1, keepdims=True))
P /= P.sum(1, keepdims=True)
lse_ref = np.log(np.exp(Sr - Sr.max(1, keepdims=True)).sum(1)) + Sr.max(1)
O = P @ V
delta = (dO * O).sum(1)
dP = dO @ V.T
dS = P * (dP - delta[:, None]) * scale
dQ_ref = dS @ K

nkc = D // 32
nd16 = D // 16
nslot1 = D // 8 - 1


def perm16(mt, l15):
    return (mt >> 1) * 32 + (l15 >> 2) * 8 + (mt & 1) * 4 + (l15 & 3)


def cpos16(mt, lg):
    return (mt >> 1) * 32 + lg * 8 + (mt & 1) * 4


def rm_idx(row, col):
    # st_idx: 16-col-subtile image (matches _hip/common.h st_idx)
    return ((col >> 4) << 10) + (row << 4) + (col & 15)


def mfma(A, B, C):
    # A[m][k] 16x32, B[k][n] 32x16, C[m][n] 16x16
    return C + A @ B


# LDS staging (logical image via rm_idx)
def stage(kv0):
    lds_k = np.zeros(64 * D, np.float32)
    lds_v = np.zeros(64 * D, np.float32)
    for key in range(64):
        kg = min(kv0 + key, S - 1)
        for slot in range(D // 8):
            for i in range(8):
                lds_k[rm_idx(key, slot * 8 + i)] = K[kg, slot * 8 + i]
                lds_v[rm_idx(key, slot * 8 + i)] = V[kg, slot * 8 + i]
    return lds_k, lds_v


def tr16_frag(lds, row0, col0, l15):
    # output lane l15 gets rows row0..row0+7 at column col0+l15
    return np.array([lds[rm_idx(row0 + j, col0 + l15)] for j in range(8)])


dQ = np.zeros((S, D), np.float32)
qtile = 0
ntq = 1
for qtile in range(1):
    for wid in range(4):
        rowb = [qtile * 128 + wid * 16, qtile * 128 + 64 + wid * 16]
        # Q/dO B-frags: B[k=d][n=qrow]: lane n=l15, k=kc*32+lg*8+j
        # (simulate as full matrices per nq: qf[nq] = Q[rows, :] )
        kv_end = min(S, qtile * 128 + 128)
        ntiles = (kv_end + 63) // 64
        lse_r = [lse_ref[np.minimum(rowb[nq] + np.arange(16), S - 1)]
                 for nq in range(2)]
        del_r = [delta[np.minimum(rowb[nq] + np.arange(16), S - 1)]
                 for nq in range(2)]
        dqacc = np.zeros((nd16, 2, 16, 16), np.float32)  # [dt][nq][m=qrow16][n=d16]
        for t in range(ntiles):
            kv0 = t * 64
            lds_k, lds_v = stage(kv0)
            act0 = kv0 <= rowb[0] + 15
            smask = kv_end < kv0 + 64
            pk_ds = np.zeros((2, 2, 16, 32), np.float32)  # [nq][kc][m=qrow l15][k j]
            for mt in range(4):
                # A rows: K[perm16(mt, l15)] (16 rows), k-dim = d
                # S^T C[m=key positions][n=qrow]: simulate per lane (l15=qrow idx)
                for nq in range(2):
                    if nq == 0 and not act0:
                        continue
                    for lg in range(4):
                        for l15 in range(16):
                            qrow = rowb[nq] + l15
                            qrc = min(qrow, S - 1)
                            for r in range(4):
                                # C position (mt, lg, r) for lane (l15, lg)
                                arow = lg * 4 + r  # m position
                                keyrow = perm16(mt, arow)
                                # A row from LDS k image
                                kvec = np.array([lds_k[rm_idx(keyrow, d)] for d in range(D)])
                                vvec = np.array([lds_v[rm_idx(keyrow, d)] for d in range(D)])
                                sval = kvec @ Q[qrc]
                                dval = vvec @ dO[qrc]
                                keyr = kv0 + cpos16(mt, lg) + r
                                diag = (kv0 + 63 > rowb[nq]) or smask
                                e = scale * sval - lse_r[nq][l15]
                                if diag and (keyr > qrow or keyr >= S or qrow >= S):
                                    p = 0.0
                                else:
                                    p = np.exp(e)
                                ds = scale * p * (dval - del_r[nq][l15])
                                kc = mt >> 1
                                j = (mt & 1) * 4 + r
                                pk_ds[nq][kc][l15][lg * 8 + j] = ds
            # dQ += dS K
            for dt in range(nd16):
                for kc in range(2):
                    # B frag via tr16: lane n=l15 -> column dt*16+l15, k=lg*8+j
                    Bf = np.zeros((32, 16), np.float32)
                    for l15 in range(16):
                        for lg in range(4):
                            fr = tr16_frag(lds_k, kc * 32 + lg * 8, dt * 16, l15)
                            Bf[lg * 8:lg * 8 + 8, l15] = fr
                    for nq in range(2):
                        if nq == 0 and not act0:
                            continue
                        dqacc[dt][nq] += pk_ds[nq][kc] @ Bf
        # store
        for nq in range(2):
            for lg in range(4):
                for r in range(4):
                    qrow = rowb[nq] + lg * 4 + r
                    if qrow >= S:
                        continue
                    for dt in range(nd16):
                        for l15 in range(16):
                            dQ[qrow, dt * 16 + l15] = dqacc[dt][nq][lg * 4 + r][l15]

def test_dq_fragment_mapping():
    rel = np.linalg.norm(dQ - dQ_ref) / np.linalg.norm(dQ_ref)
    assert rel < 1e-5, f"dq index-math mismatch: rel {rel}"
