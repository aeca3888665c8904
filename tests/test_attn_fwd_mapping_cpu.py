"""Lane-exact CPU simulation of attn_fwd_kernel's fragment/index math:
swapped-operand S^T (perm16), interleaved 16-row wave groups, online
softmax with lane-local rows, in-register P packing order, st_idx subtile
V image + tr16 transpose reads, O^T epilogue mapping — against a numpy
causal-attention reference (companion of test_attn_mapping_cpu.py)."""
import numpy as np

np.random.seed(1)
S, D = 128, 64
scale = D ** -0.5
Q = np.random.randn(S, D).astype(np.float32)
K = np.random.randn(S, D).astype(np.float32)
V = np.random.randn(S, D).astype(np.float32)

Sr = Q @ K.T * scale
Sr[~np.tril(np.ones((S, S), bool))] = -np.inf
P = np.exp(Sr - Sr.max(1, keepdims=True))
P /= P.sum(1, keepdims=True)
O_ref = P @ V
lse_ref = np.log(np.exp(Sr - Sr.max(1, keepdims=True)).sum(1)) + Sr.max(1)

nd16 = D // 16


def perm16(mt, l15):
    return (mt >> 1) * 32 + (l15 >> 2) * 8 + (mt & 1) * 4 + (l15 & 3)


def cpos16(mt, lg):
    return (mt >> 1) * 32 + lg * 8 + (mt & 1) * 4


def st_idx(row, col):
    return ((col >> 4) << 10) + (row << 4) + (col & 15)


def stage(kv0, X):
    lds = np.zeros(((D // 16) << 10), np.float32)
    for key in range(64):
        kg = min(kv0 + key, S - 1)
        for c in range(D):
            lds[st_idx(key, c)] = X[kg, c]
    return lds


def tr16_frag(lds, row0, col0, l15):
    return np.array([lds[st_idx(row0 + j, col0 + l15)] for j in range(8)])


O = np.zeros((S, D), np.float32)
LSE = np.zeros(S, np.float32)
for qtile in range(S // 128):
    for wid in range(4):
        rowb = [qtile * 128 + wid * 16, qtile * 128 + 64 + wid * 16]
        kv_end = min(S, qtile * 128 + 128)
        ntiles = (kv_end + 63) // 64
        # per lane (l15 = qrow within group): oacc[dt][nq][m=lg*4+r][l15]
        oacc = np.zeros((nd16, 2, 16, 16), np.float32)
        mrow = np.full((2, 16), -np.inf)
        lrow = np.zeros((2, 16))
        for t in range(ntiles):
            kv0 = t * 64
            k_lds = stage(kv0, K)
            v_lds = stage(kv0, V)
            for nq in range(2):
                # S^T C-frag: [mt][lg][r] per lane l15 (= qrow index)
                sfrag = np.zeros((4, 4, 4, 16), np.float32)
                for mt in range(4):
                    for lg in range(4):
                        for r in range(4):
                            arow = lg * 4 + r
                            keyrow = perm16(mt, arow)
                            kvec = np.array(
                                [k_lds[st_idx(keyrow, d)] for d in range(D)])
                            for l15 in range(16):
                                qr = min(rowb[nq] + l15, S - 1)
                                sfrag[mt][lg][r][l15] = kvec @ Q[qr] * scale
                # causal mask + online softmax per lane
                for l15 in range(16):
                    qrow = rowb[nq] + l15
                    vals = {}
                    for mt in range(4):
                        for lg in range(4):
                            for r in range(4):
                                key = kv0 + cpos16(mt, lg) + r
                                v_ = sfrag[mt][lg][r][l15]
                                if key > qrow or key >= S:
                                    v_ = -np.inf
                                vals[(mt, lg, r)] = v_
                    tmax = max(vals.values())
                    mnew = max(mrow[nq][l15], tmax)
                    alpha = 0.0 if mrow[nq][l15] == -np.inf else np.exp(
                        mrow[nq][l15] - mnew)
                    mrow[nq][l15] = mnew
                    p = {k: (0.0 if v_ == -np.inf else np.exp(v_ - mnew))
                         for k, v_ in vals.items()}
                    lrow[nq][l15] = lrow[nq][l15] * alpha + sum(p.values())
                    # pk B-operand: per lg, key j = within kc*32+lg*8+j
                    # O^T += mfma(V^T, P^T): C[m=d(lg*4+r)][n=qrow l15]
                    for dt in range(nd16):
                        for lgd in range(4):
                            for rd in range(4):
                                d = dt * 16 + lgd * 4 + rd
                                acc = oacc[dt][nq][lgd * 4 + rd][l15] * alpha
                                for kc in range(2):
                                    for lg in range(4):
                                        for j in range(8):
                                            key_local = kc * 32 + lg * 8 + j
                                            # invert cpos16: key ->
                                            # (mt, lg, r) of the C frag
                                            mt = kc * 2 + (key_local % 8) // 4
                                            r = key_local % 4
                                            lgp = (key_local % 32) // 8
                                            pv = p[(mt, lgp, r)]
                                            vv = v_lds[st_idx(key_local, d)]
                                            acc += pv * vv
                                oacc[dt][nq][lgd * 4 + rd][l15] = acc
        # epilogue
        for nq in range(2):
            for l15 in range(16):
                qrow = rowb[nq] + l15
                if qrow >= S:
                    continue
                inv = 1.0 / lrow[nq][l15]
                for dt in range(nd16):
                    for m in range(16):
                        O[qrow, dt * 16 + m] = oacc[dt][nq][m][l15] * inv
                LSE[qrow] = mrow[nq][l15] + np.log(lrow[nq][l15])


def test_fwd_fragment_mapping():
    rel = np.linalg.norm(O - O_ref) / np.linalg.norm(O_ref)
    assert rel < 1e-5, f"fwd index-math mismatch: rel {rel}"
    assert np.allclose(LSE, lse_ref, atol=1e-4)
