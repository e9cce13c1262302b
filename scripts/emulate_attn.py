"""CPU emulation of the attention forward's lane-level dataflow (one
workgroup) used to locate logic bugs without GPU round-trips.

NOTE: models the RETIRED 16x16x32 kernel variant (the shipped kernels use
32x32x16); kept as the record of the debugging method that located the
cross-lane P-repack bug (see docs/kernels.md).

MFMA semantics (validated on hardware by scripts/debug_mfma.py):
  D[i][j] = sum over g in 0..3, e in 0..7 of
            Aslot(lane=g*16+i, e) * Bslot(lane=g*16+j, e)
  C/D: lane (g,c) reg r holds C[4g+r][c].
"""

import numpy as np

BLK, D, NW = 64, 64, 4


def mfma(a_slots, b_slots, acc):
    # a_slots, b_slots: [64][8] floats; acc: [64][4]
    out = acc.copy()
    for i in range(16):
        for j in range(16):
            s = 0.0
            for g in range(4):
                s += np.dot(a_slots[g * 16 + i], b_slots[g * 16 + j])
            # write into C layout: lane (g2, j) reg r holds C[4*g2+r][j]
            g2, r = divmod(i, 4)
            out[g2 * 16 + j][r] += s
    return out


def shfl(vals, src):  # vals: [64] array, src: per-lane source index
    return np.array([vals[src[l]] for l in range(64)])


def bf16(x):
    # round-to-nearest-even bf16 quantization
    x32 = np.asarray(x, dtype=np.float32)
    u = x32.view(np.uint32)
    rounded = ((u + 0x7FFF + ((u >> 16) & 1)) & 0xFFFF0000).view(np.float32)
    return rounded


def fwd_wave(Q, K, V, qb, w, T, scale):
    """Emulate one wave (16 q rows) of attn_fwd_kernel. Q,K,V: [T][64] f32
    (already bf16-quantized). Returns O rows [16][64], lse [16]."""
    lanes = [(l >> 4, l & 15) for l in range(64)]
    q_frag = np.zeros((2, 64, 8))
    for ds in range(2):
        for l, (g, c) in enumerate(lanes):
            q_frag[ds][l] = Q[qb * BLK + w * 16 + c, ds * 32 + 8 * g:ds * 32 + 8 * g + 8]
    o_acc = np.zeros((4, 64, 4))
    m_run = np.full(64, -np.inf)
    l_run = np.zeros(64)
    for j in range(qb + 1):
        st = np.zeros((4, 64, 4))
        for sub in range(4):
            acc = np.zeros((64, 4))
            for ds in range(2):
                a = np.zeros((64, 8))
                for l, (g, c) in enumerate(lanes):
                    a[l] = K[j * BLK + sub * 16 + c, ds * 32 + 8 * g: ds * 32 + 8 * g + 8]
                acc = mfma(a, q_frag[ds], acc)
            st[sub] = acc
        # mask + softmax per lane (qrow = c)
        mt = np.full(64, -np.inf)
        for sub in range(4):
            for l, (g, c) in enumerate(lanes):
                for r in range(4):
                    key = j * BLK + sub * 16 + 4 * g + r
                    qrow = qb * BLK + w * 16 + c
                    s = st[sub][l][r] * scale
                    if key > qrow:
                        s = -np.inf
                    st[sub][l][r] = s
                    mt[l] = max(mt[l], s)
        mt = np.maximum(mt, shfl(mt, [l ^ 16 for l in range(64)]))
        mt = np.maximum(mt, shfl(mt, [l ^ 32 for l in range(64)]))
        m_new = np.maximum(m_run, mt)
        alpha = np.exp(m_run - m_new)
        p = np.zeros((4, 64, 4))
        psum = np.zeros(64)
        for sub in range(4):
            for l in range(64):
                for r in range(4):
                    pv = 0.0 if st[sub][l][r] == -np.inf else np.exp(st[sub][l][r] - m_new[l])
                    p[sub][l][r] = pv
                    psum[l] += pv
        psum = psum + shfl(psum, [l ^ 16 for l in range(64)])
        psum = psum + shfl(psum, [l ^ 32 for l in range(64)])
        l_run = l_run * alpha + psum
        m_run = m_new
        # O rescale
        for l, (g, c) in enumerate(lanes):
            for dt in range(4):
                for r in range(4):
                    ar = alpha[4 * g + r]
                    o_acc[dt][l][r] *= ar
        # PV: A-frag via repack, B = V transposed read
        pq = bf16(p)  # packing quantizes P to bf16
        for ks in range(2):
            a = np.zeros((64, 8))
            for l, (g, c) in enumerate(lanes):
                qt = 2 * ks + (g >> 1)
                L0 = 32 * (g & 1) + c
                for e in range(8):
                    src = L0 if e < 4 else L0 + 16
                    a[l][e] = pq[qt][src][e & 3]
            for dt in range(4):
                b = np.zeros((64, 8))
                for l, (g, c) in enumerate(lanes):
                    for e in range(8):
                        b[l][e] = V[j * BLK + ks * 32 + 8 * g + e, dt * 16 + c]
                o_acc[dt] = mfma(a, b, o_acc[dt])
    # epilogue
    O = np.zeros((16, 64))
    lse = np.zeros(16)
    for l, (g, c) in enumerate(lanes):
        for dt in range(4):
            for r in range(4):
                O[4 * g + r, dt * 16 + c] = o_acc[dt][l][r] / l_run[4 * g + r]
    for c in range(16):
        lse[c] = m_run[c] + np.log(l_run[c])
    return O, lse


def main():
    rng = np.random.default_rng(0)
    T = 128
    scale = 1.0 / np.sqrt(D)
    Q = bf16(rng.standard_normal((T, D)))
    K = bf16(rng.standard_normal((T, D)))
    V = bf16(rng.standard_normal((T, D)))
    # reference
    S = (Q @ K.T) * scale
    mask = np.tril(np.ones((T, T), dtype=bool))
    S = np.where(mask, S, -np.inf)
    m = S.max(axis=1, keepdims=True)
    P = np.exp(S - m)
    lse_ref = (m.squeeze() + np.log(P.sum(axis=1)))
    O_ref = (P / P.sum(axis=1, keepdims=True)) @ V

    worst = 0.0
    for qb in range(T // BLK):
        for w in range(NW):
            O, lse = fwd_wave(Q, K, V, qb, w, T, scale)
            rows = slice(qb * BLK + w * 16, qb * BLK + w * 16 + 16)
            e1 = np.abs(O - O_ref[rows]).max()
            e2 = np.abs(lse - lse_ref[rows]).max()
            worst = max(worst, e1, e2)
            print(f"qb={qb} w={w}: O err {e1:.5f}  lse err {e2:.5f}")
    print("worst:", worst)


if __name__ == "__main__":
    main()
