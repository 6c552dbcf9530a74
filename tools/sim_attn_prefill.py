"""CPU simulation of csrc/attn_prefill.hip's index flow (one wave = one
q-tile x one q-head), following the lane math exactly: k_swz'd K staging,
V^T staging with VT_PAD rows, mfma_f32_32x32x16 fragment layouts, the
causal/page masking, the online softmax, and the P LDS round trip.

Checks the result against plain causal attention for D=128 and D=64,
multiple KV pages, continuation chunks (ctx > 0) and partial tiles.
Run: python tools/sim_attn_prefill.py"""

import numpy as np

BS = 32
QTILE = 32
VT_PAD = 40
WAVE = 64


def d_row(r, hi):
    return (r & 3) + 8 * (r >> 2) + 4 * hi


def k_swz(D, row, byte_in_row):
    mask = 15 if D == 128 else 7
    return row * (2 * D) + (byte_in_row ^ ((row & mask) << 4))


def mfma32(a_lanes, b_lanes, c_lanes):
    """A[32,16] B[16,32] frags per lane -> accumulate D[32,32]."""
    A = np.zeros((32, 16))
    B = np.zeros((16, 32))
    for l in range(WAVE):
        for j in range(8):
            A[l % 32][(l // 32) * 8 + j] = a_lanes[l][j]
            B[(l // 32) * 8 + j][l % 32] = b_lanes[l][j]
    D = A @ B
    out = np.array(c_lanes, dtype=float)
    for l in range(WAVE):
        hi = l >> 5
        for r in range(16):
            out[l][r] += D[d_row(r, hi)][l % 32]
    return out


def sim_tile(D, q_rows, K, V, q0, nq_total, L, scale):
    """One wave processing q-tile rows [q0, q0+32) of a chunk with nq_total
    new tokens over cached length L. K/V: [L, D] already gathered in page
    order (the kernel reads pages via block tables; gathering is the same
    index arithmetic exercised by the decode sim)."""
    nq = min(QTILE, nq_total - q0)
    ctx = L - nq_total
    qabs_base = ctx + q0
    kv_end = qabs_base + nq
    ntiles_kv = (kv_end + BS - 1) // BS

    # Q fragments per lane: qb[kk][j] = q[my_q][hi*8 + kk*16 + j]
    def qfrag(l, kk):
        my_q = l & 31
        hi = l >> 5
        row = my_q if my_q < nq else 0
        return np.array([q_rows[row][hi * 8 + kk * 16 + j] for j in range(8)])

    acc = np.zeros((D // 32, WAVE, 16))
    m_run = np.full(WAVE, -1e30)
    l_run = np.zeros(WAVE)

    for kt in range(ntiles_kv):
        kv0 = kt * BS
        # K staging through the swizzled byte image (verifies write/read
        # consistency of k_swz)
        kview = np.zeros(BS * D)  # value per 2-byte slot (fp64 sim)
        for row in range(BS):
            for slot in range(D // 8):
                for j in range(8):
                    byte = k_swz(D, row, (slot * 8 + j) * 2)
                    kv_idx = kv0 + row
                    val = K[kv_idx][slot * 8 + j] if kv_idx < L else 0.0
                    kview[byte // 2] = val
        # V^T staging: vt[d][kv]
        vt = np.zeros((D, VT_PAD))
        for it in range(BS * (D // 8)):
            kv = it % BS
            d0 = (it // BS) * 8
            for j in range(8):
                kv_idx = kv0 + kv
                vt[d0 + j][kv] = V[kv_idx][d0 + j] if kv_idx < L else 0.0

        # QK^T
        d1 = np.zeros((WAVE, 16))
        for kk in range(D // 16):
            ka = np.zeros((WAVE, 8))
            qb = np.zeros((WAVE, 8))
            for l in range(WAVE):
                row = l & 31
                hi = l >> 5
                for j in range(8):
                    byte = k_swz(D, row, (kk * 16 + hi * 8 + j) * 2)
                    ka[l][j] = kview[byte // 2]
                qb[l] = qfrag(l, kk)
            d1 = mfma32(ka, qb, d1)

        # online softmax per q (= lane & 31, shared by lane and lane+32)
        sc = np.zeros((WAVE, 16))
        tmax = np.full(WAVE, -1e30)
        for l in range(WAVE):
            hi = l >> 5
            my_q = l & 31
            qabs = qabs_base + my_q
            for r in range(16):
                kvpos = kv0 + d_row(r, hi)
                v = d1[l][r] * scale
                if kvpos > qabs or kvpos >= L:
                    v = -1e30
                sc[l][r] = v
                tmax[l] = max(tmax[l], v)
        for l in range(WAVE):
            tmax[l] = max(tmax[l], tmax[l ^ 32])  # shfl_xor 32
        m_new = np.maximum(m_run, tmax)
        alpha = np.exp(m_run - m_new)
        psum = np.zeros(WAVE)
        for l in range(WAVE):
            for r in range(16):
                sc[l][r] = np.exp(sc[l][r] - m_new[l])
                psum[l] += sc[l][r]
        psum_tot = np.array([psum[l] + psum[l ^ 32] for l in range(WAVE)])
        l_run = l_run * alpha + psum_tot
        m_run = m_new.copy()

        # P round trip through LDS [q][kv]
        p_lds = np.zeros((QTILE, VT_PAD))
        alpha_lds = np.zeros(QTILE)
        for l in range(WAVE):
            hi = l >> 5
            my_q = l & 31
            for r in range(16):
                p_lds[my_q][d_row(r, hi)] = sc[l][r]
            if hi == 0:
                alpha_lds[my_q] = alpha[l]

        # rescale + PV
        for l in range(WAVE):
            hi = l >> 5
            for b in range(D // 32):
                for r in range(16):
                    acc[b][l][r] *= alpha_lds[d_row(r, hi)]
        for b in range(D // 32):
            for kk in range(2):
                pa = np.zeros((WAVE, 8))
                vb = np.zeros((WAVE, 8))
                for l in range(WAVE):
                    hi = l >> 5
                    my_q = l & 31
                    for j in range(8):
                        pa[l][j] = p_lds[my_q][kk * 16 + hi * 8 + j]
                        vb[l][j] = vt[b * 32 + my_q][kk * 16 + hi * 8 + j]
                acc[b] = mfma32(pa, vb, acc[b])

    # epilogue
    lsum = np.zeros(QTILE)
    for l in range(WAVE):
        if (l >> 5) == 0:
            lsum[l & 31] = l_run[l]
    out = np.zeros((nq, D))
    for l in range(WAVE):
        hi = l >> 5
        d_col = l & 31
        for b in range(D // 32):
            for r in range(16):
                qrow = d_row(r, hi)
                if qrow < nq:
                    lv = lsum[qrow]
                    out[qrow][b * 32 + d_col] = (
                        acc[b][l][r] / lv if lv > 0 else 0.0)
    return out


def reference(q_rows, K, V, q0, nq_total, L, scale):
    ctx = L - nq_total
    nq = min(QTILE, nq_total - q0)
    out = np.zeros((nq, K.shape[1]))
    for i in range(nq):
        qabs = ctx + q0 + i
        s = (K[:qabs + 1] @ q_rows[i]) * scale
        w = np.exp(s - s.max())
        w /= w.sum()
        out[i] = w @ V[:qabs + 1]
    return out


rng = np.random.default_rng(11)
for D in (128, 64):
    for (L, nq_total, q0) in [(1, 1, 0), (32, 32, 0), (70, 70, 32),
                              (90, 40, 0), (90, 40, 32), (65, 1, 0)]:
        K = rng.standard_normal((L, D))
        V = rng.standard_normal((L, D))
        nq = min(QTILE, nq_total - q0)
        q_rows = rng.standard_normal((max(nq, 1), D))
        scale = 1.0 / np.sqrt(D)
        got = sim_tile(D, q_rows, K, V, q0, nq_total, L, scale)
        ref = reference(q_rows, K, V, q0, nq_total, L, scale)
        err = np.abs(got - ref).max()
        assert err < 1e-9, (D, L, nq_total, q0, err)
    print(f"D={D}: prefill index flow exact across chunk/page/edge cases OK")
