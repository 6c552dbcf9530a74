"""CPU simulation of csrc/attn_decode_mfma.hip's index flow, to debug the
D=64 instantiation without a GPU.

Models the LDS arrays and per-lane loops exactly as written, with
mfma_f32_16x16x32_bf16 semantics implemented from the verified fragment
layout:
    A[16,32]: lane l holds A[l%16][(l/16)*8 + j]
    B[32,16]: lane l holds B[(l/16)*8 + j][l%16]
    D[16,16]: lane l holds D[(l/16)*4 + r][l%16]

Run: python tools/sim_mfma_decode.py
"""

import numpy as np

WAVE = 64
BS = 32
PV_PAD = 40


def mfma16(a_lanes, b_lanes, c_lanes):
    """a_lanes/b_lanes: [64][8] floats; c_lanes: [64][4]. Returns new [64][4]."""
    A = np.zeros((16, 32))
    B = np.zeros((32, 16))
    for l in range(WAVE):
        for j in range(8):
            A[l % 16][(l // 16) * 8 + j] = a_lanes[l][j]
            B[(l // 16) * 8 + j][l % 16] = b_lanes[l][j]
    D = A @ B
    out = np.array(c_lanes, dtype=float)
    for l in range(WAVE):
        for r in range(4):
            out[l][r] += D[(l // 16) * 4 + r][l % 16]
    return out


def qt_swz(D, head, byte_in_row):
    mask = 15 if D == 128 else 7
    return head * (2 * D) + (byte_in_row ^ ((head & mask) << 4))


def sim(D, L, G=1, seed=0):
    rng = np.random.default_rng(seed)
    Hk = 1
    kh = 0
    scale = 1.0 / np.sqrt(D)
    npages = (L + BS - 1) // BS
    K = rng.standard_normal((npages * BS, D))
    V = rng.standard_normal((npages * BS, D))
    q = rng.standard_normal((G, D))

    # LDS (element-typed; byte addressing divided by 2 for qt)
    qt = np.zeros(16 * D)          # addressed via qt_swz bytes / 2
    vt = np.zeros(D * PV_PAD)
    p = np.zeros(16 * PV_PAD)
    alpha = np.zeros(16)
    linv = np.zeros(16)

    # ---- stage Q^T
    for lane in range(WAVE):
        lo16, hi4 = lane & 15, lane >> 4
        head = lo16
        for t in range(D // 32):
            k0 = hi4 * 8 + t * 32
            val = np.zeros(8)
            if head < G:
                val = q[head][k0:k0 + 8] * scale
            base = qt_swz(D, head, k0 * 2) // 2
            qt[base:base + 8] = val

    m_run = np.full(WAVE, -1e30)
    l_run = np.zeros(WAVE)
    DB = D // 16
    acc = np.zeros((DB, WAVE, 4))

    for pg in range(npages):
        valid = min(BS, L - pg * BS)
        kpage = K[pg * BS:(pg + 1) * BS]
        vpage = V[pg * BS:(pg + 1) * BS]

        # ---- stage V^T
        for it in range((BS * D) // (WAVE * 8)):
            for lane in range(WAVE):
                flat = it * WAVE + lane
                pos = flat & 31
                d0 = (flat >> 5) * 8
                for j in range(8):
                    vt[(d0 + j) * PV_PAD + pos] = vpage[pos][d0 + j]

        # ---- QK^T
        s01 = []
        for half in range(2):
            ka = np.zeros((WAVE, 8))
            qb = np.zeros((WAVE, 8))
            d = np.zeros((WAVE, 4))
            for kk in range(D // 32):
                for lane in range(WAVE):
                    lo16, hi4 = lane & 15, lane >> 4
                    pos = half * 16 + lo16
                    ka[lane] = kpage[pos][kk * 32 + hi4 * 8: kk * 32 + hi4 * 8 + 8]
                    base = qt_swz(D, lo16, (kk * 32 + hi4 * 8) * 2) // 2
                    qb[lane] = qt[base:base + 8]
                d = mfma16(ka, qb, d)
            s01.append(d)

        # ---- softmax (per lane = head lo16)
        sv = np.zeros((WAVE, 8))
        for lane in range(WAVE):
            lo16, hi4 = lane & 15, lane >> 4
            for half in range(2):
                for r in range(4):
                    pos = half * 16 + hi4 * 4 + r
                    x = s01[half][lane][r]
                    sv[lane][half * 4 + r] = x if pos < valid else -1e30
        tmax = sv.max(axis=1)
        # shfl_xor 16, 32: combine lanes with same lo16
        for lane in range(WAVE):
            grp = [lane & 15, (lane & 15) + 16, (lane & 15) + 32, (lane & 15) + 48]
            tmax[lane] = max(sv[g].max() for g in grp)
        m_new = np.maximum(m_run, tmax)
        al = np.exp(m_run - m_new)
        m_run = m_new.copy()
        ev = np.exp(sv - m_new[:, None])
        psum = np.zeros(WAVE)
        for lane in range(WAVE):
            grp = [lane & 15, (lane & 15) + 16, (lane & 15) + 32, (lane & 15) + 48]
            psum[lane] = sum(ev[g].sum() for g in grp)
        l_run = l_run * al + psum
        for lane in range(WAVE):
            lo16, hi4 = lane & 15, lane >> 4
            if hi4 == 0:
                alpha[lo16] = al[lane]
            for half in range(2):
                for r in range(4):
                    p[lo16 * PV_PAD + half * 16 + hi4 * 4 + r] = ev[lane][half * 4 + r]

        # ---- rescale + PV
        for lane in range(WAVE):
            hi4 = lane >> 4
            for b in range(DB):
                for r in range(4):
                    acc[b][lane][r] *= alpha[hi4 * 4 + r]
        pa = np.zeros((WAVE, 8))
        for lane in range(WAVE):
            lo16, hi4 = lane & 15, lane >> 4
            pa[lane] = p[lo16 * PV_PAD + hi4 * 8: lo16 * PV_PAD + hi4 * 8 + 8]
        for b in range(DB):
            vb = np.zeros((WAVE, 8))
            for lane in range(WAVE):
                lo16, hi4 = lane & 15, lane >> 4
                base = (b * 16 + lo16) * PV_PAD + hi4 * 8
                vb[lane] = vt[base:base + 8]
            acc[b] = mfma16(pa, vb, acc[b])

    # ---- epilogue
    out = np.zeros((G, D))
    for lane in range(WAVE):
        lo16, hi4 = lane & 15, lane >> 4
        if hi4 == 0:
            linv[lo16] = 1.0 / l_run[lane]
        li = [linv[hi4 * 4 + r] for r in range(4)]
    for lane in range(WAVE):
        lo16, hi4 = lane & 15, lane >> 4
        for r in range(4):
            head = hi4 * 4 + r
            if head >= G:
                continue
            for b in range(DB):
                out[head][b * 16 + lo16] = acc[b][lane][r] / l_run[(head & 15)]
    # reference
    ref = np.zeros((G, D))
    for h in range(G):
        s = (K[:L] @ q[h]) * scale
        w = np.exp(s - s.max())
        w /= w.sum()
        ref[h] = w @ V[:L]
    return out, ref


for D in (128, 64):
    for L in (1, 16, 33):
        out, ref = sim(D, L)
        err = np.abs(out - ref).max()
        print(f"D={D} L={L}: max err {err:.2e} {'OK' if err < 1e-9 else 'MISMATCH'}")
