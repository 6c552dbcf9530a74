"""CPU simulation of csrc/gemm_tn.hip's index flow: glds source pre-swizzle
vs ds_read address swizzle consistency (modes 0/1/2), the 2(M)x4(N) wave
decomposition, mfma_f32_16x16x32 fragment layouts, K-tile double buffering
and the D-layout epilogue — checked against a plain matmul.
Run: python tools/sim_gemm_tn.py"""

import numpy as np

WAVE = 64


def swz(mode, byte):
    if mode == 1:
        return byte ^ (((byte >> 9) & 1) << 5)
    if mode == 2:
        return byte ^ (((byte >> 8) & 7) << 4)
    return byte


def stage_tile(mode, rows, gsrc, rows_valid):
    """Returns the LDS image as a [rows*64] element array (16B chunks are 8
    elements); follows stage_tile's per-lane math exactly."""
    CH = rows // 64
    lds = np.zeros(rows * 64)
    for wave in range(8):
        for j in range(CH):
            c = wave * CH + j
            for lane in range(WAVE):
                b = c * 1024 + lane * 16
                lb = swz(mode, b)
                row = min(lb >> 7, rows_valid - 1)
                koff_e = (lb & 127) // 2  # elements within the row
                for e in range(8):
                    lds[b // 2 + e] = gsrc[row][koff_e + e]
    return lds


def frag(mode, lds, row, kk16):
    lb = row * 128 + kk16 * 16
    b = swz(mode, lb)
    return lds[b // 2: b // 2 + 8]


def mfma16(a_lanes, b_lanes, c):
    A = np.zeros((16, 32))
    B = np.zeros((32, 16))
    for l in range(WAVE):
        for j in range(8):
            A[l % 16][(l // 16) * 8 + j] = a_lanes[l][j]
            B[(l // 16) * 8 + j][l % 16] = b_lanes[l][j]
    return c + A @ B


def sim(mode, BM, BN, M, N, K):
    rng = np.random.default_rng(mode * 100 + BM + BN)
    x = rng.standard_normal((M, K))
    w = rng.standard_normal((N, K))
    out = np.zeros((M, N))
    WM, WN = BM // 2, BN // 4
    MF, NF = WM // 16, WN // 16
    n_tiles = (N + BN - 1) // BN
    for bid in range((M // BM) * n_tiles):
        m0 = (bid % (M // BM)) * BM
        n0 = (bid // (M // BM)) * BN
        acc = {}  # (wave, i, j) -> D[16,16]
        KT = K // 64
        for t in range(KT):
            aT = stage_tile(mode, BM, x[m0:, t * 64:], BM)
            bT = stage_tile(mode, BN, w[n0:, t * 64:], N - n0)
            for wave in range(8):
                wm, wn = wave >> 2, wave & 3
                for kk in range(2):
                    for i in range(MF):
                        for j in range(NF):
                            af = np.zeros((WAVE, 8))
                            bf = np.zeros((WAVE, 8))
                            for l in range(WAVE):
                                kk16 = kk * 4 + (l >> 4)
                                af[l] = frag(mode, aT,
                                             wm * WM + i * 16 + (l & 15), kk16)
                                bf[l] = frag(mode, bT,
                                             wn * WN + j * 16 + (l & 15), kk16)
                            acc[(wave, i, j)] = mfma16(
                                af, bf, acc.get((wave, i, j), np.zeros((16, 16))))
        # epilogue: D[16,16] lane l holds rows (l/16)*4+r, col l%16
        for (wave, i, j), d in acc.items():
            wm, wn = wave >> 2, wave & 3
            for row in range(16):
                for col in range(16):
                    gm = m0 + wm * WM + i * 16 + row
                    gn = n0 + wn * WN + j * 16 + col
                    if gn < N:
                        out[gm][gn] = d[row][col]
    ref = x @ w.T
    return np.abs(out - ref).max()


for mode in (0, 1, 2):
    for BM, BN, M, N, K in [(256, 256, 256, 256, 128),
                            (128, 128, 128, 200, 128)]:
        err = sim(mode, BM, BN, M, N, K)
        assert err < 1e-9, (mode, BM, BN, err)
    print(f"swz mode {mode}: staging/frag/epilogue index flow exact OK")
