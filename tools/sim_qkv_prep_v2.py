"""CPU simulation of csrc/qkv_prep.hip's v2 index/shuffle flow against the
reference semantics (per-head RMSNorm + NeoX RoPE + KV scatter), following
the lane math exactly. Run: python tools/sim_qkv_prep_v2.py"""

import numpy as np

WAVE = 64


def reference(qkv_row, w, cs_row, D, eps, do_norm):
    half = D // 2
    x = qkv_row.astype(np.float64).copy()
    if do_norm:
        inv = 1.0 / np.sqrt((x * x).sum() / D + eps)
        x = x * inv * w
    out = np.empty(D)
    for d in range(half):
        c, s = cs_row[d], cs_row[d + half]
        out[d] = x[d] * c - x[d + half] * s
        out[d + half] = x[d + half] * c + x[d] * s
    return out


def v2_sim(qkv_row, w, cs_row, D, eps, do_norm):
    chunks = D // 8
    hc = chunks // 2
    half = D // 2
    # lane g holds chunk g
    x = np.zeros((chunks, 8))
    for g in range(chunks):
        x[g] = qkv_row[g * 8:(g + 1) * 8]
    if do_norm:
        ssq = (x * x).sum()  # the shfl_xor tree computes the group total
        inv = 1.0 / np.sqrt(ssq / D + eps)
        for g in range(chunks):
            x[g] = x[g] * inv * w[g * 8:(g + 1) * 8]
    out = np.zeros(D)
    for g in range(chunks):
        lo = g < hc
        for j in range(8):
            e = g * 8 + j
            p = x[g ^ hc][j]  # __shfl_xor(x[j], hc)
            c = cs_row[e if lo else e - half]
            s = cs_row[e + half if lo else e]
            out[e] = x[g][j] * c - p * s if lo else x[g][j] * c + p * s
    return out


rng = np.random.default_rng(0)
for D in (128, 64):
    half = D // 2
    for do_norm in (True, False):
        for trial in range(20):
            row = rng.standard_normal(D)
            w = rng.standard_normal(D)
            theta = rng.standard_normal(half)
            cs = np.concatenate([np.cos(theta), np.sin(theta)])
            ref = reference(row, w, cs, D, 1e-6, do_norm)
            got = v2_sim(row, w, cs, D, 1e-6, do_norm)
            err = np.abs(ref - got).max()
            assert err < 1e-12, (D, do_norm, trial, err)
    print(f"D={D}: v2 index flow exact (norm on/off x 20 trials)")
