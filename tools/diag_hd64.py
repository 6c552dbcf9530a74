"""Localize the head-dim-64 MFMA decode mismatch: run minimal decode configs
and print the diff pattern (which seq/head/dim) vs the fp32 reference."""

import math
import sys

import torch

sys.path.insert(0, ".")
from sutro_amd import ops  # noqa: E402
from sutro_amd.ops import torch_ref as R  # noqa: E402


def bf(t):
    return t.bfloat16().cuda()


def run(seq_lens, Hq, Hk, D, tag):
    bs = 32
    S = len(seq_lens)
    nb = sum((L + bs - 1) // bs for L in seq_lens) + 1
    kc = bf(torch.randn(nb, Hk, bs, D))
    vc = bf(torch.randn(nb, Hk, bs, D))
    tables, i = [], 1
    for L in seq_lens:
        n = (L + bs - 1) // bs
        tables.append(list(range(i, i + n)))
        i += n
    max_b = max(len(t) for t in tables)
    bt = torch.zeros(S, max_b, dtype=torch.int32)
    for s, t in enumerate(tables):
        bt[s, : len(t)] = torch.tensor(t, dtype=torch.int32)
    q = bf(torch.randn(S, Hq, D))
    sl = torch.tensor(seq_lens, dtype=torch.int32)
    ql = torch.arange(S + 1, dtype=torch.int32)
    scale = 1.0 / math.sqrt(D)
    got = ops.paged_attention(q, kc, vc, bt.cuda(), sl.cuda(), ql.cuda(),
                              scale, num_decodes_tail=S,
                              prefill_token_count=0)
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(), vc.float().cpu(),
                            bt, sl, ql, scale)
    d = (got.float().cpu() - ref).abs()  # [S, Hq, D]
    print(f"{tag}: max={d.max().item():.4f}")
    if d.max().item() > 3e-2:
        bad = (d > 3e-2)
        for s in range(S):
            for h in range(Hq):
                n = int(bad[s, h].sum())
                if n:
                    dims = bad[s, h].nonzero().flatten().tolist()
                    print(f"  seq {s} head {h}: {n} bad dims "
                          f"first/last {dims[0]}/{dims[-1]} "
                          f"maxdiff {d[s, h].max().item():.4f}")


torch.manual_seed(3)
run([1], 1, 1, 64, "L=1 G=1")
run([16], 1, 1, 64, "L=16 G=1")
run([32], 1, 1, 64, "L=32 G=1")
run([33], 1, 1, 64, "L=33 G=1 (2 pages)")
run([70, 33], 8, 2, 64, "test config G=4")
run([70, 33], 8, 2, 128, "control D=128 G=4")
