"""GPU numerics tests: every CDNA4 HIP kernel vs the plain-PyTorch fp32
reference (sutro_amd.ops.torch_ref) on the same bf16 inputs."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from sutro_amd import _C, ops
    from sutro_amd.ops import torch_ref as R
else:  # collected on CPU; every test is skipped by the marker filter anyway
    _C = ops = R = None


def require_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


DEV = "cuda"


def bf(x):
    return x.to(torch.bfloat16).to(DEV)


def test_hip_extension_loaded():
    require_gpu()
    assert ops.hip_available(), "sutro_amd._C must be importable on the GPU box"


def test_mfma32_layout_probe():
    require_gpu()
    torch.manual_seed(0)
    # asymmetric B catches transposed layouts (guide: A=I-check rule)
    a = torch.randn(32, 16)
    b = torch.randn(16, 32) * torch.arange(1, 33).float() / 8.0
    c = _C.mfma32_probe(bf(a), bf(b))
    ref = (bf(a).float() @ bf(b).float())
    assert torch.allclose(c.cpu(), ref.cpu(), atol=2e-2, rtol=2e-2), (
        (c.cpu() - ref.cpu()).abs().max()
    )


@pytest.mark.parametrize("rows,cols", [(64, 5120), (3, 1024), (256, 4096)])
def test_rmsnorm(rows, cols):
    require_gpu()
    torch.manual_seed(1)
    x = bf(torch.randn(rows, cols))
    w = bf(torch.randn(cols) * 0.5 + 1.0)
    got = ops.rmsnorm(x, w, 1e-6).float().cpu()
    ref = R.rmsnorm(x.cpu(), w.cpu(), 1e-6).float()
    assert torch.allclose(got, ref, atol=3e-2, rtol=3e-2)


def test_rmsnorm_small_rows():
    require_gpu()
    x = bf(torch.randn(100, 4, 128))  # qk-norm shape [T, H, D]
    w = bf(torch.ones(128) * 1.3)
    got = ops.rmsnorm(x, w, 1e-6).float().cpu()
    ref = R.rmsnorm(x.cpu(), w.cpu(), 1e-6).float()
    assert torch.allclose(got, ref, atol=3e-2, rtol=3e-2)


def test_fused_add_rmsnorm():
    require_gpu()
    torch.manual_seed(2)
    x = bf(torch.randn(64, 2048))
    res = bf(torch.randn(64, 2048))
    x_cpu, res_cpu = x.cpu(), res.cpu()
    w = bf(torch.randn(2048) * 0.1 + 1.0)
    y, new_res = ops.fused_add_rmsnorm(x, res, w, 1e-6)
    ref_y, ref_res = R.fused_add_rmsnorm(x_cpu, res_cpu, w.cpu(), 1e-6)
    assert torch.allclose(new_res.float().cpu(), ref_res.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(y.float().cpu(), ref_y.float(), atol=3e-2, rtol=3e-2)


def test_silu_mul():
    require_gpu()
    x = bf(torch.randn(128, 2 * 1024))
    got = ops.silu_mul(x).float().cpu()
    ref = R.silu_mul(x.cpu()).float()
    assert torch.allclose(got, ref, atol=2e-2, rtol=2e-2)


def _mk_cache(nb, hk, bs, d):
    return (torch.zeros(nb, hk, bs, d, dtype=torch.bfloat16, device=DEV),
            torch.zeros(nb, hk, bs, d, dtype=torch.bfloat16, device=DEV))


def test_rope_and_cache():
    require_gpu()
    torch.manual_seed(3)
    T, Hq, Hk, D, bs = 33, 8, 2, 128, 32
    q = bf(torch.randn(T, Hq, D))
    k = bf(torch.randn(T, Hk, D))
    v = bf(torch.randn(T, Hk, D))
    q0, k0, v0 = q.cpu(), k.cpu(), v.cpu()
    pos = torch.randint(0, 500, (T,), dtype=torch.long)
    slots = torch.randperm(4 * bs)[:T].to(torch.long)
    cs = R.rope_cos_sin(512, D, 1e6).to(DEV)
    kc, vc = _mk_cache(4, Hk, bs, D)
    q_out = ops.rope_and_cache(q, k, v, pos.to(DEV), slots.to(DEV), kc, vc, cs)
    rq, rk = R.apply_rope(q0.float(), k0.float(), pos, cs.cpu())
    kc_ref = torch.zeros(4, Hk, bs, D)
    vc_ref = torch.zeros(4, Hk, bs, D)
    R.write_kv_cache(rk, v0.float(), kc_ref, vc_ref, slots)
    assert torch.allclose(q_out.float().cpu(), rq.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(kc.float().cpu(), kc_ref.to(torch.bfloat16).float(),
                          atol=3e-2, rtol=3e-2)
    assert torch.allclose(vc.float().cpu(), vc_ref.to(torch.bfloat16).float(),
                          atol=3e-2, rtol=3e-2)


def test_qkv_prep_fused():
    require_gpu()
    torch.manual_seed(7)
    T, Hq, Hk, D, bs = 29, 8, 2, 128, 32
    qkv = bf(torch.randn(T, (Hq + 2 * Hk) * D))
    pos = torch.randint(0, 400, (T,), dtype=torch.long)
    slots = torch.randperm(4 * bs)[:T].to(torch.long)
    cs = R.rope_cos_sin(512, D, 1e6).to(DEV)
    qw = bf(torch.randn(D) * 0.2 + 1.0)
    kw = bf(torch.randn(D) * 0.2 + 1.0)
    kc, vc = _mk_cache(4, Hk, bs, D)
    q_out = ops.fused_qkv_prep(qkv, Hq, Hk, D, pos.to(DEV), slots.to(DEV),
                               kc, vc, cs, qw, kw, 1e-6)
    # CPU reference composition on the same bf16 inputs
    kc_ref = torch.zeros(4, Hk, bs, D)
    vc_ref = torch.zeros(4, Hk, bs, D)
    q_ref = ops.fused_qkv_prep(qkv.cpu(), Hq, Hk, D, pos, slots, kc_ref,
                               vc_ref, cs.cpu(), qw.cpu(), kw.cpu(), 1e-6)
    assert torch.allclose(q_out.float().cpu(), q_ref.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(kc.float().cpu(), kc_ref.float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(vc.float().cpu(), vc_ref.float(), atol=3e-2, rtol=3e-2)


def _attn_setup(seq_lens, new_counts, Hq=8, Hk=2, D=128, bs=32, seed=5):
    """Build a scattered paged cache + q batch; returns gpu tensors + cpu refs."""
    torch.manual_seed(seed)
    S = len(seq_lens)
    total_blocks = sum((L + bs - 1) // bs for L in seq_lens) + 3
    perm = torch.randperm(total_blocks).tolist()
    kc, vc = _mk_cache(total_blocks, Hk, bs, D)
    tables = []
    i = 0
    for L in seq_lens:
        n = (L + bs - 1) // bs
        tables.append(perm[i:i + n])
        i += n
    max_b = max(len(t) for t in tables)
    bt = torch.zeros(S, max_b, dtype=torch.int32)
    for s, t in enumerate(tables):
        bt[s, :len(t)] = torch.tensor(t, dtype=torch.int32)
    qs = []
    for s, L in enumerate(seq_lens):
        K = bf(torch.randn(L, Hk, D))
        V = bf(torch.randn(L, Hk, D))
        slots = torch.tensor(
            [tables[s][p // bs] * bs + p % bs for p in range(L)], dtype=torch.long)
        R.write_kv_cache(K, V, kc, vc, slots.to(DEV))
        qs.append(bf(torch.randn(new_counts[s], Hq, D)))
    q = torch.cat(qs, 0)
    qlocs = torch.tensor([0] + list(torch.tensor(new_counts).cumsum(0)),
                         dtype=torch.int32)
    sl = torch.tensor(seq_lens, dtype=torch.int32)
    return q, kc, vc, bt.to(DEV), sl.to(DEV), qlocs.to(DEV), sl, qlocs, bt


def test_attn_decode_matches_ref():
    require_gpu()
    seq_lens = [1, 31, 32, 33, 100, 257]
    new_counts = [1] * len(seq_lens)
    q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(seq_lens, new_counts)
    scale = 1.0 / math.sqrt(128)
    got = ops.paged_attention(q, kc, vc, bt, sl, ql, scale,
                              num_decodes_tail=len(seq_lens),
                              prefill_token_count=0)
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(), vc.float().cpu(),
                            bt_c, sl_c, ql_c, scale)
    assert torch.allclose(got.float().cpu(), ref, atol=3e-2, rtol=3e-2), (
        (got.float().cpu() - ref).abs().max()
    )


def test_attn_prefill_matches_ref():
    require_gpu()
    # mixed: fresh prefill (ctx=0), chunked continuation (ctx>0), ragged sizes
    seq_lens = [40, 64, 100, 7]
    new_counts = [40, 32, 33, 7]
    q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(seq_lens, new_counts)
    tiles_s, tiles_q0 = [], []
    for s, c in enumerate(new_counts):
        for j in range(0, c, 32):
            tiles_s.append(s)
            tiles_q0.append(j)
    scale = 1.0 / math.sqrt(128)
    got = ops.paged_attention(
        q, kc, vc, bt, sl, ql, scale, num_decodes_tail=0,
        tile_seq=torch.tensor(tiles_s, dtype=torch.int32, device=DEV),
        tile_q0=torch.tensor(tiles_q0, dtype=torch.int32, device=DEV),
        prefill_token_count=int(sum(new_counts)))
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(), vc.float().cpu(),
                            bt_c, sl_c, ql_c, scale)
    err = (got.float().cpu() - ref).abs().max()
    assert torch.allclose(got.float().cpu(), ref, atol=5e-2, rtol=5e-2), err


def test_attn_mixed_batch():
    require_gpu()
    seq_lens = [50, 90, 33, 65, 129]
    new_counts = [50, 40, 1, 1, 1]  # 2 prefills + 3 decodes
    q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(seq_lens, new_counts)
    tiles_s, tiles_q0 = [], []
    for s in range(2):
        for j in range(0, new_counts[s], 32):
            tiles_s.append(s)
            tiles_q0.append(j)
    scale = 1.0 / math.sqrt(128)
    got = ops.paged_attention(
        q, kc, vc, bt, sl, ql, scale, num_decodes_tail=3,
        tile_seq=torch.tensor(tiles_s, dtype=torch.int32, device=DEV),
        tile_q0=torch.tensor(tiles_q0, dtype=torch.int32, device=DEV),
        prefill_token_count=90)
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(), vc.float().cpu(),
                            bt_c, sl_c, ql_c, scale)
    assert torch.allclose(got.float().cpu(), ref, atol=5e-2, rtol=5e-2)


def test_gqa_group_sizes():
    require_gpu()
    for hq, hk in ((8, 8), (8, 4), (16, 2)):
        seq_lens = [70, 33]
        q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(
            seq_lens, [1, 1], Hq=hq, Hk=hk)
        scale = 1.0 / math.sqrt(128)
        got = ops.paged_attention(q, kc, vc, bt, sl, ql, scale,
                                  num_decodes_tail=2, prefill_token_count=0)
        ref = R.paged_attention(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), bt_c, sl_c, ql_c, scale)
        assert torch.allclose(got.float().cpu(), ref, atol=3e-2, rtol=3e-2), (hq, hk)


def test_mean_pool_normalize():
    require_gpu()
    h = bf(torch.randn(50, 1024))
    locs = torch.tensor([0, 10, 11, 50], dtype=torch.int32, device=DEV)
    got = ops.mean_pool_normalize(h, locs).cpu()
    ref = R.mean_pool_normalize(h.float().cpu(), locs.cpu())
    assert torch.allclose(got, ref, atol=2e-2, rtol=2e-2)


def test_engine_e2e_gpu():
    require_gpu()
    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import get_model_spec

    cfg = EngineConfig(spec=get_model_spec("qwen-3-0.6b"), device="cuda",
                       max_model_len=512, num_kv_blocks=512,
                       max_tokens_per_step=2048, max_num_seqs=64)
    eng = LLMEngine(cfg)
    outs = eng.generate([f"row {i}" for i in range(8)],
                        sampling=SamplingParams(max_tokens=16, temperature=0.8))
    assert len(outs) == 8
    assert eng.total_output_tokens >= 8


def test_engine_guided_json_gpu():
    require_gpu()
    import json

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import get_model_spec

    cfg = EngineConfig(spec=get_model_spec("qwen-3-0.6b"), device="cuda",
                       max_model_len=2048, num_kv_blocks=512,
                       max_tokens_per_step=2048)
    eng = LLMEngine(cfg)
    schema = {"type": "object", "properties": {
        "label": {"enum": ["A", "B"]},
        "score": {"type": "integer", "minimum": 0, "maximum": 9}}}
    outs = eng.generate(["classify"],
                        sampling=SamplingParams(max_tokens=512, temperature=1.0),
                        schema=schema)
    d = json.loads(outs[0])
    assert d["label"] in ("A", "B") and 0 <= d["score"] <= 9


def test_attn_head_dim_64():
    """gpt-oss/embeddinggemma class models: head_dim 64 paths."""
    require_gpu()
    # decode
    q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(
        [70, 33], [1, 1], Hq=8, Hk=2, D=64)
    scale = 1.0 / math.sqrt(64)
    got = ops.paged_attention(q, kc, vc, bt, sl, ql, scale,
                              num_decodes_tail=2, prefill_token_count=0)
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(),
                            vc.float().cpu(), bt_c, sl_c, ql_c, scale)
    assert torch.allclose(got.float().cpu(), ref, atol=3e-2, rtol=3e-2)
    # prefill
    q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(
        [50, 40], [50, 40], Hq=8, Hk=2, D=64, seed=11)
    tiles_s, tiles_q0 = [], []
    for s, c in enumerate([50, 40]):
        for j in range(0, c, 32):
            tiles_s.append(s)
            tiles_q0.append(j)
    got = ops.paged_attention(
        q, kc, vc, bt, sl, ql, scale, num_decodes_tail=0,
        tile_seq=torch.tensor(tiles_s, dtype=torch.int32, device=DEV),
        tile_q0=torch.tensor(tiles_q0, dtype=torch.int32, device=DEV),
        prefill_token_count=90)
    ref = R.paged_attention(q.float().cpu(), kc.float().cpu(),
                            vc.float().cpu(), bt_c, sl_c, ql_c, scale)
    assert torch.allclose(got.float().cpu(), ref, atol=5e-2, rtol=5e-2)


def test_fp8_kv_attention():
    """e4m3 KV cache: kernels vs CPU reference using the SAME quantized cache
    (both sides round-trip through torch.float8_e4m3fn, so tolerance stays
    bf16-tight)."""
    require_gpu()
    torch.manual_seed(13)
    bs, Hq, Hk, D = 32, 8, 2, 128
    seq_lens = [70, 40, 33, 1]
    new_counts = [70, 40, 1, 1]   # 2 prefills + 2 decodes
    S = len(seq_lens)
    nb = sum((L + bs - 1) // bs for L in seq_lens) + 2
    kc = torch.zeros(nb, Hk, bs, D, dtype=torch.float8_e4m3fn, device=DEV)
    vc = torch.zeros_like(kc)
    perm = torch.randperm(nb).tolist()
    tables, i = [], 0
    for L in seq_lens:
        n = (L + bs - 1) // bs
        tables.append(perm[i:i + n]); i += n
    max_b = max(len(t) for t in tables)
    bt = torch.zeros(S, max_b, dtype=torch.int32)
    for s, t in enumerate(tables):
        bt[s, :len(t)] = torch.tensor(t, dtype=torch.int32)
    qs = []
    for s, L in enumerate(seq_lens):
        K = bf(torch.randn(L, Hk, D))
        V = bf(torch.randn(L, Hk, D))
        slots = torch.tensor([tables[s][p // bs] * bs + p % bs
                              for p in range(L)], dtype=torch.long)
        R.write_kv_cache(K, V, kc, vc, slots.to(DEV))
        qs.append(bf(torch.randn(new_counts[s], Hq, D)))
    q = torch.cat(qs, 0)
    qlocs = torch.tensor([0] + list(torch.tensor(new_counts).cumsum(0)),
                         dtype=torch.int32)
    sl = torch.tensor(seq_lens, dtype=torch.int32)
    tiles_s, tiles_q0 = [], []
    for s in range(2):
        for j in range(0, new_counts[s], 32):
            tiles_s.append(s); tiles_q0.append(j)
    scale = 1.0 / math.sqrt(D)
    got = ops.paged_attention(
        q, kc, vc, bt.to(DEV), sl.to(DEV), qlocs.to(DEV), scale,
        num_decodes_tail=2,
        tile_seq=torch.tensor(tiles_s, dtype=torch.int32, device=DEV),
        tile_q0=torch.tensor(tiles_q0, dtype=torch.int32, device=DEV),
        prefill_token_count=110)
    ref = R.paged_attention(q.float().cpu(), kc.cpu().float(),
                            vc.cpu().float(), bt, sl, qlocs, scale)
    assert torch.allclose(got.float().cpu(), ref, atol=5e-2, rtol=5e-2), (
        (got.float().cpu() - ref).abs().max())


def test_fp8_qkv_prep_write():
    """qkv_prep writes e4m3 K/V codes identical to torch's RNE cast."""
    require_gpu()
    torch.manual_seed(17)
    T, Hq, Hk, D, bs = 15, 4, 2, 128, 32
    qkv = bf(torch.randn(T, (Hq + 2 * Hk) * D))
    pos = torch.randint(0, 100, (T,), dtype=torch.long)
    slots = torch.randperm(2 * bs)[:T].to(torch.long)
    cs = R.rope_cos_sin(128, D, 1e6).to(DEV)
    kc = torch.zeros(2, Hk, bs, D, dtype=torch.float8_e4m3fn, device=DEV)
    vc = torch.zeros_like(kc)
    q_out = ops.fused_qkv_prep(qkv, Hq, Hk, D, pos.to(DEV), slots.to(DEV),
                               kc, vc, cs, None, None, 1e-6)
    # CPU composition with the same fp8 cache dtype
    kc_ref = torch.zeros(2, Hk, bs, D, dtype=torch.float8_e4m3fn)
    vc_ref = torch.zeros_like(kc_ref)
    q_ref = ops.fused_qkv_prep(qkv.cpu(), Hq, Hk, D, pos, slots, kc_ref,
                               vc_ref, cs.cpu(), None, None, 1e-6)
    assert torch.allclose(q_out.float().cpu(), q_ref.float(), atol=3e-2,
                          rtol=3e-2)
    # e4m3 codes agree except where 1-ulp fp32 RoPE differences land on a
    # quantization boundary: bound the mismatch FRACTION and magnitude
    for got_c, ref_c in ((kc, kc_ref), (vc, vc_ref)):
        diff = (got_c.cpu().float() - ref_c.float()).abs()
        frac = (diff > 1e-6).float().mean().item()
        assert frac < 0.01, frac
        assert diff.max() < 0.5, diff.max()


def test_engine_e2e_fp8_gpu():
    require_gpu()
    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import get_model_spec

    cfg = EngineConfig(spec=get_model_spec("qwen-3-0.6b"), device="cuda",
                       max_model_len=512, num_kv_blocks=512,
                       max_tokens_per_step=2048, max_num_seqs=64,
                       kv_dtype="fp8_e4m3")
    eng = LLMEngine(cfg)
    outs = eng.generate([f"fp8 row {i}" for i in range(4)],
                        sampling=SamplingParams(max_tokens=12, temperature=0.8))
    assert len(outs) == 4 and eng.total_output_tokens >= 4


def test_mfma16_layout_probe():
    require_gpu()
    torch.manual_seed(21)
    a = torch.randn(16, 32)
    b = torch.randn(32, 16) * torch.arange(1, 17).float() / 4.0
    c = _C.mfma16_probe(bf(a), bf(b))
    ref = (bf(a).float() @ bf(b).float())
    assert torch.allclose(c.cpu(), ref.cpu(), atol=2e-2, rtol=2e-2), (
        (c.cpu() - ref.cpu()).abs().max())


def test_attn_decode_mfma_matches_ref(monkeypatch):
    require_gpu()
    monkeypatch.setenv("SUTRO_DECODE_MFMA", "1")
    for hq, hk in ((8, 1), (8, 2), (16, 2)):
        seq_lens = [1, 31, 32, 33, 100, 257]
        q, kc, vc, bt, sl, ql, sl_c, ql_c, bt_c = _attn_setup(
            seq_lens, [1] * len(seq_lens), Hq=hq, Hk=hk, seed=23)
        scale = 1.0 / math.sqrt(128)
        got = ops.paged_attention(q, kc, vc, bt, sl, ql, scale,
                                  num_decodes_tail=len(seq_lens),
                                  prefill_token_count=0)
        ref = R.paged_attention(q.float().cpu(), kc.float().cpu(),
                                vc.float().cpu(), bt_c, sl_c, ql_c, scale)
        err = (got.float().cpu() - ref).abs().max()
        assert torch.allclose(got.float().cpu(), ref, atol=3e-2,
                              rtol=3e-2), (hq, hk, err)


@pytest.mark.gpu
def test_gemm_tn_matches_torch():
    """gemm_tn (MFMA glds double-buffer) vs fp32 torch reference across tile
    configs, XCD swizzle, N-edge and the fused residual epilogue."""
    from sutro_amd import _C

    torch.manual_seed(7)
    for (M, N, K), cfgs in [
        ((256, 512, 256), [(256, 256), (128, 256), (128, 128), (256, 128)]),
        ((1024, 1000, 640), [(256, 256), (128, 128)]),  # N edge, K%128 != 0
        ((128, 256, 128), [(128, 128)]),
    ]:
        x = (torch.randn(M, K, device="cuda") * 0.5).bfloat16()
        w = (torch.randn(N, K, device="cuda") * 0.1).bfloat16()
        ref = x.float() @ w.t().float()
        scale = ref.abs().max().item()
        for bm, bn in cfgs:
            if M % bm:
                continue
            for swz in (0, 1, 2):
                for xswz in (0, 1):
                    out = _C.gemm_tn(x, w, None, bm, bn, swz, xswz)
                    rel = (out.float() - ref).abs().max().item() / scale
                    assert rel < 2e-2, (M, N, K, bm, bn, swz, xswz, rel)

    # fused residual
    x = (torch.randn(256, 384, device="cuda") * 0.5).bfloat16()
    w = (torch.randn(512, 384, device="cuda") * 0.1).bfloat16()
    res = torch.randn(256, 512, device="cuda").bfloat16()
    out = _C.gemm_tn(x, w, res, 128, 128, 2, 0)
    ref = x.float() @ w.t().float() + res.float()
    rel = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert rel < 2e-2
