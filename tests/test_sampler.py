"""Sampler correctness: greedy, masks, seeds, logprobs, top-k/top-p."""

import math

import numpy as np
import torch

from sutro_amd.engine.request import Request, SamplingParams
from sutro_amd.engine.sampler import Sampler, seeded_uniform


def _reqs(n, **kw):
    return [Request(req_id=i, prompt_token_ids=[3, 4],
                    sampling=SamplingParams(**kw)) for i in range(n)]


def test_greedy_is_argmax():
    s = Sampler("cpu", vocab_limit=16)
    logits = torch.randn(4, 32)
    toks, lps = s.sample(logits, _reqs(4, temperature=0.0))
    assert toks == logits[:, :16].argmax(dim=-1).tolist()


def test_logprob_matches_log_softmax():
    s = Sampler("cpu", vocab_limit=8)
    logits = torch.randn(1, 8)
    toks, lps = s.sample(logits, _reqs(1, temperature=0.0))
    ref = torch.log_softmax(logits[0, :8], dim=-1)[toks[0]].item()
    assert abs(lps[0] - ref) < 1e-5


def test_fsm_mask_restricts_support():
    s = Sampler("cpu", vocab_limit=8)
    logits = torch.zeros(3, 8)
    mask = torch.zeros(3, 8, dtype=torch.bool)
    mask[:, 5] = True
    toks, _ = s.sample(logits, _reqs(3, temperature=1.0), fsm_mask=mask)
    assert toks == [5, 5, 5]


def test_top_k_restricts_support():
    s = Sampler("cpu", vocab_limit=16)
    logits = torch.arange(16, dtype=torch.float32).repeat(64, 1)
    toks, _ = s.sample(logits, _reqs(64, temperature=1.0, top_k=2))
    assert set(toks).issubset({14, 15})


def test_top_p_restricts_support():
    s = Sampler("cpu", vocab_limit=4)
    # probs ~ [0.97, 0.01, 0.01, 0.01] with top_p=0.5 -> always token 0
    logits = torch.tensor([[8.0, 0.0, 0.0, 0.0]]).repeat(32, 1)
    toks, _ = s.sample(logits, _reqs(32, temperature=1.0, top_p=0.5))
    assert set(toks) == {0}


def test_seeded_stream_deterministic():
    u1 = seeded_uniform(np.array([7, 7], dtype=np.uint64),
                        np.array([1, 2], dtype=np.uint64))
    u2 = seeded_uniform(np.array([7, 7], dtype=np.uint64),
                        np.array([1, 2], dtype=np.uint64))
    assert (u1 == u2).all()
    assert u1[0] != u1[1]
    assert ((0 <= u1) & (u1 < 1)).all()


def test_temperature_sampling_distribution():
    """Sampling should roughly follow softmax probabilities."""
    s = Sampler("cpu", seed=123, vocab_limit=4)
    logits = torch.tensor([[2.0, 1.0, 0.0, -1.0]]).repeat(2000, 1)
    toks, _ = s.sample(logits, _reqs(2000, temperature=1.0))
    counts = np.bincount(toks, minlength=4) / 2000
    expect = torch.softmax(torch.tensor([2.0, 1.0, 0.0, -1.0]), 0).numpy()
    assert np.abs(counts - expect).max() < 0.05


def test_cumulative_logprob_confidence_bounds():
    lp = -0.5
    conf = math.exp(lp)
    assert 0 < conf < 1


def test_sampler_support_properties():
    """Property: sampled tokens always lie inside the top-k/top-p restricted
    support of the (possibly FSM-masked) distribution."""
    from hypothesis import given, settings, strategies as st

    from sutro_amd.engine.request import Request, SamplingParams
    from sutro_amd.engine.sampler import Sampler

    @settings(max_examples=50, deadline=None)
    @given(st.data())
    def run(data):
        n = data.draw(st.integers(1, 8))
        vl = 32
        torch.manual_seed(data.draw(st.integers(0, 10_000)))
        logits = torch.randn(n, vl) * data.draw(st.floats(0.5, 4.0))
        reqs = []
        for i in range(n):
            reqs.append(Request(
                req_id=i, prompt_token_ids=[3, 4],
                sampling=SamplingParams(
                    temperature=data.draw(st.floats(0.0, 2.0)),
                    top_p=data.draw(st.floats(0.05, 1.0)),
                    top_k=data.draw(st.sampled_from([0, 1, 3, 8])),
                    seed=data.draw(st.one_of(st.none(), st.integers(0, 99))),
                    max_tokens=4)))
        s = Sampler("cpu", vocab_limit=vl)
        toks, lps = s.sample(logits, reqs)
        for i, (tok, lp) in enumerate(zip(toks, lps)):
            sp = reqs[i].sampling
            row = logits[i]
            assert 0 <= tok < vl
            assert lp <= 1e-5  # log-probability
            if sp.temperature < 1e-5:
                assert tok == int(row.argmax())
                continue
            order = torch.argsort(row, descending=True).tolist()
            k = sp.top_k if sp.top_k > 0 else vl
            topk = set(order[:k])
            assert tok in topk, "outside top-k"
            # top-p: the kept prefix (smallest set with mass >= p) always
            # includes rank 0; token must be inside the kept prefix
            probs = torch.softmax(row / sp.temperature, -1)
            sp_sorted = probs[order]
            cdf = torch.cumsum(sp_sorted, 0)
            keep = int((cdf - sp_sorted < sp.top_p).sum())
            kept = set(order[:max(1, keep)])
            assert tok in kept & topk, "outside top-p support"

    run()


def test_seeded_sampling_independent_of_batch_composition():
    """Property: a seeded row draws the same token regardless of which other
    rows share the batch."""
    from sutro_amd.engine.request import Request, SamplingParams
    from sutro_amd.engine.sampler import Sampler

    vl = 24
    torch.manual_seed(5)
    base_logits = torch.randn(1, vl)

    def draw(extra_rows):
        s = Sampler("cpu", vocab_limit=vl)
        reqs = [Request(req_id=0, prompt_token_ids=[3, 4, 5],
                        sampling=SamplingParams(temperature=0.9, seed=1234,
                                                max_tokens=4))]
        logits = [base_logits]
        for i in range(extra_rows):
            reqs.append(Request(req_id=100 + i, prompt_token_ids=[3],
                                sampling=SamplingParams(temperature=1.1,
                                                        seed=None,
                                                        max_tokens=4)))
            logits.append(torch.randn(1, vl))
        toks, _ = s.sample(torch.cat(logits), reqs)
        return toks[0]

    ref = draw(0)
    for extra in (1, 3, 7):
        assert draw(extra) == ref


def test_threshold_semantics_ties_all_kept():
    """Value-threshold top-k/top-p: ALL ties at a boundary are kept (the
    sort-free semantics the fused kernel implements — no sort-order
    tie-breaking)."""
    from sutro_amd.engine.sampler import sample_torch_reference

    # four-way tie at the top; top_k=2 must keep all four tied tokens
    logits = torch.tensor([[5.0, 5.0, 5.0, 5.0, 1.0, 0.0]]).repeat(512, 1)
    n = logits.shape[0]
    temps = torch.ones(n)
    top_ps = torch.ones(n)
    top_ks = torch.full((n,), 2, dtype=torch.int32)
    g = torch.Generator().manual_seed(0)
    u = torch.rand(n, generator=g)
    toks, _ = sample_torch_reference(logits, temps, top_ps, top_ks, u, 6)
    seen = set(toks.tolist())
    assert seen <= {0, 1, 2, 3}          # never the non-tied tail
    assert len(seen) == 4                 # every tied token reachable


def test_threshold_semantics_top_p_mass_gt():
    """keep_i iff mass{s_j > s_i} < top_p * Z: the boundary token itself is
    kept even when the strictly-greater mass is just below the threshold."""
    from sutro_amd.engine.sampler import sample_torch_reference

    # probs ~ [0.6, 0.3, 0.1]: top_p=0.7 -> mass_gt(token1)=0.6 < 0.7 keeps
    # token 1; mass_gt(token2)=0.9 >= 0.7 drops token 2
    p = torch.tensor([0.6, 0.3, 0.1])
    logits = torch.log(p).unsqueeze(0).repeat(2048, 1)
    n = logits.shape[0]
    temps = torch.ones(n)
    top_ps = torch.full((n,), 0.7)
    top_ks = torch.full((n,), 3, dtype=torch.int32)
    g = torch.Generator().manual_seed(1)
    u = torch.rand(n, generator=g)
    toks, _ = sample_torch_reference(logits, temps, top_ps, top_ks, u, 3)
    seen = set(toks.tolist())
    assert 2 not in seen
    assert seen == {0, 1}


def test_reference_matches_bruteforce_oracle():
    """sample_torch_reference vs a direct O(V^2) transcription of the
    documented semantics (independent oracle; quantized logits keep float
    boundaries well-separated so comparisons are exact)."""
    import numpy as np

    from sutro_amd.engine.sampler import sample_torch_reference

    rng = np.random.default_rng(7)
    for trial in range(200):
        vl = int(rng.integers(3, 24))
        logits = torch.tensor(
            rng.integers(-6, 7, size=vl).astype(np.float32) / 2.0)
        temp = float(rng.choice([0.5, 1.0, 1.7]))
        top_p = float(rng.choice([0.3, 0.7, 0.95, 1.0]))
        top_k = int(rng.choice([1, 2, 5, vl]))
        u = float(rng.uniform(0.01, 0.99))

        # oracle
        s = (logits - logits.max()) / temp
        p = torch.exp(s)
        z = float(p.sum())
        keep = []
        for i in range(vl):
            cnt_gt = sum(1 for j in range(vl) if s[j] > s[i])
            mass_gt = sum(float(p[j]) for j in range(vl) if s[j] > s[i])
            keep.append(cnt_gt < top_k and mass_gt < top_p * z)
        cum, tok_oracle = 0.0, None
        target = u * sum(float(p[i]) for i in range(vl) if keep[i])
        for i in range(vl):
            if keep[i]:
                cum += float(p[i])
                if cum > target:
                    tok_oracle = i
                    break
        if tok_oracle is None:
            tok_oracle = max(i for i in range(vl) if keep[i])

        toks, _ = sample_torch_reference(
            logits.unsqueeze(0),
            torch.tensor([temp]), torch.tensor([top_p]),
            torch.tensor([top_k], dtype=torch.int32),
            torch.tensor([u]), vl)
        assert int(toks[0]) == tok_oracle, (
            f"trial {trial}: vl={vl} T={temp} p={top_p} k={top_k} u={u} "
            f"logits={logits.tolist()}")
