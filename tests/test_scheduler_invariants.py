"""Property-based stress of the scheduler + paged-KV accounting.

Random interleavings of submissions, steps and finishes must preserve:
- a request is never scheduled twice in one batch;
- every scheduled prefill chunk is in-bounds and contiguous;
- decode requests satisfy num_computed == total_len - 1;
- KV block accounting: free + held == total, no block held twice;
- all work eventually drains (no starvation / lost requests).
"""

import torch
from hypothesis import given, settings, strategies as st

from sutro_amd.engine.config import EngineConfig
from sutro_amd.engine.kv_cache import PagedKVCache
from sutro_amd.engine.request import FinishReason, Request, SamplingParams
from sutro_amd.engine.scheduler import Scheduler
from sutro_amd.models.registry import tiny_spec_for_tests


def mk_env(num_blocks, max_seqs, budget, thr=0):
    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_num_seqs=max_seqs, max_model_len=256,
                       max_tokens_per_step=budget,
                       min_prefill_batch_tokens=thr,
                       num_kv_blocks=num_blocks)
    kv = PagedKVCache(num_layers=1, num_blocks=num_blocks, num_kv_heads=1,
                      block_size=cfg.kv_block_size, head_dim=8,
                      dtype=torch.float32, device="cpu")
    return cfg, kv, Scheduler(cfg, kv)


def check_accounting(sch, kv, all_reqs):
    held = []
    for t in kv.block_tables.values():
        held.extend(t)
    free = kv.allocator._free
    assert len(held) + len(free) == kv.allocator.num_blocks
    assert len(set(held)) == len(held), "block held twice"
    assert len(set(free)) == len(free), "block double-freed"


@settings(max_examples=40, deadline=None)
@given(st.data())
def test_scheduler_random_interleaving(data):
    num_blocks = data.draw(st.integers(6, 48), label="blocks")
    max_seqs = data.draw(st.integers(1, 12), label="max_seqs")
    budget = data.draw(st.integers(8, 128), label="budget")
    thr = data.draw(st.sampled_from([0, 16, None]), label="thr")
    cfg, kv, sch = mk_env(num_blocks, max_seqs, budget,
                          thr if thr is not None else 0)
    if thr is None:
        cfg.min_prefill_batch_tokens = None

    rid = [0]
    all_reqs = []

    def submit():
        plen = data.draw(st.integers(1, 96), label="plen")
        mt = data.draw(st.integers(1, 12), label="max_tokens")
        pri = data.draw(st.integers(0, 1), label="pri")
        r = Request(req_id=rid[0], prompt_token_ids=list(range(3, 3 + plen)),
                    sampling=SamplingParams(max_tokens=mt))
        rid[0] += 1
        all_reqs.append(r)
        sch.add_request(r, priority=pri)

    def run_step():
        sb = sch.schedule()
        seen = set()
        for req, c in zip(sb.reqs, sb.num_new_tokens):
            assert id(req) not in seen, "request scheduled twice in one batch"
            seen.add(id(req))
            if req.in_prefill:
                assert 1 <= c <= req.num_prompt_tokens - req.num_computed_tokens
            else:
                assert c == 1
                assert req.num_computed_tokens == req.total_len - 1
            # the KV table must cover every position this step touches
            table = kv.block_tables[req.req_id]
            assert (req.num_computed_tokens + c) <= len(table) * kv.block_size
        # emulate the engine: advance, sample one token for completing rows,
        # finish rows that hit max_tokens
        for req, c in zip(sb.reqs, sb.num_new_tokens):
            req.num_computed_tokens += c
            if not req.in_prefill and req.num_computed_tokens == req.total_len:
                req.output_token_ids.append(65)
                if len(req.output_token_ids) >= req.sampling.max_tokens:
                    sch.finish(req, FinishReason.LENGTH)
        check_accounting(sch, kv, all_reqs)
        return sb

    n_ops = data.draw(st.integers(5, 30), label="n_ops")
    for _ in range(n_ops):
        op = data.draw(st.sampled_from(["submit", "abort", "step", "step",
                                        "step"]), label="op")
        if op == "submit" and rid[0] < 24:
            submit()
        elif op == "abort":
            live = [r for r in all_reqs if r.finish_reason is None]
            if live:
                sch.abort_request(data.draw(st.sampled_from(live),
                                            label="victim"))
                check_accounting(sch, kv, all_reqs)
            else:
                run_step()
        else:
            run_step()

    # drain: everything must finish within a bounded number of steps
    for _ in range(800):
        if not sch.has_work():
            break
        run_step()
    assert not sch.has_work(), (
        f"undrained: running={len(sch.running)} "
        f"p0={len(sch.waiting_p0)} p1={len(sch.waiting_p1)}")
    for r in all_reqs:
        assert r.finish_reason is not None
    # all KV released at the end
    assert kv.allocator.num_free == kv.allocator.num_blocks
