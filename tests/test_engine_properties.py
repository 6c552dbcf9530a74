"""Property-based stress of the FULL engine on CPU: random rows with random
feature combinations (seeds, stop strings, stop tokens, FSM guidance,
priorities, aborts, async decode) must drain to terminal states that honor
every per-row contract."""

import json

from hypothesis import given, settings, strategies as st

from sutro_amd.engine.config import EngineConfig
from sutro_amd.engine.engine import LLMEngine
from sutro_amd.engine.request import SamplingParams
from sutro_amd.engine.tokenizer import get_tokenizer
from sutro_amd.models.registry import tiny_spec_for_tests

SCHEMA = {"type": "object",
          "properties": {"k": {"type": "integer", "minimum": 0,
                               "maximum": 99}}}


@settings(max_examples=25, deadline=None)
@given(st.data())
def test_engine_random_workload(data):
    async_mode = data.draw(st.booleans(), label="async")
    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_num_seqs=data.draw(st.integers(2, 8), label="seqs"),
                       max_model_len=256,
                       max_tokens_per_step=data.draw(
                           st.sampled_from([32, 64, 128]), label="budget"),
                       min_prefill_batch_tokens=data.draw(
                           st.sampled_from([0, 32]), label="thr"),
                       num_kv_blocks=data.draw(st.integers(12, 64),
                                               label="blocks"),
                       seed=7, async_decode=async_mode)
    eng = LLMEngine(cfg)
    fsm_id = eng.register_fsm(SCHEMA)
    tok = get_tokenizer()

    rows = []
    n_rows = data.draw(st.integers(1, 8), label="rows")
    for i in range(n_rows):
        guided = data.draw(st.booleans(), label="guided")
        sp = SamplingParams(
            max_tokens=data.draw(st.integers(1, 24), label="max_tokens"),
            temperature=data.draw(st.sampled_from([0.0, 0.7, 1.3]),
                                  label="temp"),
            seed=data.draw(st.one_of(st.none(), st.integers(0, 9)),
                           label="seed"),
            stop_token_ids=data.draw(
                st.one_of(st.none(), st.lists(st.integers(3, 258),
                                              max_size=2)), label="stops"),
            stop=None if guided else data.draw(
                st.one_of(st.none(), st.just(["zz"])), label="stopstr"),
        )
        req = eng.add_request(
            list(range(3, 3 + data.draw(st.integers(1, 60), label="plen"))),
            sp, fsm_id=fsm_id if guided else None,
            priority=data.draw(st.integers(0, 1), label="pri"))
        rows.append((req, guided, sp))

    aborted = set()
    steps = 0
    while eng.has_work() and steps < 4000:
        steps += 1
        eng.step()
        if (steps == 3 and len(rows) > 2
                and data.draw(st.booleans(), label="do_abort")):
            victim = rows[data.draw(st.integers(0, len(rows) - 1),
                                    label="victim")][0]
            if victim.finish_reason is None:
                eng.abort_request(victim)
                aborted.add(victim.req_id)
    eng.step()  # flush async lag
    assert not eng.has_work(), "engine failed to drain"

    for req, guided, sp in rows:
        assert req.finish_reason is not None
        if req.req_id in aborted:
            continue
        assert len(req.output_token_ids) <= sp.max_tokens
        out = req.output_token_ids
        if sp.stop_token_ids and not guided:
            # no stop token may REMAIN in the output (it ends the row);
            # guided rows honor only EOS (schema validity wins)
            assert not (set(out) & set(sp.stop_token_ids))
        if sp.stop and not guided:
            assert not tok.decode(out).endswith("zz") or not out
        if guided and req.finish_reason.value == "stop":
            text = tok.decode(out)
            parsed = json.loads(text)
            assert 0 <= parsed["k"] <= 99
