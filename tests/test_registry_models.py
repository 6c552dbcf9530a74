"""Every registry model must instantiate and generate (CPU dev proxies)."""

import pytest

from sutro_amd.engine.config import EngineConfig
from sutro_amd.engine.engine import LLMEngine
from sutro_amd.engine.request import SamplingParams
from sutro_amd.models.registry import MODEL_REGISTRY
from sutro_amd.service.models_map import dev_proxy


@pytest.mark.parametrize("name", sorted(MODEL_REGISTRY))
def test_model_generates(name):
    spec = dev_proxy(MODEL_REGISTRY[name])
    cfg = EngineConfig(spec=spec, device="cpu", max_model_len=128,
                       num_kv_blocks=32, max_tokens_per_step=64)
    eng = LLMEngine(cfg)
    if spec.embedding:
        req = eng.add_request(eng.tokenizer.encode("embed me"),
                              SamplingParams())
        while eng.has_work():
            eng.step()
        assert eng.embeddings[req.req_id].shape == (spec.hidden_size,)
    else:
        outs = eng.generate(["registry row"],
                            sampling=SamplingParams(max_tokens=4,
                                                    temperature=0.5))
        assert len(outs) == 1
        assert eng.total_output_tokens >= 1


def test_registry_specs_are_coherent():
    """Structural sanity across ALL 40 servable specs: GQA divisibility, TP
    divisibility for the recommended degree, MoE fields paired."""
    from sutro_amd.models.registry import MODEL_REGISTRY

    for name, s in MODEL_REGISTRY.items():
        assert s.num_heads % s.num_kv_heads == 0, name
        assert s.head_dim in (64, 128), name
        assert s.hidden_size % s.num_heads != -1  # defined
        tp = s.recommended_tp
        assert tp >= 1 and s.num_heads % tp == 0, (name, tp)
        assert s.num_kv_heads % tp == 0 or s.num_kv_heads < tp, (name, tp)
        if s.num_experts:
            assert s.experts_per_token >= 1, name
            assert s.moe_intermediate_size > 0, name
        else:
            assert s.intermediate_size > 0, name
        assert s.max_context >= 2048, name
        assert s.param_count() > 0 and s.active_param_count() > 0, name
        assert s.active_param_count() <= s.param_count(), name
