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
