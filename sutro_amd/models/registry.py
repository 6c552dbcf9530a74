"""Model registry: servable model names -> concrete architecture specs.

The reference keeps only a flat Literal union of names "kept in sync with the
backend" (`/root/reference/sutro/common.py:13-50`). Here the backend is in-repo,
so every name resolves to a real :class:`ModelSpec` the engine can instantiate
(random-init weights; there is no network for checkpoints in this environment,
but ``weights_path`` may point at a local safetensors dir).

Architecture family is MI355X-first: all dense/MoE decoder models share one
Qwen3-style block (RMSNorm, RoPE-NeoX, GQA, SwiGLU) so a single set of CDNA4
HIP kernels covers the whole registry.
"""

from __future__ import annotations

from dataclasses import dataclass, replace
from typing import Dict


@dataclass(frozen=True)
class ModelSpec:
    name: str
    arch: str = "qwen3"  # "qwen3" (dense), "qwen3_moe", "embedding"
    hidden_size: int = 1024
    num_layers: int = 28
    num_heads: int = 16
    num_kv_heads: int = 8
    head_dim: int = 128
    intermediate_size: int = 3072
    vocab_size: int = 151936
    rope_theta: float = 1_000_000.0
    rms_eps: float = 1e-6
    max_context: int = 32768
    tie_embeddings: bool = False
    qk_norm: bool = True           # Qwen3 applies RMSNorm to q/k heads
    # MoE
    num_experts: int = 0
    experts_per_token: int = 0
    moe_intermediate_size: int = 0
    # behavior flags
    reasoning: bool = False        # "-thinking" models emit {content, reasoning_content}
    embedding: bool = False        # embedding models: mean-pool + L2 normalize, no decode
    # parallelism default
    recommended_tp: int = 1

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def param_count(self) -> int:
        """Approximate parameter count (for memory budgeting / cost estimates)."""
        h, L = self.hidden_size, self.num_layers
        emb = self.vocab_size * h * (1 if self.tie_embeddings else 2)
        attn = h * (self.q_size + 2 * self.kv_size) + self.q_size * h
        if self.num_experts > 0:
            mlp = 3 * h * self.moe_intermediate_size * self.num_experts + h * self.num_experts
        else:
            mlp = 3 * h * self.intermediate_size
        norms = 2 * h
        return emb + L * (attn + mlp + norms) + h

    def active_param_count(self) -> int:
        """Params touched per token (MoE: only routed experts)."""
        if self.num_experts == 0:
            return self.param_count()
        h, L = self.hidden_size, self.num_layers
        emb = self.vocab_size * h * (1 if self.tie_embeddings else 2)
        attn = h * (self.q_size + 2 * self.kv_size) + self.q_size * h
        mlp = 3 * h * self.moe_intermediate_size * self.experts_per_token
        return emb + L * (attn + mlp + 2 * h) + h


def _dense(name: str, h: int, L: int, heads: int, kv: int, inter: int, *,
           tie: bool = False, vocab: int = 151936, hd: int = 128, tp: int = 1,
           **kw) -> ModelSpec:
    return ModelSpec(name=name, arch="qwen3", hidden_size=h, num_layers=L,
                     num_heads=heads, num_kv_heads=kv, head_dim=hd,
                     intermediate_size=inter, vocab_size=vocab,
                     tie_embeddings=tie, recommended_tp=tp, **kw)


def _moe(name: str, h: int, L: int, heads: int, kv: int, *, experts: int,
         topk: int, moe_inter: int, vocab: int = 151936, hd: int = 128,
         tp: int = 1, **kw) -> ModelSpec:
    return ModelSpec(name=name, arch="qwen3_moe", hidden_size=h, num_layers=L,
                     num_heads=heads, num_kv_heads=kv, head_dim=hd,
                     intermediate_size=0, vocab_size=vocab,
                     num_experts=experts, experts_per_token=topk,
                     moe_intermediate_size=moe_inter, recommended_tp=tp, **kw)


_BASE: Dict[str, ModelSpec] = {}


def _reg(spec: ModelSpec) -> None:
    _BASE[spec.name] = spec


# ---- benchmark / north-star models (BASELINE.json configs) ----
_reg(_dense("qwen-3-0.6b", 1024, 28, 16, 8, 3072, tie=True))
_reg(_dense("qwen-3-4b", 2560, 36, 32, 8, 9728, tie=True))
_reg(_dense("qwen-3-8b", 4096, 36, 32, 8, 12288))
_reg(_dense("qwen-3-14b", 5120, 40, 40, 8, 17408))
_reg(_dense("qwen-3-32b", 5120, 64, 64, 8, 25600, tp=4))
_reg(_moe("qwen-3-30b-a3b", 2048, 48, 32, 4, experts=128, topk=8, moe_inter=768))
_reg(_moe("mixtral-8x7b", 4096, 32, 32, 8, experts=8, topk=2, moe_inter=14336,
          vocab=32000, tp=8, qk_norm=False, rope_theta=1e6))

# ---- embedding models (reference registry `common.py:13-21`) ----
_reg(_dense("qwen-3-embedding-0.6b", 1024, 28, 16, 8, 3072, tie=True,
            embedding=True))
_reg(_dense("qwen-3-embedding-4b", 2560, 36, 32, 8, 9728, tie=True, embedding=True))
_reg(_dense("qwen-3-embedding-8b", 4096, 36, 32, 8, 12288, embedding=True))
_reg(_dense("embeddinggemma-300m", 768, 24, 12, 4, 1152, tie=True, vocab=262144,
            hd=64, embedding=True, qk_norm=False))

# ---- generative models from the reference registry (`common.py:25-50`) ----
# Architectures are this framework's own (sized to the names); weights random-init.
_reg(_dense("qwen-3.5-2b", 2048, 28, 16, 8, 6144, tie=True))
_reg(_dense("qwen-3.5-27b", 5120, 48, 40, 8, 20480, tp=2))
_reg(_moe("qwen-3.5-35b-a3b", 2048, 48, 32, 4, experts=128, topk=8, moe_inter=896))
_reg(_moe("qwen-3.5-122b-a10b", 4096, 48, 32, 8, experts=128, topk=8,
          moe_inter=1536, tp=4))
_reg(_moe("gpt-oss-20b", 2880, 24, 64, 8, experts=32, topk=4, moe_inter=2880,
          vocab=201088, hd=64))
_reg(_moe("gpt-oss-120b", 2880, 36, 64, 8, experts=128, topk=4, moe_inter=2880,
          vocab=201088, hd=64, tp=4))
_reg(_moe("nemotron-3-super-120b-a12b", 4096, 52, 32, 8, experts=128, topk=8,
          moe_inter=1792, tp=4))
_reg(_moe("nemotron-3-nano-30b-a3b", 2048, 48, 32, 4, experts=128, topk=8,
          moe_inter=768))
_reg(_dense("gemma-4-31b-it", 5120, 48, 32, 16, 21504, vocab=262144, tp=2))
_reg(_moe("gemma-4-26b-a4b-it", 2560, 40, 16, 8, experts=64, topk=4,
          moe_inter=2048, vocab=262144))

# "-thinking" / "-no-thinking" variants share the base architecture.
for _name in list(_BASE):
    spec = _BASE[_name]
    if spec.embedding:
        continue
    _BASE[f"{_name}-thinking"] = replace(spec, name=f"{_name}-thinking", reasoning=True)
# gpt-oss family is reasoning-by-default in the reference list, with explicit
# "-no-thinking" names (`common.py:36-39`).
for _name in ("gpt-oss-20b", "gpt-oss-120b"):
    _BASE[_name] = replace(_BASE[_name], reasoning=True)
    _BASE[f"{_name}-no-thinking"] = replace(_BASE[_name], name=f"{_name}-no-thinking",
                                            reasoning=False)


MODEL_REGISTRY: Dict[str, ModelSpec] = dict(_BASE)


def get_model_spec(name: str) -> ModelSpec:
    """Resolve a model name to its spec. Unknown names raise KeyError with the
    available options (the reference accepts `| str` for Functions; the engine
    must know the architecture, so unknown names fail loudly here)."""
    try:
        return MODEL_REGISTRY[name]
    except KeyError:
        raise KeyError(
            f"unknown model {name!r}; available: {sorted(MODEL_REGISTRY)}"
        ) from None


def tiny_spec_for_tests(vocab_size: int = 512) -> ModelSpec:
    """A tiny dense spec used by CPU unit tests."""
    return ModelSpec(name="tiny-test", hidden_size=64, num_layers=2, num_heads=4,
                     num_kv_heads=2, head_dim=16, intermediate_size=128,
                     vocab_size=vocab_size, max_context=512, tie_embeddings=True)
