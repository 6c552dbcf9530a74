"""Checkpoint round trip: save -> load (full and TP-sharded)."""

import torch

from sutro_amd.models.loader import load_weights, save_weights
from sutro_amd.models.qwen3 import Qwen3Model
from sutro_amd.models.registry import tiny_spec_for_tests
from sutro_amd.parallel.tp import TPContext


def _mk(seed=3, tp=None):
    m = Qwen3Model(tiny_spec_for_tests(), torch.float32, 128, tp)
    m.init_random_weights(seed)
    return m


def test_save_load_roundtrip(tmp_path):
    src = _mk(seed=5)
    save_weights(src, str(tmp_path))
    dst = _mk(seed=0)  # different init
    n = load_weights(dst, str(tmp_path))
    assert n > 0
    for (na, pa), (nb, pb) in zip(sorted(src.named_parameters()),
                                  sorted(dst.named_parameters())):
        assert na == nb
        torch.testing.assert_close(pa, pb)


def test_load_tp_shards_match_full(tmp_path):
    full = _mk(seed=7)
    save_weights(full, str(tmp_path))
    for r in range(2):
        shard = Qwen3Model(tiny_spec_for_tests(), torch.float32, 128,
                           TPContext(size=2, rank=r))
        load_weights(shard, str(tmp_path))
        # column-parallel qkv: rank r holds its q/k/v sections
        spec = tiny_spec_for_tests()
        qs = spec.num_heads * spec.head_dim
        w_full = full.layers[0].self_attn.qkv_proj.weight
        w_shard = shard.layers[0].self_attn.qkv_proj.weight
        expect_q = w_full[r * qs // 2:(r + 1) * qs // 2]
        torch.testing.assert_close(w_shard[: qs // 2], expect_q)


def test_load_split_projections(tmp_path):
    """HF-style checkpoints ship split q/k/v and gate/up tensors."""
    from safetensors.torch import save_file

    src = _mk(seed=9)
    spec = tiny_spec_for_tests()
    qs = spec.num_heads * spec.head_dim
    kvs = spec.num_kv_heads * spec.head_dim
    state = {}
    for name, p in src.named_parameters():
        t = p.detach().clone()
        if name.endswith("qkv_proj.weight"):
            prefix = "model." + name[: -len("qkv_proj.weight")]
            state[prefix + "q_proj.weight"] = t[:qs]
            state[prefix + "k_proj.weight"] = t[qs:qs + kvs]
            state[prefix + "v_proj.weight"] = t[qs + kvs:]
        elif name.endswith("gate_up_proj.weight"):
            inter = t.shape[0] // 2
            prefix = "model." + name[: -len("gate_up_proj.weight")]
            state[prefix + "gate_proj.weight"] = t[:inter]
            state[prefix + "up_proj.weight"] = t[inter:]
        else:
            state["model." + name] = t
    save_file(state, str(tmp_path / "model.safetensors"))
    dst = _mk(seed=0)
    load_weights(dst, str(tmp_path))
    for (na, pa), (nb, pb) in zip(sorted(src.named_parameters()),
                                  sorted(dst.named_parameters())):
        torch.testing.assert_close(pa, pb, msg=na)


def _mk_moe(seed=3, tp=None, ep=False):
    from sutro_amd.models.registry import ModelSpec

    spec = ModelSpec(name="tiny-moe-load", hidden_size=64, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=16,
                     intermediate_size=0, vocab_size=512, max_context=512,
                     tie_embeddings=True, num_experts=4, experts_per_token=2,
                     moe_intermediate_size=64)
    m = Qwen3Model(spec, torch.float32, 128, tp, moe_ep=ep)
    m.init_random_weights(seed)
    return m


def test_load_per_expert_hf_names(tmp_path):
    """HF MoE checkpoints ship mlp.experts.N.{gate,up,down}_proj tensors;
    the [E, N, K] parameter layout stacks them directly."""
    from safetensors.torch import save_file

    src = _mk_moe(seed=11)
    state = {}
    for name, p in src.named_parameters():
        t = p.detach().clone()
        if name.endswith(".mlp.gate_up"):
            prefix = "model." + name[: -len("gate_up")]
            m = t.shape[1] // 2
            for e in range(t.shape[0]):
                state[f"{prefix}experts.{e}.gate_proj.weight"] = t[e, :m]
                state[f"{prefix}experts.{e}.up_proj.weight"] = t[e, m:]
        elif name.endswith(".mlp.down"):
            prefix = "model." + name[: -len("down")]
            for e in range(t.shape[0]):
                state[f"{prefix}experts.{e}.down_proj.weight"] = t[e]
        else:
            state["model." + name] = t
    save_file(state, str(tmp_path / "model.safetensors"))
    dst = _mk_moe(seed=0)
    load_weights(dst, str(tmp_path))
    for (na, pa), (nb, pb) in zip(sorted(src.named_parameters()),
                                  sorted(dst.named_parameters())):
        assert na == nb
        torch.testing.assert_close(pa, pb)


def test_load_strict_rejects_unknown_tensors(tmp_path):
    from safetensors.torch import save_file

    import pytest

    src = _mk(seed=2)
    state = {"model." + n: p.detach().clone()
             for n, p in src.named_parameters()}
    state["model.layers.0.totally_unknown.weight"] = torch.zeros(3, 3)
    save_file(state, str(tmp_path / "model.safetensors"))
    dst = _mk(seed=0)
    with pytest.raises(ValueError, match="map to no parameter"):
        load_weights(dst, str(tmp_path))
    load_weights(dst, str(tmp_path), strict=False)  # non-strict: ignored
