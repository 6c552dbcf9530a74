"""Tensor-parallel correctness on CPU (gloo, world_size 2).

TP=2 must match TP=1: the parallel linears partition deterministic full
weights, so logits agree to fp32 rounding and greedy generation is identical.
"""

import multiprocessing as mp
import os

import torch


def _run_tp_worker(rank, world, port, result_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        from sutro_amd.engine.config import EngineConfig
        from sutro_amd.engine.engine import LLMEngine
        from sutro_amd.engine.request import SamplingParams
        from sutro_amd.models.registry import tiny_spec_for_tests

        cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                           max_model_len=256, num_kv_blocks=64,
                           max_tokens_per_step=128, tp_size=world, seed=0)
        eng = LLMEngine(cfg)
        reqs = [eng.add_request(eng.tokenizer.encode(t),
                                SamplingParams(max_tokens=8, temperature=0))
                for t in ("tensor parallel row a", "row b")]
        while eng.has_work():
            eng.step()
        outs = [list(r.output_token_ids) for r in reqs]
        result_q.put((rank, outs))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        result_q.put((rank, f"ERROR: {type(e).__name__}: {e}"))


def _tp1_reference():
    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import tiny_spec_for_tests

    cfg = EngineConfig(spec=tiny_spec_for_tests(), device="cpu",
                       max_model_len=256, num_kv_blocks=64,
                       max_tokens_per_step=128, seed=0)
    eng = LLMEngine(cfg)
    reqs = [eng.add_request(eng.tokenizer.encode(t),
                            SamplingParams(max_tokens=8, temperature=0))
            for t in ("tensor parallel row a", "row b")]
    while eng.has_work():
        eng.step()
    return [list(r.output_token_ids) for r in reqs]


def test_tp2_greedy_matches_tp1():
    ref = _tp1_reference()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29551
    procs = [ctx.Process(target=_run_tp_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, outs = q.get(timeout=300)
        assert not isinstance(outs, str), outs
        results[rank] = outs
    for p in procs:
        p.join(timeout=60)
    # both ranks lockstep-identical, and equal to the TP=1 reference
    assert results[0] == results[1]
    assert results[0] == ref


def test_tp_shard_init_consistency():
    """Sharded params concatenated across ranks == the unsharded param."""
    from sutro_amd.models.qwen3 import Qwen3Model
    from sutro_amd.models.registry import tiny_spec_for_tests
    from sutro_amd.parallel.tp import TPContext

    spec = tiny_spec_for_tests()
    full = Qwen3Model(spec, torch.float32, 128)
    full.init_random_weights(0)
    shards = []
    for r in range(2):
        m = Qwen3Model(spec, torch.float32, 128, TPContext(size=2, rank=r))
        m.init_random_weights(0)
        shards.append(m)
    # column-parallel: qkv rows per section
    w_full = full.layers[0].self_attn.qkv_proj.weight
    qs = spec.num_heads * spec.head_dim
    kvs = spec.num_kv_heads * spec.head_dim
    w0 = shards[0].layers[0].self_attn.qkv_proj.weight
    w1 = shards[1].layers[0].self_attn.qkv_proj.weight
    torch.testing.assert_close(w0[: qs // 2], w_full[: qs // 2])
    torch.testing.assert_close(w1[: qs // 2], w_full[qs // 2: qs])
    # row-parallel: o_proj columns
    o_full = full.layers[0].self_attn.o_proj.weight
    o0 = shards[0].layers[0].self_attn.o_proj.weight
    o1 = shards[1].layers[0].self_attn.o_proj.weight
    torch.testing.assert_close(torch.cat([o0, o1], dim=1), o_full)


def test_tp_row_parallel_linear_numerics():
    from sutro_amd.parallel.tp import RowParallelLinear, TPContext

    torch.manual_seed(0)
    x = torch.randn(4, 8)
    w = torch.randn(6, 8)
    ref = x @ w.t()
    parts = []
    for r in range(2):
        lin = RowParallelLinear(8, 6, TPContext(size=1), torch.float32)
        # emulate 2-way row parallelism without a process group
        lin.weight.data = w[:, r * 4:(r + 1) * 4].clone()
        lin.in_per_rank = 4
        parts.append(lin(x[:, r * 4:(r + 1) * 4]))
    torch.testing.assert_close(parts[0] + parts[1], ref)


def _run_ep_worker(rank, world, port, result_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        from sutro_amd.engine.config import EngineConfig
        from sutro_amd.engine.engine import LLMEngine
        from sutro_amd.engine.request import SamplingParams
        from sutro_amd.models.registry import ModelSpec

        spec = ModelSpec(name="tiny-moe-ep", hidden_size=64, num_layers=2,
                         num_heads=4, num_kv_heads=2, head_dim=16,
                         intermediate_size=0, vocab_size=512, max_context=512,
                         tie_embeddings=True, num_experts=4,
                         experts_per_token=2, moe_intermediate_size=64)
        cfg = EngineConfig(spec=spec, device="cpu", max_model_len=256,
                           num_kv_blocks=64, max_tokens_per_step=128,
                           tp_size=world, seed=0, moe_ep=True)
        eng = LLMEngine(cfg)
        req = eng.add_request(eng.tokenizer.encode("expert parallel row"),
                              SamplingParams(max_tokens=8, temperature=0))
        while eng.has_work():
            eng.step()
        result_q.put((rank, list(req.output_token_ids)))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        result_q.put((rank, f"ERROR: {type(e).__name__}: {e}"))


def test_moe_expert_parallel_matches_single_rank():
    """EP=2 (experts sharded + all-reduce combine) greedy == single-rank MoE."""
    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import ModelSpec

    spec = ModelSpec(name="tiny-moe-ep", hidden_size=64, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=16,
                     intermediate_size=0, vocab_size=512, max_context=512,
                     tie_embeddings=True, num_experts=4, experts_per_token=2,
                     moe_intermediate_size=64)
    cfg = EngineConfig(spec=spec, device="cpu", max_model_len=256,
                       num_kv_blocks=64, max_tokens_per_step=128, seed=0)
    eng = LLMEngine(cfg)
    req = eng.add_request(eng.tokenizer.encode("expert parallel row"),
                          SamplingParams(max_tokens=8, temperature=0))
    while eng.has_work():
        eng.step()
    ref = list(req.output_token_ids)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_run_ep_worker, args=(r, 2, 29553, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, outs = q.get(timeout=300)
        assert not isinstance(outs, str), outs
        results[rank] = outs
    for p in procs:
        p.join(timeout=60)
    assert results[0] == results[1] == ref


def test_bench_tp2_torchrun_cpu():
    """The driver-contract bench entrypoint with --tp 2 runs end to end over
    gloo on CPU (2 ranks, one TP replica) and prints the JSON line."""
    import json as _json
    import subprocess
    import sys

    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", "29753", "bench.py", "--gpus", "2", "--tp", "2",
           "--steps", "2", "--warmup", "1"]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = _json.loads(line)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp1xtp2"
    assert d["value"] > 0


def _run_pipelined_worker(rank, world, port, result_q):
    import os

    import torch.distributed as dist

    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sutro_amd.parallel.tp as tpmod
        from sutro_amd.parallel.tp import RowParallelLinear, TPContext

        tpmod.PIPELINE_MIN_TOKENS = 64  # force the pipelined path
        torch.manual_seed(0)
        T, K, N = 200, 16, 12  # T not divisible by chunks: exercises bounds
        x = torch.randn(T, K)
        w = torch.randn(N, K)
        tp = TPContext(size=world, rank=rank, group=None)
        lin = RowParallelLinear(K, N, TPContext(size=world), torch.float32)
        lin.weight.data = w[:, rank * (K // world):(rank + 1) * (K // world)].clone()
        lin.tp = tp
        y = lin(x[:, rank * (K // world):(rank + 1) * (K // world)])
        if rank == 0:
            result_q.put(("ok", y, x @ w.t()))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        result_q.put(("err", repr(e), None))


def test_tp_pipelined_all_reduce_matches_dense():
    """Chunk-pipelined row-parallel path (async all-reduce overlap) is
    numerically identical to the dense product (gloo, world 2)."""
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_run_pipelined_worker, args=(r, 2, 29513, q))
          for r in range(2)]
    for p in ps:
        p.start()
    kind, y, ref = q.get(timeout=120)
    for p in ps:
        p.join(timeout=60)
    assert kind == "ok", y
    torch.testing.assert_close(y, ref, atol=1e-5, rtol=1e-5)


def _run_a2a_worker(rank, world, port, result_q):
    import os

    import torch.distributed as dist

    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from sutro_amd.models.qwen3 import Qwen3MoE
        from sutro_amd.models.registry import ModelSpec
        from sutro_amd.parallel.tp import TPContext

        spec = ModelSpec(name="tiny-moe-a2a", hidden_size=32, num_layers=1,
                         num_heads=2, num_kv_heads=1, head_dim=16,
                         intermediate_size=0, vocab_size=128, num_experts=4,
                         experts_per_token=2, moe_intermediate_size=32)
        tp = TPContext(size=world, rank=rank, group=None)
        torch.manual_seed(0)  # identical router/experts on every rank...
        full = Qwen3MoE(spec, torch.float32)  # ...single-rank reference
        for p in full.parameters():
            torch.nn.init.normal_(p, std=0.1)
        shard = Qwen3MoE(spec, torch.float32, tp, ep=True)
        with torch.no_grad():
            shard.router.weight.copy_(full.router.weight)
            e_l = spec.num_experts // world
            shard.gate_up.copy_(full.gate_up[rank * e_l:(rank + 1) * e_l])
            shard.down.copy_(full.down[rank * e_l:(rank + 1) * e_l])

        torch.manual_seed(7)
        x_all = torch.randn(26, 32)  # uneven shards: 13 tokens each
        x_local = x_all[rank * 13:(rank + 1) * 13]
        with torch.no_grad():
            y_local = shard.forward_a2a(x_local)
            y_ref = full._forward_loop(x_all)[rank * 13:(rank + 1) * 13]
        result_q.put(("ok", rank, y_local.detach(), y_ref.detach()))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        result_q.put(("err", rank, repr(e), None))


def test_moe_all_to_all_dispatch_matches_single_rank():
    """EP all-to-all: token-sharded ranks exchange assignments to expert
    owners and back; every rank's output equals the single-module result on
    the concatenated tokens (gloo, world 2)."""
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_run_a2a_worker, args=(r, 2, 29517, q))
          for r in range(2)]
    for p in ps:
        p.start()
    got = [q.get(timeout=180) for _ in range(2)]
    for p in ps:
        p.join(timeout=60)
    for kind, rank, y, ref in got:
        assert kind == "ok", y
        torch.testing.assert_close(y, ref, atol=1e-5, rtol=1e-5)


def test_moe_grouped_ep_sharded_matches_loop():
    """EP-sharded grouped path (expert_base offset + invalid-assignment
    redirect slot) computes exactly the local contribution: summing both
    ranks' grouped outputs equals the full model (no process group needed —
    shards are emulated)."""
    from sutro_amd.models.qwen3 import Qwen3MoE
    from sutro_amd.models.registry import ModelSpec
    from sutro_amd.parallel.tp import TPContext

    spec = ModelSpec(name="tiny-moe-epg", hidden_size=32, num_layers=1,
                     num_heads=2, num_kv_heads=1, head_dim=16,
                     intermediate_size=0, vocab_size=128, num_experts=4,
                     experts_per_token=2, moe_intermediate_size=64)
    torch.manual_seed(3)
    full = Qwen3MoE(spec, torch.float32)
    for p in full.parameters():
        torch.nn.init.normal_(p, std=0.1)
    x = torch.randn(23, 32)
    ref = full._forward_loop(x)
    total = torch.zeros_like(ref)
    for r in range(2):
        shard = Qwen3MoE(spec, torch.float32,
                         TPContext(size=2, rank=r), ep=True)
        with torch.no_grad():
            shard.router.weight.copy_(full.router.weight)
            shard.gate_up.copy_(full.gate_up[r * 2:(r + 1) * 2])
            shard.down.copy_(full.down[r * 2:(r + 1) * 2])
        shard.tp = TPContext(size=1)  # no collective: sum manually
        shard.experts_per_rank = 2
        shard.expert_base = r * 2
        total += shard._forward_grouped(x)
    torch.testing.assert_close(total, ref, atol=1e-4, rtol=1e-4)
