"""Flagship benchmark: Qwen3-32B `so.infer`-shaped batch inference throughput.

Measures whole-node output tokens/sec of the continuous-batching engine on
synthetic data with random-init weights (BASELINE.json metric). Data-parallel:
one rank per GPU over RCCL, each rank runs a full engine replica on its own
row shard (weak scaling: per-GPU work fixed).

    python bench.py --gpus 1 --steps 32 --warmup 8
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 bench.py --gpus 8 --steps 32 --warmup 8

One engine `step()` = one scheduler iteration (prefill chunks + one decode
token for every running sequence). The engine is kept saturated by refilling
finished rows from a synthetic job queue, so the timed region is the
steady state of a large batch job (prefill+decode mix included).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np
import torch


def build_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=48)
    p.add_argument("--warmup", type=int, default=12)
    p.add_argument("--model", type=str, default="qwen-3-32b")
    p.add_argument("--batch", type=int, default=2048,
                   help="rows in flight per GPU (max_num_seqs); 2048 rows of "
                        "ctx 256 use ~134 GB of KV — sized for 288 GB HBM3E, "
                        "and M=2048 decode GEMMs run 18-60%% more efficient "
                        "than M=1024 (profiles/PROFILES.md)")
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--max-new", type=int, default=128)
    p.add_argument("--tokens-per-step", type=int, default=32768)
    p.add_argument("--async-decode", action="store_true",
                   help="one-step-lagged decode (EngineConfig.async_decode); "
                        "A/B flag until GPU-validated")
    p.add_argument("--graph-prefill", action="store_true",
                   help="hipGraph-captured prefill (EngineConfig.graph_prefill)")
    p.add_argument("--min-prefill", type=int, default=None,
                   help="prefill accumulation threshold (tokens); default = "
                        "engine default")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--kv-blocks", type=int, default=None)
    p.add_argument("--no-refill", action="store_true",
                   help="measure pure decode of the initial batch (no new rows)")
    p.add_argument("--tune", action="store_true",
                   help="run TunableOp GEMM tuning and write the results file")
    p.add_argument("--tp", type=int, default=1,
                   help="tensor-parallel degree (ranks per engine replica; "
                        "dp = world_size // tp)")
    p.add_argument("--kv-fp8", action="store_true",
                   help="OCP e4m3 KV cache (halves KV bytes; labeled variant, "
                        "not the default bf16 measurement)")
    p.add_argument("--schema", action="store_true",
                   help="FSM-guided structured JSON extraction workload "
                        "(BASELINE 1M-row p1 config shape)")
    p.add_argument("--e2e", action="store_true",
                   help="service-level bench: real so.infer() jobs through "
                        "the JobService (tokenize, schedule, materialize "
                        "input-ordered results) — the BASELINE rows/hour "
                        "metric end to end. Under torchrun each rank runs "
                        "its own service on its GPU (DP row shard); "
                        "standalone with --gpus N the service's own dp_pool "
                        "spawns one worker per GPU.")
    p.add_argument("--e2e-rows", type=int, default=3000,
                   help="rows per GPU for --e2e")
    return p.parse_args()


def run_e2e(args, rank: int, world: int, local_rank: int, have_gpu: bool):
    """so.infer() end to end: submit -> schedule -> materialize, report
    whole-node rows/hour + output tokens/sec (BASELINE.json metric)."""
    import tempfile

    import torch.distributed as dist_mod

    from sutro_amd.sdk import Sutro

    from sutro_amd.models.registry import get_model_spec as _gms

    device = (f"cuda:{local_rank}" if have_gpu else "cpu")
    dist = dist_mod if world > 1 else None
    model = args.model
    if not have_gpu and not _gms(model).embedding:
        model = "qwen-3-0.6b"  # CPU smoke: small generative stand-in
    rows = args.e2e_rows if have_gpu else 24
    max_new = args.max_new if have_gpu else 8
    home = tempfile.mkdtemp(prefix=f"sutro-bench-{rank}-")
    ek = {"max_num_seqs": args.batch,
          "max_model_len": max(256, args.prompt_len + max_new + 32)}
    if not have_gpu:
        ek["num_kv_blocks"] = 512
    client = Sutro(home=home, device=device, engine_kwargs=ek)

    # synthetic rows ~ prompt_len tokens of text each (tokenized for real by
    # the service — that cost is part of the measurement)
    rng = np.random.default_rng(77 + rank)
    words = ["alpha", "binary", "cluster", "delta", "engine", "filter",
             "graph", "hidden", "input", "joint", "kernel", "linear"]
    texts = [" ".join(rng.choice(words) for _ in range(args.prompt_len // 2))
             for _ in range(rows)]
    sp = {"max_tokens": max_new, "temperature": 0.8, "top_p": 0.95}

    if _gms(model).embedding:
        # BASELINE config 3 shape: embedding job through so.embed()
        _ = client.embed(texts[:8], model=model)      # warmup
        if dist is not None:
            dist.barrier()
        t0 = time.time()
        df = client.embed(texts, model=model)
        t1 = time.time()
        assert len(df) == rows
        elapsed = t1 - t0
        if dist is not None:
            te = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(te, op=dist.ReduceOp.MAX)
            elapsed = float(te.item())
            tr = torch.tensor([float(rows)], dtype=torch.float64)
            dist.all_reduce(tr)
            rows = int(tr.item())
        n_gpus = world if world > 1 else args.gpus
        if rank == 0:
            print(json.dumps({
                "metric": "embedding_rows_per_sec",
                "value": round(rows / elapsed, 2), "unit": "rows/s",
                "n_gpus": n_gpus, "steps": args.steps, "warmup": args.warmup,
                "ms_per_step": None, "higher_is_better": True,
                "scaling": "weak", "vs_baseline": None,
                "dtype": "bf16" if have_gpu else "fp32", "data": "synthetic",
                "config": {"model": model, "mode": "e2e_service_so.embed",
                           "rows": rows, "parallelism": f"dp{n_gpus}",
                           "rows_per_hour": round(rows / elapsed * 3600, 1),
                           "elapsed_s": round(elapsed, 1)}}))
        client.shutdown()
        return

    schema = None
    if args.schema:
        schema = {"type": "object", "properties": {
            "name": {"type": "string", "maxLength": 48},
            "category": {"enum": ["news", "review", "spam", "other"]},
            "sentiment": {"enum": ["positive", "negative", "neutral"]},
            "score": {"type": "integer", "minimum": 0, "maximum": 100}}}

    # warmup job: engine init, hipGraph capture, first GEMM algo selection
    wid = client.infer(texts[:8], model=model, job_priority=1,
                       stay_attached=False, sampling_params=sp,
                       output_schema=schema)
    client.await_job_completion(wid, obtain_results=False, timeout=600)

    # p0 interactive latency: single row submit -> materialized result
    # (admission bypasses the p1 accumulation hold)
    t_p0 = time.time()
    pid = client.infer([texts[0]], model=model, job_priority=0,
                       stay_attached=False, output_schema=schema,
                       sampling_params={**sp, "max_tokens": 16})
    client.await_job_completion(pid, obtain_results=False, timeout=300)
    client.get_job_results(pid)
    p0_latency_ms = (time.time() - t_p0) * 1000.0

    if dist is not None:
        dist.barrier()
    t0 = time.time()
    job_id = client.infer(texts, model=model, job_priority=1,
                          stay_attached=False, sampling_params=sp,
                          output_schema=schema)
    client.await_job_completion(job_id, obtain_results=False, timeout=3600)
    df = client.get_job_results(job_id)
    t1 = time.time()
    assert len(df) == rows, f"{len(df)} != {rows}"
    job = client.do_request("GET", f"/jobs/{job_id}").json()["job"]
    out_tokens = float(job["output_tokens"])
    elapsed = t1 - t0
    if dist is not None:
        te = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
        tt = torch.tensor([out_tokens, float(rows)], dtype=torch.float64)
        dist.all_reduce(tt)
        out_tokens, rows = float(tt[0].item()), int(tt[1].item())
    n_gpus = world if world > 1 else args.gpus
    if rank == 0:
        result = {
            "metric": "output_tokens_per_sec",
            "value": round(out_tokens / elapsed, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": None,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if have_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": model, "mode": "e2e_service_so.infer",
                "rows": rows, "rows_per_hour": round(rows / elapsed * 3600, 1),
                "global_batch": args.batch * n_gpus,
                "prompt_len": args.prompt_len, "max_new_tokens": max_new,
                "parallelism": f"dp{n_gpus}",
                "elapsed_s": round(elapsed, 1),
                "p0_row_latency_ms": round(p0_latency_ms, 1),
                "guided": bool(schema),
            },
        }
        print(json.dumps(result))
    client.shutdown()


def main():
    args = build_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    have_gpu = torch.cuda.is_available()
    device = args.device or ("cuda" if have_gpu else "cpu")
    if have_gpu:
        torch.cuda.set_device(local_rank)

    if args.tune:
        os.environ["SUTRO_AMD_TUNABLEOP_TUNE"] = "1"
        os.environ.setdefault("SUTRO_AMD_TUNABLEOP_FILE",
                              "gpurun_out/tunableop_gfx950.csv")
        os.makedirs("gpurun_out", exist_ok=True)

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        backend = "nccl" if have_gpu else "gloo"
        dist.init_process_group(backend=backend)

    if args.e2e:
        run_e2e(args, rank, world, local_rank, have_gpu)
        if dist is not None:
            dist.destroy_process_group()
        return

    from sutro_amd.engine.config import EngineConfig
    from sutro_amd.engine.engine import LLMEngine
    from sutro_amd.engine.request import SamplingParams
    from sutro_amd.models.registry import get_model_spec, tiny_spec_for_tests

    if device == "cpu":
        # no GPU in this container: run the contract end-to-end on a tiny spec
        spec = tiny_spec_for_tests()
        args.batch = min(args.batch, 32)
        args.prompt_len = min(args.prompt_len, 32)
    else:
        spec = get_model_spec(args.model)

    tp = max(1, args.tp)
    dp_idx = rank // tp
    cfg = EngineConfig(
        spec=spec,
        device=device,
        max_model_len=max(256, args.prompt_len + args.max_new + 32),
        max_num_seqs=args.batch,
        max_tokens_per_step=args.tokens_per_step,
        async_decode=args.async_decode,
        graph_prefill=args.graph_prefill,
        **({"min_prefill_batch_tokens": args.min_prefill}
           if args.min_prefill is not None else {}),
        num_kv_blocks=args.kv_blocks if device != "cpu" else 512,
        seed=dp_idx,  # identical within a TP group (lockstep), unique per replica
        tp_size=tp,
        kv_dtype="fp8_e4m3" if args.kv_fp8 else "bf16",
    )
    t_init0 = time.time()
    eng = LLMEngine(cfg)
    init_s = time.time() - t_init0

    # synthetic row stream: prompts of prompt_len random BPE token ids over
    # the model's real vocab, seeded per DP replica (identical within a TP
    # group so all its ranks run lockstep)
    vocab_hi = eng.tokenizer.vocab_size
    rng = np.random.default_rng(1234 + dp_idx)

    def make_prompt():
        body = rng.integers(3, vocab_hi, size=args.prompt_len - 1).tolist()
        return [1] + body  # BOS + random tokens

    sp_kwargs = dict(max_tokens=args.max_new, temperature=0.8, top_p=0.95)
    fsm_id = None
    if args.schema:
        schema = {"type": "object", "properties": {
            "name": {"type": "string", "maxLength": 48},
            "category": {"enum": ["news", "review", "spam", "other"]},
            "sentiment": {"enum": ["positive", "negative", "neutral"]},
            "score": {"type": "integer", "minimum": 0, "maximum": 100},
            "tags": {"type": "array", "items": {"type": "string",
                                                "maxLength": 16},
                     "minItems": 1, "maxItems": 4}}}
        fsm_id = eng.register_fsm(schema)
        sp_kwargs = dict(max_tokens=max(args.max_new, 512), temperature=0.9)
    arrival = [0]

    def refill():
        if args.no_refill and arrival[0] >= args.batch:
            return
        # p1 = production batch job: admission is accumulation-batched so
        # decode steps stay pure hipGraph replays (p0 would bypass that)
        sch = eng.scheduler
        while (len(sch.running) + len(sch.waiting_p0) + len(sch.waiting_p1)
               < args.batch):
            eng.add_request(make_prompt(), SamplingParams(**sp_kwargs),
                            fsm_id=fsm_id, priority=1, arrival_idx=arrival[0])
            arrival[0] += 1

    ramp = not eng.spec.embedding and not args.no_refill
    if not ramp:
        refill()

    def sync():
        if have_gpu:
            torch.cuda.synchronize()
        if dist is not None:
            dist.barrier()
            if have_gpu:
                torch.cuda.synchronize()

    # prime to STEADY STATE. A thundering-herd start (all rows admitted at
    # once) makes the job periodic with period ~max_new steps: short driver
    # windows then catch a prefill-heavy or decode-only phase and report
    # either half or double the true steady rate (r1 driver run measured
    # 4.0k tok/s on a 20-step window vs 7.8k on 32 steps, same build).
    # Fix: ramp admission over one row lifetime so rows finish (and refill)
    # uniformly — every window then carries the same prefill/decode mix.
    # Generative only — embedding rows never leave prefill.
    if ramp:
        per_step = max(1, args.batch // max(1, args.max_new))
        admitted = 0
        for _ in range(4 * args.batch):
            if admitted >= args.batch and not any(
                    r.in_prefill for r in eng.scheduler.running):
                break
            for _ in range(min(per_step, args.batch - admitted)):
                eng.add_request(make_prompt(), SamplingParams(**sp_kwargs),
                                fsm_id=fsm_id, priority=1,
                                arrival_idx=arrival[0])
                arrival[0] += 1
                admitted += 1
            eng.step()
    elif not eng.spec.embedding:
        for _ in range(256):
            sch = eng.scheduler
            if not sch.running or all(not r.in_prefill for r in sch.running):
                break
            eng.step()

    # warmup: W untimed steps
    for _ in range(args.warmup):
        refill()
        eng.step()

    sync()
    t0 = time.time()
    out_tokens = 0
    sched_tokens = 0
    count_me = (rank % tp) == 0  # one counter per TP replica
    rows_done = 0
    # SUTRO_BENCH_TRACE=1: per-step wall times on stderr (adds a per-step
    # sync, so it is NOT on for headline runs)
    trace = os.environ.get("SUTRO_BENCH_TRACE") == "1"
    step_kinds = []
    for _ in range(args.steps):
        refill()
        ts = time.time()
        stats = eng.step()
        if trace:
            if have_gpu:
                torch.cuda.synchronize()
            step_kinds.append((stats.prefill_tokens, (time.time() - ts) * 1e3))
        if count_me:
            out_tokens += stats.output_tokens
            sched_tokens += stats.scheduled_tokens
            rows_done += len(stats.finished)
    if have_gpu:
        torch.cuda.synchronize()
    t1 = time.time()
    if trace and rank == 0:
        dec = [m for p_, m in step_kinds if p_ == 0]
        mix = [(p_, m) for p_, m in step_kinds if p_ > 0]
        print(f"[steps] pure-decode: {len(dec)} x {sum(dec)/max(1,len(dec)):.1f} ms; "
              f"mixed: {[(p_, round(m,1)) for p_, m in mix]}", file=sys.stderr)
    elapsed = t1 - t0
    sync()

    if dist is not None:
        te = torch.tensor([elapsed], dtype=torch.float64,
                          device=device if have_gpu else "cpu")
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
        tt = torch.tensor([out_tokens, sched_tokens], dtype=torch.float64,
                          device=device if have_gpu else "cpu")
        dist.all_reduce(tt, op=dist.ReduceOp.SUM)
        out_tokens, sched_tokens = float(tt[0].item()), float(tt[1].item())

    n_gpus = world if world > 1 else args.gpus
    if spec.embedding:
        # embedding jobs have no decode; throughput = rows (and prompt tokens)
        if dist is not None:
            tr = torch.tensor([float(rows_done)],
                              device=device if have_gpu else "cpu")
            dist.all_reduce(tr)
            rows_done = float(tr.item())
        value = rows_done / elapsed if elapsed > 0 else 0.0
        metric = "embedding_rows_per_sec"
    else:
        value = out_tokens / elapsed if elapsed > 0 else 0.0
        metric = "output_tokens_per_sec"
    if rank == 0:
        result = {
            "metric": metric,
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed * 1000.0 / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if device != "cpu" else "fp32",
            "kv_dtype": "fp8_e4m3" if args.kv_fp8 else
                        ("bf16" if device != "cpu" else "fp32"),
            "data": "synthetic",
            "config": {
                "model": spec.name,
                "global_batch": args.batch * max(1, n_gpus // tp),
                "seq_len": args.prompt_len + args.max_new,
                "parallelism": (f"dp{n_gpus}" if tp == 1
                                else f"dp{max(1, n_gpus // tp)}xtp{tp}"),
                "prompt_len": args.prompt_len,
                "max_new_tokens": args.max_new,
                "rows_per_hour": round(value / max(1, args.max_new) * 3600, 1),
                "scheduled_tokens_per_sec": round(sched_tokens / elapsed, 1),
                "engine_init_s": round(init_s, 1),
            },
        }
        print(json.dumps(result))
    if args.tune and have_gpu:
        import torch.cuda.tunable as tunable

        if hasattr(tunable, "write_file"):
            tunable.write_file()
        # newer torch flushes tuned results to the filename automatically
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
