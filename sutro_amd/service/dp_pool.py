"""Data-parallel engine pool: one engine worker PROCESS per GPU.

Batch jobs are embarrassingly parallel over rows; on an 8-GPU MI355X node the
job service shards each job's rows across N worker processes (one per GPU,
`cuda:i` each), collects per-row results over queues, and merges them
input-ordered into the JobRecord. No collectives on this path — row scatter
and result gather are host-side and tiny next to the compute.

Workers are plain `spawn` processes running a continuous-batching engine loop;
the same pool runs on CPU (dev/tests) with device="cpu".
"""

from __future__ import annotations

import os
import queue as queue_mod
import threading
import time
from typing import Any, Dict, List, Optional

from ..interfaces import JobStatus


def _worker_main(rank: int, model: str, device: str, engine_kwargs: dict,
                 in_q, out_q) -> None:
    try:
        from ..engine.request import SamplingParams
        from ..engine.tokenizer import get_tokenizer
        from .models_map import resolve_engine_config
        from ..engine.engine import LLMEngine

        if device == "auto":
            import torch

            device = f"cuda:{rank}" if torch.cuda.is_available() else "cpu"
        elif device.startswith("cuda"):
            device = f"cuda:{rank}"
        cfg = resolve_engine_config(model, device=device, **(engine_kwargs or {}))
        eng = LLMEngine(cfg)
        tok = get_tokenizer()
        spec = eng.spec
        req_meta: Dict[int, tuple] = {}   # req_id -> (job_id, row_idx, auto_reasoning)
        fsm_cache: Dict[str, Optional[int]] = {}
        cancelled: set = set()
        out_q.put(("ready", rank, None, None))

        def admit(msg):
            _, job_id, rows, opts = msg
            schema = opts.get("json_schema")
            auto_reasoning = False
            if schema is None and spec.reasoning and not spec.embedding:
                schema = {"type": "object", "properties": {
                    "reasoning_content": {"type": "string", "maxLength": 512},
                    "content": {"type": "string", "maxLength": 512}}}
                auto_reasoning = True
            key = f"{job_id}"
            if key not in fsm_cache:
                fsm_cache[key] = (eng.register_fsm(schema)
                                  if schema is not None and not spec.embedding
                                  else None)
            fsm_id = fsm_cache[key]
            default_max = 1024 if schema is not None else cfg.default_max_new_tokens
            for row_idx, text in rows:
                ids = tok.render_prompt(text, opts.get("system_prompt"))
                sp = SamplingParams.from_dict(opts.get("sampling_params"),
                                              default_max)
                if opts.get("random_seed_per_input"):
                    sp.seed = row_idx
                req = eng.add_request(ids, sp, fsm_id=fsm_id,
                                      priority=opts.get("priority", 0),
                                      arrival_idx=row_idx,
                                      truncate=opts.get("truncate_rows", True))
                req_meta[req.req_id] = (job_id, row_idx, auto_reasoning)

        while True:
            # drain control queue
            try:
                while True:
                    msg = in_q.get_nowait()
                    if msg[0] == "shutdown":
                        return
                    if msg[0] == "cancel":
                        cancelled.add(msg[1])
                        for req_id, (jid, _, _) in list(req_meta.items()):
                            if jid == msg[1]:
                                r = _find_req(eng, req_id)
                                if r is not None:
                                    eng.abort_request(r)
                                req_meta.pop(req_id, None)
                    elif msg[0] == "run_rows":
                        if msg[1] not in cancelled:
                            admit(msg)
            except queue_mod.Empty:
                pass
            if eng.has_work():
                stats = eng.step()
                for req in stats.finished:
                    meta = req_meta.pop(req.req_id, None)
                    if meta is None:
                        continue
                    job_id, row_idx, _auto = meta
                    import math

                    if spec.embedding:
                        emb = eng.embeddings.pop(req.req_id, None)
                        payload = {"emb": emb.tolist() if emb is not None else None,
                                   "output": None,
                                   "in_tokens": len(req.prompt_token_ids),
                                   "out_tokens": 0}
                    else:
                        text = tok.decode(req.output_token_ids)
                        n = max(1, len(req.output_token_ids))
                        payload = {
                            "output": text,
                            "cumulative_logprob": req.cumulative_logprob,
                            "confidence": float(min(1.0, max(0.0, math.exp(
                                req.cumulative_logprob / n)))),
                            "in_tokens": len(req.prompt_token_ids),
                            "out_tokens": len(req.output_token_ids),
                        }
                    out_q.put(("row_done", rank, job_id, (row_idx, payload)))
            else:
                try:
                    msg = in_q.get(timeout=0.05)
                except queue_mod.Empty:
                    continue
                if msg[0] == "shutdown":
                    return
                if msg[0] == "run_rows" and msg[1] not in cancelled:
                    admit(msg)
                elif msg[0] == "cancel":
                    cancelled.add(msg[1])
    except Exception as e:  # pragma: no cover
        out_q.put(("worker_error", rank, None, f"{type(e).__name__}: {e}"))


def _find_req(eng, req_id):
    sch = eng.scheduler
    for pool in (sch.running, sch.waiting_p0, sch.waiting_p1):
        for r in pool:
            if r.req_id == req_id:
                return r
    return None


class MultiProcEngineWorker:
    """EngineWorker-compatible facade over N engine processes (DP row shard)."""

    def __init__(self, model: str, n_workers: int, device: str = "auto",
                 engine_kwargs: Optional[dict] = None):
        import torch.multiprocessing as mp

        from ..models.registry import get_model_spec

        self.model = model
        self.spec = get_model_spec(model)
        self.n = n_workers
        ctx = mp.get_context("spawn")
        self.in_qs = [ctx.Queue() for _ in range(n_workers)]
        self.out_q = ctx.Queue()
        self.procs = [
            ctx.Process(target=_worker_main,
                        args=(r, model, device, engine_kwargs or {},
                              self.in_qs[r], self.out_q), daemon=True)
            for r in range(n_workers)
        ]
        for p in self.procs:
            p.start()
        ready = 0
        deadline = time.time() + 600
        while ready < n_workers and time.time() < deadline:
            kind, rank, _, info = self.out_q.get(timeout=600)
            if kind == "ready":
                ready += 1
            elif kind == "worker_error":
                raise RuntimeError(f"engine worker {rank} failed: {info}")
        self._jobs: Dict[str, tuple] = {}  # job_id -> (JobRecord, service, t0)
        self._collector = threading.Thread(target=self._collect, daemon=True)
        self._collector.start()

    def submit_job(self, job, service) -> None:
        job.status = JobStatus.RUNNING
        job.datetime_started = time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime())
        self._jobs[job.job_id] = (job, service, time.time())
        service.persist_job(job)
        opts = {
            "system_prompt": job.system_prompt,
            "json_schema": job.json_schema,
            "sampling_params": job.sampling_params,
            "random_seed_per_input": job.random_seed_per_input,
            "truncate_rows": job.truncate_rows,
            "priority": job.priority,
        }
        # contiguous row shards, one per worker
        rows = [(i, r if isinstance(r, str) else __import__("json").dumps(r))
                for i, r in enumerate(job.inputs)]
        per = (len(rows) + self.n - 1) // self.n
        for w in range(self.n):
            shard = rows[w * per:(w + 1) * per]
            if shard:
                self.in_qs[w].put(("run_rows", job.job_id, shard, opts))

    def cancel_job(self, job_id: str) -> None:
        for q in self.in_qs:
            q.put(("cancel", job_id))

    def shutdown(self) -> None:
        for q in self.in_qs:
            q.put(("shutdown",))
        for p in self.procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()

    def _collect(self) -> None:
        import math

        while True:
            try:
                kind, rank, job_id, data = self.out_q.get(timeout=0.5)
            except queue_mod.Empty:
                if not any(p.is_alive() for p in self.procs):
                    return
                continue
            if kind == "worker_error":
                for job, service, _ in self._jobs.values():
                    if not JobStatus.is_terminal(job.status):
                        job.status = JobStatus.FAILED
                        job.failure_reason = {"message": str(data)}
                        service.persist_job(job)
                return
            if kind != "row_done" or job_id not in self._jobs:
                continue
            job, service, t0 = self._jobs[job_id]
            row_idx, payload = data
            with job.lock:
                if self.spec.embedding:
                    job.embeddings[row_idx] = payload.get("emb")
                    job.outputs[row_idx] = None
                else:
                    text = payload["output"]
                    if self.spec.reasoning and job.json_schema is not None:
                        import json as _json

                        text = _json.dumps({"content": text,
                                            "reasoning_content": ""})
                    job.outputs[row_idx] = text
                    job.cumulative_logprobs[row_idx] = payload.get(
                        "cumulative_logprob")
                    job.confidence_scores[row_idx] = payload.get("confidence")
                job.input_tokens += payload.get("in_tokens", 0)
                job.output_tokens += payload.get("out_tokens", 0)
                job.completed_rows += 1
                el = time.time() - t0
                if el > 0:
                    job.tokens_per_second = (
                        (job.input_tokens + job.output_tokens) / el)
                if job.completed_rows == job.num_rows:
                    job.status = JobStatus.SUCCEEDED
                    job.datetime_completed = time.strftime(
                        "%Y-%m-%dT%H:%M:%S", time.gmtime())
                    job.job_cost = service.compute_cost(
                        self.spec, job.input_tokens, job.output_tokens)
                    service.persist_job(job, with_results=True)
