"""Job service: the engine-side implementation of the reference's API contract.

Implements every behavior the reference client expects from `api.sutro.sh`
(endpoint table reconstructed in SURVEY.md §2.3 from `/root/reference/sutro/sdk.py`):
job submission with p0/p1 priorities, the QUEUED→…→SUCCEEDED status machine,
line-JSON progress streaming (`update_type: progress|tokens`), input-ordered
columnar results with cumulative_logprobs / confidence_score, cost estimation
(`cost_estimate` dry-run jobs), per-priority quotas, cancellation, and job
persistence for detach/reattach.

Execution: one EngineWorker thread per model engine, continuous-batching many
jobs' rows through a shared LLMEngine.
"""

from __future__ import annotations

import json
import math
import os
import threading
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..engine.engine import LLMEngine
from ..engine.request import Request, SamplingParams
from ..engine.tokenizer import get_tokenizer
from ..interfaces import JobStatus
from ..models.registry import get_model_spec
from .models_map import resolve_engine_config

SUTRO_HOME = os.environ.get("SUTRO_AMD_HOME", os.path.expanduser("~/.sutro-amd"))

NAME_MAX = 45
DESC_MAX = 512

# deterministic $/1M-token pricing derived from active parameter count
_IN_RATE_PER_B = 0.02   # $ per 1M input tokens per 1B active params
_OUT_RATE_PER_B = 0.08

DEFAULT_QUOTAS = [
    {"row_quota": 100_000, "token_quota": 50_000_000},      # p0 prototyping
    {"row_quota": 50_000_000, "token_quota": 10_000_000_000},  # p1 production
]


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime())


@dataclass
class JobRecord:
    job_id: str
    model: str
    status: JobStatus = JobStatus.QUEUED
    priority: int = 0
    name: Optional[str] = None
    description: Optional[str] = None
    inputs: List[Any] = field(default_factory=list)
    system_prompt: Optional[str] = None
    json_schema: Optional[dict] = None
    sampling_params: Optional[dict] = None
    random_seed_per_input: bool = False
    truncate_rows: bool = True
    cost_estimate_only: bool = False
    id_column_name: Optional[str] = None
    id_column_values: Optional[List[Any]] = None

    num_rows: int = 0
    completed_rows: int = 0
    input_tokens: int = 0
    output_tokens: int = 0
    tokens_per_second: float = 0.0
    cost_estimate: Optional[float] = None
    job_cost: float = 0.0
    failure_reason: Optional[Dict[str, str]] = None

    datetime_created: str = field(default_factory=_now)
    datetime_started: Optional[str] = None
    datetime_completed: Optional[str] = None

    # columnar result arrays, input-ordered
    outputs: List[Optional[str]] = field(default_factory=list)
    cumulative_logprobs: List[Optional[float]] = field(default_factory=list)
    confidence_scores: List[Optional[float]] = field(default_factory=list)
    embeddings: List[Optional[list]] = field(default_factory=list)

    lock: threading.Lock = field(default_factory=threading.Lock, repr=False)

    def to_public(self) -> Dict[str, Any]:
        """The `job` object shape the client parses (`sdk.py:1053-1080`)."""
        return {
            "job_id": self.job_id,
            "status": self.status.value,
            "model": self.model,
            "name": self.name,
            "description": self.description,
            "num_rows": self.num_rows,
            "input_tokens": self.input_tokens,
            "output_tokens": self.output_tokens,
            "cost_estimate": self.cost_estimate,
            "job_cost": round(self.job_cost, 5),
            "failure_reason": self.failure_reason,
            "datetime_created": self.datetime_created,
            "datetime_added": self.datetime_created,
            "datetime_started": self.datetime_started,
            "datetime_completed": self.datetime_completed,
            "job_priority": self.priority,
        }


class EngineWorker:
    """Owns one LLMEngine and pumps jobs' rows through it on a thread."""

    def __init__(self, model: str, device: str = "auto",
                 engine_kwargs: Optional[dict] = None):
        self.model = model
        cfg = resolve_engine_config(model, device=device, **(engine_kwargs or {}))
        self.engine = LLMEngine(cfg)
        self.spec = self.engine.spec
        self.tokenizer = self.engine.tokenizer  # model-vocab-sized BPE
        self._inbox: List[tuple] = []
        self._req_meta: Dict[int, tuple] = {}  # req_id -> (job, row_idx)
        # compiled guided-decoding FSMs, shared across jobs with the same
        # schema (DFA compilation for deep schemas can take seconds)
        self._fsm_by_schema: Dict[str, int] = {}
        self._cancelled_jobs: set = set()
        self.dead: Optional[str] = None  # failure message once the loop dies
        self._lock = threading.Lock()
        self._wake = threading.Event()
        self._stop = False
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name=f"engine-{model}")
        self._thread.start()

    def submit_job(self, job: JobRecord, service: "JobService") -> None:
        if self.dead is not None:
            with job.lock:
                job.status = JobStatus.FAILED
                job.failure_reason = {"message": self.dead}
                job.datetime_completed = _now()
            service.persist_job(job)
            return
        with self._lock:
            self._inbox.append((job, service, None))
        self._wake.set()

    def submit_partial(self, job: JobRecord, service: "JobService",
                       rows_idx) -> None:
        """Admit only the given row indices (engine-restart resume)."""
        if self.dead is not None:
            with job.lock:
                job.status = JobStatus.FAILED
                job.failure_reason = {"message": self.dead}
                job.datetime_completed = _now()
            service.persist_job(job)
            return
        with self._lock:
            self._inbox.append((job, service, list(rows_idx)))
        self._wake.set()

    def cancel_job(self, job_id: str) -> None:
        with self._lock:
            self._cancelled_jobs.add(job_id)
        self._wake.set()

    def shutdown(self) -> None:
        self._stop = True
        self._wake.set()
        self._thread.join(timeout=5)

    # ---- worker loop ----

    def _admit(self, job: JobRecord, service: "JobService",
               rows_idx=None) -> None:
        if JobStatus.is_terminal(job.status):
            return  # cancelled while queued — never resurrect to RUNNING
        eng = self.engine
        job.datetime_started = job.datetime_started or _now()
        job.status = JobStatus.RUNNING
        service.persist_job(job)
        fsm_id = None
        schema = job.json_schema
        if self.spec.reasoning and not self.spec.embedding:
            # real two-field generation: FSM forces reasoning_content +
            # content (user schema nests under content)
            from ..engine.guided import reasoning_wrapper_schema

            schema = reasoning_wrapper_schema(schema)
            job._auto_reasoning_schema = True  # type: ignore[attr-defined]
        if schema is not None and not self.spec.embedding:
            key = json.dumps(schema, sort_keys=True)
            fsm_id = self._fsm_by_schema.get(key)
            if fsm_id is None:
                fsm_id = eng.register_fsm(schema)
                self._fsm_by_schema[key] = fsm_id
        default_max = 1024 if schema is not None else eng.cfg.default_max_new_tokens
        t_start = time.time()
        indices = list(rows_idx if rows_idx is not None
                       else range(len(job.inputs)))
        texts = [job.inputs[i] if isinstance(job.inputs[i], str)
                 else json.dumps(job.inputs[i]) for i in indices]
        all_ids = self.tokenizer.render_prompts(texts, job.system_prompt)
        for i, ids in zip(indices, all_ids):
            sp = SamplingParams.from_dict(job.sampling_params, default_max)
            if job.random_seed_per_input:
                sp.seed = i
            req = eng.add_request(ids, sp, fsm_id=fsm_id, priority=job.priority,
                                  arrival_idx=i, truncate=job.truncate_rows)
            if rows_idx is None:
                # resume re-admissions were already counted at original
                # admission (persisted input_tokens) — don't double-count
                job.input_tokens += len(req.prompt_token_ids)
            self._req_meta[req.req_id] = (job, i, service)
        job._t_start = t_start  # type: ignore[attr-defined]
        service.persist_job(job)

    def _finish_row(self, req: Request) -> None:
        meta = self._req_meta.pop(req.req_id, None)
        if meta is None:
            return
        job, row_idx, service = meta
        with job.lock:
            if self.spec.embedding:
                emb = self.engine.embeddings.pop(req.req_id, None)
                job.embeddings[row_idx] = emb.tolist() if emb is not None else None
                job.outputs[row_idx] = None
            else:
                # reasoning models emit the {reasoning_content, content}
                # wrapper JSON directly via the forced FSM (_admit)
                text = self.engine.output_text(req)
                job.outputs[row_idx] = text
                n = max(1, len(req.output_token_ids))
                job.cumulative_logprobs[row_idx] = req.cumulative_logprob
                job.confidence_scores[row_idx] = float(
                    min(1.0, max(0.0, math.exp(req.cumulative_logprob / n)))
                )
            job.output_tokens += len(req.output_token_ids)
            job.completed_rows += 1
            el = time.time() - getattr(job, "_t_start", time.time())
            if el > 0:
                job.tokens_per_second = (job.input_tokens + job.output_tokens) / el
            if job.completed_rows % 256 == 0:
                # resumable partial results: a detached client (or a restarted
                # service) can still read completed shards
                service.persist_job(job, with_results=True)
            if (job.completed_rows == job.num_rows
                    and not JobStatus.is_terminal(job.status)):
                job.status = JobStatus.SUCCEEDED
                job.datetime_completed = _now()
                job.job_cost = service.compute_cost(self.spec, job.input_tokens,
                                                    job.output_tokens)
                service.persist_job(job, with_results=True)

    def _run(self) -> None:
        while not self._stop:
            with self._lock:
                inbox, self._inbox = self._inbox, []
                cancelled, self._cancelled_jobs = self._cancelled_jobs, set()
            for job, service, rows_idx in inbox:
                if job.job_id in cancelled:
                    continue  # cancelled before admission
                try:
                    self._admit(job, service, rows_idx)
                except Exception as e:  # admission failure -> FAILED
                    job.status = JobStatus.FAILED
                    job.failure_reason = {"message": f"{type(e).__name__}: {e}"}
                    job.datetime_completed = _now()
                    service.persist_job(job)
            if cancelled:
                for req_id, (job, row_idx, service) in list(self._req_meta.items()):
                    if job.job_id in cancelled:
                        req = self._find_req(req_id)
                        if req is not None:
                            self.engine.abort_request(req)
                        self._req_meta.pop(req_id, None)
            if self.engine.has_work():
                try:
                    stats = self.engine.step()
                except Exception as e:
                    # a fatal step error (e.g. HIP OOM) must not strand
                    # in-flight jobs as RUNNING forever: fail them all
                    msg = f"engine step failed: {type(e).__name__}: {e}"
                    self.dead = msg
                    failed = {}
                    for req_id, (job, _row, service) in list(self._req_meta.items()):
                        failed[job.job_id] = (job, service)
                        self._req_meta.pop(req_id, None)
                    for job, service in failed.values():
                        with job.lock:
                            if not JobStatus.is_terminal(job.status):
                                job.status = JobStatus.FAILED
                                job.failure_reason = {"message": msg}
                                job.datetime_completed = _now()
                                service.persist_job(job, with_results=True)
                    raise
                for req in stats.finished:
                    self._finish_row(req)
            else:
                self._wake.wait(timeout=0.05)
                self._wake.clear()

    def _find_req(self, req_id: int) -> Optional[Request]:
        sch = self.engine.scheduler
        for pool in (sch.running, sch.waiting_p0, sch.waiting_p1):
            for r in pool:
                if r.req_id == req_id:
                    return r
        return None


class JobService:
    """In-process service implementing the full job/dataset/quota contract."""

    def __init__(self, home: Optional[str] = None, device: str = "auto",
                 engine_kwargs: Optional[dict] = None):
        self.home = home or SUTRO_HOME
        os.makedirs(os.path.join(self.home, "jobs"), exist_ok=True)
        os.makedirs(os.path.join(self.home, "job-results"), exist_ok=True)
        self.device = device
        self.engine_kwargs = engine_kwargs or {}
        self.jobs: Dict[str, JobRecord] = {}
        self.workers: Dict[str, EngineWorker] = {}
        self.quotas = [dict(q) for q in DEFAULT_QUOTAS]
        self._lock = threading.Lock()
        # event-driven completion waits (p0 interactive latency: clients
        # block here instead of sleep-polling status)
        self._done_cv = threading.Condition()
        self._load_persisted_jobs()

    # ---- cost model ----

    @staticmethod
    def compute_cost(spec, input_tokens: int, output_tokens: int) -> float:
        b = spec.active_param_count() / 1e9
        return (input_tokens * _IN_RATE_PER_B * b
                + output_tokens * _OUT_RATE_PER_B * b) / 1e6

    def _estimate_cost(self, job: JobRecord) -> float:
        spec = get_model_spec(job.model)
        tok = get_tokenizer(spec.vocab_size)
        # large p1 estimates sample a ~1M-token prefix (README.md:173 behavior)
        budget = 1_000_000
        in_tokens = 0
        rows_counted = 0
        for row in job.inputs:
            text = row if isinstance(row, str) else json.dumps(row)
            in_tokens += len(tok.encode(text)) + 16
            rows_counted += 1
            if in_tokens >= budget:
                break
        if rows_counted < len(job.inputs):
            in_tokens = int(in_tokens * len(job.inputs) / max(1, rows_counted))
        sp = SamplingParams.from_dict(job.sampling_params, 256)
        est_out = len(job.inputs) * max(16, sp.max_tokens // 2)
        return round(self.compute_cost(spec, in_tokens, est_out), 5)

    # ---- submission ----

    def submit_job(self, payload: Dict[str, Any]) -> Dict[str, Any]:
        name = payload.get("name")
        if name and len(name) > NAME_MAX:
            raise ValueError(f"name exceeds {NAME_MAX} characters")
        desc = payload.get("description")
        if desc and len(desc) > DESC_MAX:
            raise ValueError(f"description exceeds {DESC_MAX} characters")
        model = payload["model"]
        try:
            get_model_spec(model)  # validate early
        except KeyError:
            # `model` may be a Function name (`sdk.py:710-721` passes the
            # function name as the model field for batch Functions runs)
            from .functions import FunctionStore

            fn = FunctionStore(self.home).get(model)
            if fn is None:
                raise
            model = fn["model"]
            payload = dict(payload)
            payload.setdefault("system_prompt", None)
            payload["system_prompt"] = payload["system_prompt"] or fn.get("system_prompt")
            payload["json_schema"] = payload.get("json_schema") or fn.get("output_schema")
        inputs = payload["inputs"]
        if isinstance(inputs, str):
            inputs, id_values = self._materialize_inputs(
                inputs, payload.get("column_name"), payload.get("id_column_name"))
        else:
            id_values = None
        priority = int(payload.get("job_priority", 0))
        q = self.quotas[min(priority, len(self.quotas) - 1)]
        if len(inputs) > q["row_quota"]:
            raise ValueError(
                f"job exceeds row quota for priority {priority}: "
                f"{len(inputs)} > {q['row_quota']}")
        # token quota: fast estimate at ~3 utf-8 bytes per BPE token (the
        # shipped tokenizer measures 2.7-3 on English/JSON; exact counts
        # would tokenize the whole job twice)
        est_tokens = sum(
            len(r.encode() if isinstance(r, str) else json.dumps(r).encode())
            for r in inputs) // 3 + len(inputs)
        if est_tokens > q["token_quota"]:
            raise ValueError(
                f"job exceeds token quota for priority {priority}: "
                f"~{est_tokens} > {q['token_quota']}")
        job = JobRecord(
            job_id=f"job-{uuid.uuid4().hex[:12]}",
            model=model,
            priority=priority,
            name=name,
            description=desc,
            inputs=list(inputs),
            system_prompt=payload.get("system_prompt"),
            json_schema=payload.get("json_schema"),
            sampling_params=payload.get("sampling_params"),
            random_seed_per_input=bool(payload.get("random_seed_per_input", False)),
            truncate_rows=bool(payload.get("truncate_rows", True)),
            cost_estimate_only=bool(payload.get("cost_estimate", False)),
            id_column_name=payload.get("id_column_name"),
            id_column_values=id_values,
        )
        job.num_rows = len(job.inputs)
        job.outputs = [None] * job.num_rows
        job.cumulative_logprobs = [None] * job.num_rows
        job.confidence_scores = [None] * job.num_rows
        job.embeddings = [None] * job.num_rows
        with self._lock:
            self.jobs[job.job_id] = job
        if job.cost_estimate_only:
            job.status = JobStatus.RUNNING
            job.cost_estimate = self._estimate_cost(job)
            job.status = JobStatus.SUCCEEDED
            job.datetime_completed = _now()
            self.persist_job(job)
        else:
            worker = self._get_worker(model)
            worker.submit_job(job, self)
        return {"results": job.job_id}

    def _materialize_inputs(self, ref: str, column: Optional[str],
                            id_column: Optional[str]):
        """Dataset-ID or URL inputs -> row list (+ optional id column values)."""
        import pandas as pd

        if ref.startswith("dataset-"):
            from .datasets import DatasetStore

            store = DatasetStore(self.home)
            df = store.read_all(ref)
        elif ref.startswith(("http://", "https://")):
            if ref.endswith(".parquet"):
                df = pd.read_parquet(ref)
            else:
                df = pd.read_csv(ref)
        else:
            raise ValueError(f"unsupported inputs reference: {ref!r}")
        if column is None:
            raise ValueError("column_name required for dataset/URL inputs")
        if isinstance(column, list):
            from ..common import do_dataframe_column_concatenation

            rows = do_dataframe_column_concatenation(df, column)
        else:
            rows = df[column].astype(str).tolist()
        id_values = df[id_column].tolist() if id_column else None
        return rows, id_values

    def _num_dp_workers(self) -> int:
        env = os.environ.get("SUTRO_AMD_NUM_WORKERS")
        if env is not None:
            return max(1, int(env))
        try:
            import torch

            n = torch.cuda.device_count()
        except Exception:
            n = 0
        return n if n > 1 else 1

    def _get_worker(self, model: str):
        with self._lock:
            w = self.workers.get(model)
            if w is None:
                n = self._num_dp_workers()
                if n > 1:
                    from .dp_pool import MultiProcEngineWorker

                    spec = get_model_spec(model)
                    tp_env = os.environ.get("SUTRO_AMD_TP")
                    tp = int(tp_env) if tp_env else min(spec.recommended_tp, n)
                    tp = max(1, tp)
                    while n % tp:
                        tp -= 1
                    w = MultiProcEngineWorker(model, n, self.device,
                                              self.engine_kwargs, tp=tp)
                else:
                    w = EngineWorker(model, self.device, self.engine_kwargs)
                self.workers[model] = w
            return w

    def _get_local_worker(self, model: str) -> EngineWorker:
        """Always an in-process engine (online Functions need direct access)."""
        key = f"local:{model}"
        with self._lock:
            w = self.workers.get(key)
            if w is None or not isinstance(w, EngineWorker):
                w = self.workers.get(model)
                if not isinstance(w, EngineWorker):
                    w = EngineWorker(model, self.device, self.engine_kwargs)
                    self.workers[key] = w
            return w

    # ---- lifecycle queries ----

    def get_job(self, job_id: str) -> JobRecord:
        job = self.jobs.get(job_id)
        if job is None:
            raise KeyError(f"unknown job {job_id!r}")
        return job

    def job_status(self, job_id: str) -> Dict[str, Any]:
        job = self.get_job(job_id)
        return {"job_status": {job_id: job.status.value}}

    def job_details(self, job_id: str) -> Dict[str, Any]:
        return {"job": self.get_job(job_id).to_public()}

    def list_jobs(self) -> Dict[str, Any]:
        jobs = sorted(self.jobs.values(), key=lambda j: j.datetime_created,
                      reverse=True)
        return {"jobs": [j.to_public() for j in jobs]}

    def cancel_job(self, job_id: str) -> Dict[str, Any]:
        job = self.get_job(job_id)
        if JobStatus.is_terminal(job.status):
            return {"job_status": {job_id: job.status.value}}
        job.status = JobStatus.CANCELLING
        worker = self.workers.get(job.model)
        if worker is not None:
            worker.cancel_job(job_id)
        job.status = JobStatus.CANCELLED
        job.datetime_completed = _now()
        self.persist_job(job)
        return {"job_status": {job_id: job.status.value}}

    def stream_progress(self, job_id: str, poll: float = 0.1):
        """Yield the line-JSON progress protocol (`sdk.py:354-390`)."""
        job = self.get_job(job_id)
        last_rows = -1
        last_tokens = (-1, -1)
        while True:
            rows = job.completed_rows
            if rows != last_rows:
                yield {"update_type": "progress", "result": rows}
                last_rows = rows
            tok = (job.input_tokens, job.output_tokens)
            if tok != last_tokens:
                yield {
                    "update_type": "tokens",
                    "result": {
                        "input_tokens": job.input_tokens,
                        "output_tokens": job.output_tokens,
                        "total_tokens_processed_per_second": round(
                            job.tokens_per_second, 2),
                    },
                }
                last_tokens = tok
            if JobStatus.is_terminal(job.status):
                return
            time.sleep(poll)

    def job_results(self, job_id: str, include_inputs: bool = False,
                    include_cumulative_logprobs: bool = False) -> Dict[str, Any]:
        job = self.get_job(job_id)
        if job.status != JobStatus.SUCCEEDED:
            raise RuntimeError(f"job {job_id} is {job.status.value}, not SUCCEEDED")
        spec = get_model_spec(job.model)
        results: Dict[str, Any] = {}
        if spec.embedding:
            results["outputs"] = job.embeddings
        else:
            results["outputs"] = job.outputs
        if include_inputs:
            results["inputs"] = [
                r if isinstance(r, str) else json.dumps(r) for r in job.inputs
            ]
        if include_cumulative_logprobs:
            results["cumulative_logprobs"] = job.cumulative_logprobs
        if not spec.embedding and any(c is not None for c in job.confidence_scores):
            results["confidence_score"] = job.confidence_scores
        if job.id_column_name and job.id_column_values is not None:
            results[job.id_column_name] = job.id_column_values
        return {"results": results}

    def get_quotas(self) -> Dict[str, Any]:
        return {"quotas": [dict(q) for q in self.quotas]}

    def try_authentication(self) -> Dict[str, Any]:
        return {"authenticated": True}

    # ---- persistence (detach/reattach + engine-restart resume) ----

    def wait_terminal(self, job_id: str, timeout: float) -> Optional[str]:
        """Block until the job reaches a terminal status (or timeout);
        returns the status or None. Notified by persist_job; a short
        condition-wait cap keeps missed notifications harmless."""
        deadline = time.time() + timeout
        while True:
            job = self.jobs.get(job_id)
            if job is not None and JobStatus.is_terminal(job.status):
                return job.status
            rem = deadline - time.time()
            if rem <= 0:
                return None
            with self._done_cv:
                self._done_cv.wait(min(rem, 0.25))

    def persist_job(self, job: JobRecord, with_results: bool = False) -> None:
        if JobStatus.is_terminal(job.status):
            with self._done_cv:
                self._done_cv.notify_all()
        path = os.path.join(self.home, "jobs", f"{job.job_id}.json")
        with open(path, "w") as f:
            json.dump(job.to_public(), f)
        if with_results:
            rpath = os.path.join(self.home, "job-results", f"{job.job_id}.json")
            with open(rpath, "w") as f:
                json.dump({
                    "outputs": job.outputs,
                    "embeddings": job.embeddings,
                    "cumulative_logprobs": job.cumulative_logprobs,
                    "confidence_score": job.confidence_scores,
                    "id_column_name": job.id_column_name,
                    "id_column_values": job.id_column_values,
                    "inputs": [r if isinstance(r, str) else json.dumps(r)
                               for r in job.inputs],
                    "json_schema": job.json_schema,
                    "system_prompt": job.system_prompt,
                    "sampling_params": job.sampling_params,
                }, f)

    def _load_persisted_jobs(self) -> None:
        jdir = os.path.join(self.home, "jobs")
        for fn in os.listdir(jdir):
            if not fn.endswith(".json"):
                continue
            try:
                with open(os.path.join(jdir, fn)) as f:
                    pub = json.load(f)
                job = JobRecord(job_id=pub["job_id"], model=pub.get("model", ""))
                job.status = JobStatus(pub.get("status", "UNKNOWN"))
                job.name = pub.get("name")
                job.description = pub.get("description")
                job.num_rows = pub.get("num_rows", 0)
                job.completed_rows = job.num_rows
                job.input_tokens = pub.get("input_tokens", 0)
                job.output_tokens = pub.get("output_tokens", 0)
                job.cost_estimate = pub.get("cost_estimate")
                job.job_cost = pub.get("job_cost", 0.0)
                job.datetime_created = pub.get("datetime_created", _now())
                job.datetime_started = pub.get("datetime_started")
                job.datetime_completed = pub.get("datetime_completed")
                rpath = os.path.join(self.home, "job-results", f"{job.job_id}.json")
                if os.path.exists(rpath):
                    with open(rpath) as f:
                        r = json.load(f)
                    job.outputs = r.get("outputs", [])
                    job.embeddings = r.get("embeddings", [])
                    job.cumulative_logprobs = r.get("cumulative_logprobs", [])
                    job.confidence_scores = r.get("confidence_score", [])
                    job.id_column_name = r.get("id_column_name")
                    job.id_column_values = r.get("id_column_values")
                    job.inputs = r.get("inputs", [])
                    job.json_schema = r.get("json_schema")
                    job.system_prompt = r.get("system_prompt")
                    job.sampling_params = r.get("sampling_params")
                if not JobStatus.is_terminal(job.status):
                    self._resume_or_fail(job)
                self.jobs[job.job_id] = job
            except Exception:
                continue

    def _resume_or_fail(self, job: JobRecord) -> None:
        """A job that was live when the service died: finish the MISSING rows
        from the persisted shards (resume), or fail it if inputs are gone."""
        try:
            is_embedding = get_model_spec(job.model).embedding
        except KeyError:
            is_embedding = False
        # embedding rows record their vector in `embeddings` (outputs stay
        # None by design) — judge completeness by the right column
        if is_embedding:
            done = [e is not None for e in (job.embeddings or [])]
            done += [False] * (job.num_rows - len(done))
        else:
            done = [o is not None for o in job.outputs]
        if not job.inputs or len(job.inputs) != job.num_rows:
            job.status = JobStatus.FAILED
            job.failure_reason = {
                "message": "service restarted mid-job with no persisted shards"}
            job.datetime_completed = _now()
            self.persist_job(job)
            return
        job.completed_rows = sum(done)
        if job.completed_rows == job.num_rows:
            job.status = JobStatus.SUCCEEDED
            job.datetime_completed = _now()
            self.persist_job(job, with_results=True)
            return
        # re-run only the incomplete rows through a fresh worker
        missing = [i for i, d in enumerate(done) if not d]
        job.status = JobStatus.RUNNING
        worker = self._get_worker(job.model)
        self.jobs[job.job_id] = job
        _submit_rows_into(worker, self, job, missing)

    def shutdown(self) -> None:
        for w in self.workers.values():
            w.shutdown()


def _submit_rows_into(worker, service: "JobService", job: JobRecord,
                      rows_idx) -> None:
    """Feed specific row indices of an existing JobRecord into a worker."""
    if hasattr(worker, "in_qs"):  # MultiProcEngineWorker
        opts = {
            "system_prompt": job.system_prompt,
            "json_schema": job.json_schema,
            "sampling_params": job.sampling_params,
            "random_seed_per_input": job.random_seed_per_input,
            "truncate_rows": job.truncate_rows,
            "priority": job.priority,
        }
        rows = [(i, job.inputs[i] if isinstance(job.inputs[i], str)
                 else json.dumps(job.inputs[i])) for i in rows_idx]
        worker._jobs[job.job_id] = (job, service, time.time())
        per = (len(rows) + worker.n_replicas - 1) // worker.n_replicas
        for rep in range(worker.n_replicas):
            shard = rows[rep * per:(rep + 1) * per]
            if shard:
                worker.in_qs[rep * worker.tp].put(
                    ("run_rows", job.job_id, shard, opts))
        return
    # in-process EngineWorker: admit only the missing rows of the SAME record
    worker.submit_partial(job, service, rows_idx)
