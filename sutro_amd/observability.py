"""Tracing: per-row batch traces + online run traces.

The reference traces to LangSmith (`/root/reference/sutro/observability.py`);
this environment has no network, so the same trace lifecycle is written to a
local JSONL sink (SUTRO_AMD_HOME/traces.jsonl), preserving the reference's key
design points:

- deterministic per-row trace UUIDs `uuid5(NAMESPACE, f"{job_id}-{row}")`
  (`observability.py:15-20`) so submission-time creation and retrieval-time
  completion line up without shared state;
- batch traces are created at submission and completed (with outputs and
  amortized token usage) at result retrieval;
- tracing failures degrade to warnings and never block the job.

Enable with SUTRO_TRACING=true (or LANGSMITH_TRACING=true for parity).
"""

from __future__ import annotations

import json
import logging
import os
import time
import uuid
from typing import Any, Optional

logger = logging.getLogger(__name__)

TRACE_NAMESPACE = uuid.UUID("2f6c8f6e-0000-4000-8000-6d6933353578")


def _row_run_id(job_id: str, row_index: int) -> str:
    return str(uuid.uuid5(TRACE_NAMESPACE, f"{job_id}-{row_index}"))


def tracing_enabled() -> bool:
    return (os.environ.get("SUTRO_TRACING", "").lower() == "true"
            or os.environ.get("LANGSMITH_TRACING", "").lower() == "true")


class ObservabilityMixin:
    """Mixed into the Sutro client; no-ops unless tracing is enabled."""

    def _init_observability(self) -> None:
        self._trace_path: Optional[str] = None
        if not tracing_enabled():
            return
        from .service.jobs import SUTRO_HOME

        home = getattr(self, "_home", None) or SUTRO_HOME
        os.makedirs(home, exist_ok=True)
        self._trace_path = os.path.join(home, "traces.jsonl")

    def _emit_trace(self, record: dict) -> None:
        if self._trace_path is None:
            return
        try:
            record.setdefault("project",
                              os.environ.get("SUTRO_TRACING_PROJECT",
                                             os.environ.get("LANGSMITH_PROJECT",
                                                            "default")))
            record.setdefault("ts", time.time())
            with open(self._trace_path, "a") as f:
                f.write(json.dumps(record) + "\n")
        except Exception as e:  # never block on tracing
            logger.warning("trace emit failed: %s", e)

    # ---- batch path ----

    def _create_batch_traces(self, job_id: str, rows: Any) -> None:
        if self._trace_path is None or not isinstance(rows, list):
            return
        for i, row in enumerate(rows):
            self._emit_trace({
                "event": "create", "run_id": _row_run_id(job_id, i),
                "job_id": job_id, "row": i,
                "inputs": row if isinstance(row, str) else json.dumps(row),
                "status": "open",
            })

    def _has_open_batch_traces(self, job_id: str) -> bool:
        if self._trace_path is None or not os.path.exists(self._trace_path):
            return False
        open_ids = set()
        with open(self._trace_path) as f:
            for line in f:
                try:
                    r = json.loads(line)
                except json.JSONDecodeError:
                    continue
                if r.get("job_id") != job_id:
                    continue
                if r.get("event") == "create":
                    open_ids.add(r.get("run_id"))
                elif r.get("event") == "complete":
                    open_ids.discard(r.get("run_id"))
        return bool(open_ids)

    def _complete_batch_traces(self, job_id: str, df, output_column: str) -> None:
        if self._trace_path is None:
            return
        if not self._has_open_batch_traces(job_id):
            return
        try:
            job = self._fetch_job(job_id)  # type: ignore[attr-defined]
            n = max(1, len(df))
            in_tok = (job.get("input_tokens") or 0) // n
            out_tok = (job.get("output_tokens") or 0) // n
            outputs = df[output_column].tolist() if output_column in df else []
            for i, out in enumerate(outputs):
                self._emit_trace({
                    "event": "complete", "run_id": _row_run_id(job_id, i),
                    "job_id": job_id, "row": i, "output": out,
                    "usage": {"input_tokens": in_tok, "output_tokens": out_tok},
                    "status": "closed",
                })
        except Exception as e:
            logger.warning("batch trace completion failed: %s", e)

    # ---- online path ----

    def _trace_online_run(self, name: str, input_data: Any, response: dict,
                          latency_s: float) -> None:
        if self._trace_path is None:
            return
        self._emit_trace({
            "event": "online_run", "name": name,
            "run_id": response.get("run_id"),
            "inputs": input_data, "response": response.get("response"),
            "confidence": response.get("confidence"),
            "usage": response.get("usage"),
            "latency_s": round(latency_s, 4),
        })
