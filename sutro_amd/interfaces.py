"""Job-status model and the client protocol.

Behavioral contract mirrors the reference client (`/root/reference/sutro/interfaces.py:11-91`):
the same status names, the same terminal-state logic, and a protocol class so the
task-template mixins can call ``infer`` / ``await_job_completion`` without importing
the concrete client.
"""

from __future__ import annotations

from enum import Enum
from typing import Any, Dict, List, Optional, Protocol, Type, Union, runtime_checkable


class JobStatus(str, Enum):
    """Lifecycle states of a batch job.

    State machine: QUEUED -> STARTING -> RUNNING -> {SUCCEEDED, FAILED, CANCELLED}
    with CANCELLING as the transient state entered by a cancel request.
    """

    QUEUED = "QUEUED"
    STARTING = "STARTING"
    RUNNING = "RUNNING"
    SUCCEEDED = "SUCCEEDED"
    FAILED = "FAILED"
    CANCELLING = "CANCELLING"
    CANCELLED = "CANCELLED"
    UNKNOWN = "UNKNOWN"

    @classmethod
    def terminal_statuses(cls) -> List["JobStatus"]:
        return [cls.SUCCEEDED, cls.FAILED, cls.CANCELLED]

    @classmethod
    def is_terminal(cls, status: Union["JobStatus", str]) -> bool:
        try:
            status = cls(status)
        except ValueError:
            return False
        return status in cls.terminal_statuses()


@runtime_checkable
class BaseSutroClient(Protocol):
    """Protocol implemented by :class:`sutro_amd.sdk.Sutro`.

    Task-template mixins (classify/embed/score/rank) are written against this
    interface (reference: `sutro/interfaces.py:11-66`).
    """

    def infer(
        self,
        data: Any,
        model: str = ...,
        name: Optional[str] = None,
        description: Optional[str] = None,
        column: Union[str, List[str], None] = None,
        output_column: str = "inference_result",
        job_priority: int = 0,
        output_schema: Union[Dict[str, Any], Type, None] = None,
        sampling_params: Optional[dict] = None,
        system_prompt: Optional[str] = None,
        dry_run: bool = False,
        stay_attached: Optional[bool] = None,
        random_seed_per_input: bool = False,
        truncate_rows: bool = True,
        id_column: Optional[str] = None,
    ) -> Optional[str]: ...

    def await_job_completion(
        self,
        job_id: str,
        timeout: int = 7200,
        obtain_results: bool = True,
        with_original_df: Any = None,
        output_column: str = "inference_result",
        unpack_json: bool = True,
    ) -> Any: ...

    def get_job_results(self, job_id: str, **kwargs: Any) -> Any: ...

    def get_job_status(self, job_id: str) -> Optional[str]: ...
