"""embed(): embedding job submit + await (reference `templates/embed.py:9-53`)."""

from __future__ import annotations

from typing import List, Optional, Union

from ..common import EmbeddingModelOptions
from ..interfaces import BaseSutroClient


class EmbeddingTemplates(BaseSutroClient):
    def embed(
        self,
        data,
        model: EmbeddingModelOptions = "qwen-3-embedding-0.6b",
        job_priority: int = 0,
        name: Optional[str] = None,
        description: Optional[str] = None,
        output_column: str = "inference_result",
        column: Union[str, List[str], None] = None,
        truncate_rows: bool = True,
    ):
        """Generate embeddings for each row; blocks until the job completes and
        returns the results frame (vectors in `output_column`)."""
        job_id = self.infer(
            data,
            model=model,
            name=name,
            description=description,
            column=column,
            output_column=output_column,
            job_priority=job_priority,
            truncate_rows=truncate_rows,
            stay_attached=False,
        )
        return self.await_job_completion(job_id, output_column=output_column,
                                         unpack_json=False)
