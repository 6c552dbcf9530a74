"""classify(): structured-output classification with scratchpad reasoning.

Behavioral parity with reference `templates/classification.py:12-117`:
classes as list or {label: description} dict, a system prompt enumerating
them, a pydantic output schema {scratchpad, classification}, blocking until
completion, and optional scratchpad stripping.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Union

import pandas as pd
from pydantic import BaseModel

from ..common import DEFAULT_MODEL, ModelOptions
from ..interfaces import BaseSutroClient


class ClassificationTemplates(BaseSutroClient):
    def classify(
        self,
        data,
        classes: Union[Dict[str, str], List[str]],
        model: ModelOptions = DEFAULT_MODEL,
        job_priority: int = 0,
        name: Optional[str] = None,
        description: Optional[str] = None,
        output_column: str = "inference_result",
        column: Union[str, List[str], None] = None,
        truncate_rows: bool = True,
        include_scratchpad: bool = False,
    ):
        """Classify each row into one of `classes`; blocks until done.

        Returns a DataFrame: with `include_scratchpad` both `scratchpad` and
        `classification` columns are kept; otherwise only `classification`.
        """
        if isinstance(classes, dict):
            formatted = "\n".join(f"- {label}: {desc}" for label, desc in classes.items())
            labels = list(classes)
        else:
            formatted = "\n".join(f"- {c}" for c in classes)
            labels = list(classes)

        system_prompt = (
            "You are an expert classifier. Categorize the input into exactly one "
            "of the following classes.\n\n## Classes\n" + formatted + "\n\n"
            "Think step by step in the scratchpad, then give the final class."
        )

        class ClassificationOutput(BaseModel):
            scratchpad: str
            classification: str

        schema = ClassificationOutput.model_json_schema()
        # constrain the final field to the class labels (guided decoding
        # enforces it token-by-token)
        schema["properties"]["classification"] = {"enum": labels}
        schema["properties"]["scratchpad"]["maxLength"] = 512

        job_id = self.infer(
            data,
            model=model,
            name=name,
            description=description,
            column=column,
            output_column=output_column,
            job_priority=job_priority,
            output_schema=schema,
            system_prompt=system_prompt,
            truncate_rows=truncate_rows,
            stay_attached=False,
        )
        res = self.await_job_completion(job_id, output_column=output_column)
        if res is None:
            return None
        if not include_scratchpad and isinstance(res, pd.DataFrame):
            if "scratchpad" in res.columns:
                res = res.drop(columns=["scratchpad"])
        return res
