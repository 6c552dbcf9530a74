"""Input preparation, schema normalization, model-name options, terminal utils.

Behavioral contract follows the reference (`/root/reference/sutro/common.py`):
- `prepare_input_data` (`common.py:116-167`): list -> as-is; DataFrame -> column
  extraction or multi-column concatenation; "dataset-..." -> passthrough;
  http(s) URL -> passthrough; local .csv/.parquet/.txt -> read from disk.
- `do_dataframe_column_concatenation` (`common.py:77-113`): list of column names
  interleaved with literal separator strings.
- `normalize_output_schema` (`common.py:170-181`): pydantic BaseModel -> JSON
  schema dict; dict passes through.

DataFrames are pandas here (the reference also accepts polars; polars is not in
this environment — pyarrow Tables are accepted instead).
"""

from __future__ import annotations

import os
import sys
from typing import Any, Dict, List, Literal, Optional, Type, Union

import pandas as pd
import pyarrow as pa
from pydantic import BaseModel

from .models.registry import MODEL_REGISTRY

# Typed options for autocompletion; `| str` escape hatch like the reference.
EmbeddingModelOptions = Union[
    Literal[
        "embeddinggemma-300m",
        "qwen-3-embedding-0.6b",
        "qwen-3-embedding-4b",
        "qwen-3-embedding-8b",
    ],
    str,
]

ModelOptions = Union[EmbeddingModelOptions, str]

DEFAULT_MODEL = "qwen-3-0.6b"


def list_models() -> List[str]:
    return sorted(MODEL_REGISTRY)


def is_jupyter() -> bool:
    return not sys.stdout.isatty()


def make_clickable_link(url: str, text: Optional[str] = None) -> str:
    """OSC-8 clickable hyperlink for supporting terminals."""
    text = text or url
    return f"\033]8;;{url}\033\\{text}\033]8;;\033\\"


_COLORS = {"success": "\033[32m", "fail": "\033[31m", "callout": "\033[36m"}
_RESET = "\033[0m"


def to_colored_text(
    text: str, state: Optional[Literal["success", "fail", "callout"]] = None
) -> str:
    """Color text by state: success=green, fail=red, callout=cyan, default plain."""
    color = _COLORS.get(state or "")
    if not color or is_jupyter():
        return text
    return f"{color}{text}{_RESET}"


def _as_pandas(df: Any) -> pd.DataFrame:
    if isinstance(df, pd.DataFrame):
        return df
    if isinstance(df, pa.Table):
        return df.to_pandas()
    raise TypeError(f"unsupported DataFrame type: {type(df)!r}")


def do_dataframe_column_concatenation(
    df: Any, column: List[str]
) -> List[str]:
    """Concatenate columns into one string per row.

    `column` is a list whose items are either column names or literal separator
    strings (any item that is not a column of `df` is treated as a separator),
    mirroring reference `common.py:77-113`.
    """
    pdf = _as_pandas(df)
    parts: List[pd.Series] = []
    for item in column:
        if item in pdf.columns:
            parts.append(pdf[item].astype(str))
        else:
            parts.append(pd.Series([item] * len(pdf), index=pdf.index))
    if not parts:
        raise ValueError("empty column list")
    out = parts[0]
    for p in parts[1:]:
        out = out.str.cat(p)
    return out.tolist()


def prepare_input_data(
    data: Union[List[Any], pd.DataFrame, pa.Table, str],
    column: Union[str, List[str], None] = None,
) -> Union[List[Any], str]:
    """Normalize user input into the job payload's `inputs` field.

    Returns either a list of rows or a passthrough string (dataset ID / URL).
    Reference semantics: `common.py:116-167`.
    """
    if isinstance(data, list):
        return data
    if isinstance(data, (pd.DataFrame, pa.Table)):
        if column is None:
            raise ValueError("a `column` is required when passing a DataFrame")
        if isinstance(column, list):
            return do_dataframe_column_concatenation(data, column)
        pdf = _as_pandas(data)
        if column not in pdf.columns:
            raise ValueError(f"column {column!r} not found in DataFrame")
        return pdf[column].tolist()
    if isinstance(data, str):
        if data.startswith("dataset-"):
            return data  # dataset ID passthrough; column travels separately
        if data.startswith(("http://", "https://")):
            return data  # URL passthrough
        ext = os.path.splitext(data)[1].lower()
        if ext == ".csv":
            df = pd.read_csv(data)
        elif ext == ".parquet":
            df = pd.read_parquet(data)
        elif ext == ".txt":
            with open(data) as f:
                return [line.rstrip("\n") for line in f if line.strip()]
        else:
            raise ValueError(f"unsupported file type: {data!r}")
        if column is None:
            raise ValueError(f"a `column` is required when passing a file path ({data!r})")
        if isinstance(column, list):
            return do_dataframe_column_concatenation(df, column)
        if column not in df.columns:
            raise ValueError(f"column {column!r} not found in {data!r}")
        return df[column].tolist()
    raise TypeError(f"unsupported data type: {type(data)!r}")


def normalize_output_schema(
    output_schema: Union[Dict[str, Any], Type[BaseModel]],
) -> Dict[str, Any]:
    """pydantic BaseModel class -> model_json_schema(); dict passes through."""
    if isinstance(output_schema, dict):
        return output_schema
    if isinstance(output_schema, type) and issubclass(output_schema, BaseModel):
        return output_schema.model_json_schema()
    raise TypeError(
        "output_schema must be a dict JSON schema or a pydantic BaseModel subclass"
    )


def fancy_tqdm(total: int, desc: str = "", style: int = 1, postfix: Optional[str] = None):
    """A styled tqdm progress bar (reference: `common.py:214-270`)."""
    from tqdm import tqdm

    return tqdm(
        total=total,
        desc=desc,
        bar_format="{l_bar}{bar}| {n_fmt}/{total_fmt} [{elapsed}<{remaining}]"
        + (f" {postfix}" if postfix else ""),
        colour="magenta" if style == 1 else "cyan",
        dynamic_ncols=True,
    )
