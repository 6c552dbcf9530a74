"""Dataset storage backing the dataset endpoints.

Reference contract (`/root/reference/sutro/sdk.py:1369-1596`): create returns a
"dataset-..." ID; files are uploaded/downloaded by name; list-datasets reports
schema + timestamps. Storage is a local directory tree under SUTRO_AMD_HOME.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from typing import Any, Dict, List

import pandas as pd


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime())


class DatasetStore:
    def __init__(self, home: str):
        self.root = os.path.join(home, "datasets")
        os.makedirs(self.root, exist_ok=True)

    def _dir(self, dataset_id: str) -> str:
        d = os.path.join(self.root, dataset_id)
        if not os.path.isdir(d):
            raise KeyError(f"unknown dataset {dataset_id!r}")
        return d

    def create(self) -> Dict[str, Any]:
        dataset_id = f"dataset-{uuid.uuid4().hex[:12]}"
        d = os.path.join(self.root, dataset_id)
        os.makedirs(os.path.join(d, "files"))
        with open(os.path.join(d, "meta.json"), "w") as f:
            json.dump({"dataset_id": dataset_id, "datetime_added": _now(),
                       "updated_at": _now(), "schema": {}}, f)
        return {"dataset_id": dataset_id}

    def upload(self, dataset_id: str, file_name: str, data: bytes) -> None:
        d = self._dir(dataset_id)
        path = os.path.join(d, "files", os.path.basename(file_name))
        with open(path, "wb") as f:
            f.write(data)
        meta_path = os.path.join(d, "meta.json")
        with open(meta_path) as f:
            meta = json.load(f)
        meta["updated_at"] = _now()
        try:
            df = self._read_file(path)
            meta["schema"] = {c: str(t) for c, t in df.dtypes.items()}
        except Exception:
            pass
        with open(meta_path, "w") as f:
            json.dump(meta, f)

    @staticmethod
    def _read_file(path: str) -> pd.DataFrame:
        ext = os.path.splitext(path)[1].lower()
        if ext == ".parquet":
            return pd.read_parquet(path)
        if ext == ".csv":
            return pd.read_csv(path)
        if ext == ".txt":
            with open(path) as f:
                return pd.DataFrame({"text": [l.rstrip("\n") for l in f]})
        raise ValueError(f"unsupported dataset file type: {path}")

    def list_datasets(self) -> Dict[str, Any]:
        out: List[Dict[str, Any]] = []
        for name in sorted(os.listdir(self.root)):
            meta_path = os.path.join(self.root, name, "meta.json")
            if os.path.exists(meta_path):
                with open(meta_path) as f:
                    out.append(json.load(f))
        return {"datasets": out}

    def list_files(self, dataset_id: str) -> Dict[str, Any]:
        d = self._dir(dataset_id)
        return {"files": sorted(os.listdir(os.path.join(d, "files")))}

    def download(self, dataset_id: str, file_name: str) -> bytes:
        d = self._dir(dataset_id)
        with open(os.path.join(d, "files", os.path.basename(file_name)), "rb") as f:
            return f.read()

    def read_all(self, dataset_id: str) -> pd.DataFrame:
        """Concatenate every file of the dataset into one DataFrame."""
        d = self._dir(dataset_id)
        frames = []
        for fn in sorted(os.listdir(os.path.join(d, "files"))):
            frames.append(self._read_file(os.path.join(d, "files", fn)))
        if not frames:
            raise ValueError(f"dataset {dataset_id} has no files")
        return pd.concat(frames, ignore_index=True)
