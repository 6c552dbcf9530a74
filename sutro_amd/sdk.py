"""The Sutro client: job submission, lifecycle, results, datasets, quotas.

API surface mirrors the reference client (`/root/reference/sutro/sdk.py`), with
two transports behind the single `do_request` choke point (`sdk.py:103-177`):

- "local" (default): an in-process :class:`sutro_amd.service.jobs.JobService`
  running the MI355X engine (or its CPU dev path) — no network.
- an http(s) base_url: the same endpoint contract over HTTP (served by
  `sutro_amd.service.http_api`), with the reference's retry policy: only
  524-style timeouts retry, with exponential backoff, and job submission never
  retries (non-idempotent POST, `sdk.py:247-254`).
"""

from __future__ import annotations

import json
import os
import time
import uuid
from typing import Any, Dict, List, Optional, Type, Union

import pandas as pd
import pyarrow as pa
from pydantic import BaseModel

from .common import (
    DEFAULT_MODEL,
    ModelOptions,
    fancy_tqdm,
    normalize_output_schema,
    prepare_input_data,
    to_colored_text,
)
from .interfaces import JobStatus
from .observability import ObservabilityMixin
from .templates.classification import ClassificationTemplates
from .templates.embed import EmbeddingTemplates
from .templates.evals import EvalTemplates
from .validation import check_for_api_key, load_config

JOB_NAME_MAX_LEN = 45
JOB_DESC_MAX_LEN = 512
POLL_INTERVAL = 0.25  # local service is in-process; poll fast


class _LocalResponse:
    """requests.Response-alike wrapping a local service call."""

    def __init__(self, payload: Any = None, status_code: int = 200,
                 stream=None, content: bytes = b""):
        self._payload = payload
        self.status_code = status_code
        self._stream = stream
        self.content = content
        self.text = json.dumps(payload) if payload is not None else ""

    def json(self) -> Any:
        return self._payload

    def iter_lines(self, decode_unicode: bool = False):
        for item in self._stream or ():
            line = json.dumps(item)
            yield line if decode_unicode else line.encode()


class LocalTransport:
    """Routes the endpoint table onto the in-process service."""

    def __init__(self, home: Optional[str] = None, device: str = "auto",
                 engine_kwargs: Optional[dict] = None):
        from .service.datasets import DatasetStore
        from .service.functions import FunctionStore
        from .service.jobs import JobService

        self.service = JobService(home=home, device=device,
                                  engine_kwargs=engine_kwargs)
        self.datasets = DatasetStore(self.service.home)
        self.functions = FunctionStore(self.service.home)

    def request(self, method: str, path: str, payload: Optional[dict] = None,
                files: Optional[dict] = None, stream: bool = False):
        svc = self.service
        payload = payload or {}
        try:
            if path == "/batch-inference":
                return _LocalResponse(svc.submit_job(payload))
            if path.startswith("/stream-job-progress/"):
                job_id = path.rsplit("/", 1)[1]
                return _LocalResponse(stream=svc.stream_progress(job_id))
            if path.startswith("/job-status/"):
                return _LocalResponse(svc.job_status(path.rsplit("/", 1)[1]))
            if path.startswith("/jobs/"):
                return _LocalResponse(svc.job_details(path.rsplit("/", 1)[1]))
            if path == "/job-results":
                return _LocalResponse(svc.job_results(
                    payload["job_id"],
                    include_inputs=payload.get("include_inputs", False),
                    include_cumulative_logprobs=payload.get(
                        "include_cumulative_logprobs", False)))
            if path == "/list-jobs":
                return _LocalResponse(svc.list_jobs())
            if path.startswith("/job-cancel/"):
                return _LocalResponse(svc.cancel_job(path.rsplit("/", 1)[1]))
            if path == "/try-authentication":
                return _LocalResponse(svc.try_authentication())
            if path == "/get-quotas":
                return _LocalResponse(svc.get_quotas())
            if path == "/create-dataset":
                return _LocalResponse(self.datasets.create())
            if path == "/upload-to-dataset":
                dataset_id = payload["dataset_id"]
                for name, fobj in (files or {}).items():
                    data = fobj.read() if hasattr(fobj, "read") else fobj
                    self.datasets.upload(dataset_id, name, data)
                return _LocalResponse({"ok": True})
            if path == "/list-datasets":
                return _LocalResponse(self.datasets.list_datasets())
            if path == "/list-dataset-files":
                return _LocalResponse(self.datasets.list_files(payload["dataset_id"]))
            if path == "/download-from-dataset":
                data = self.datasets.download(payload["dataset_id"],
                                              payload["file_name"])
                return _LocalResponse(content=data)
            if path == "/functions/run":
                return _LocalResponse(
                    self.functions.run(self.service, payload["name"],
                                       payload.get("input_data")))
            if path == "/functions/create":
                return _LocalResponse(self.functions.create(payload))
            return _LocalResponse({"error": f"unknown path {path}"}, 404)
        except KeyError as e:
            return _LocalResponse({"error": str(e)}, 404)
        except (ValueError, RuntimeError) as e:
            return _LocalResponse({"error": str(e)}, 400)


class Sutro(ObservabilityMixin, EmbeddingTemplates, ClassificationTemplates,
            EvalTemplates):
    def __init__(
        self,
        api_key: Optional[str] = None,
        base_url: Optional[str] = None,
        serving_base_url: Optional[str] = None,
        home: Optional[str] = None,
        device: str = "auto",
        engine_kwargs: Optional[dict] = None,
    ):
        config = load_config()
        self.api_key = api_key or check_for_api_key() or "local"
        self.base_url = base_url or config.get("base_url") or "local"
        self.serving_base_url = serving_base_url or config.get("serving_base_url") \
            or self.base_url
        self._home = home
        self._device = device
        self._engine_kwargs = engine_kwargs
        self._local: Optional[LocalTransport] = None
        self._init_observability()

    # ---- configuration ----

    def set_api_key(self, api_key: str) -> None:
        self.api_key = api_key

    def set_base_url(self, base_url: str) -> None:
        self.base_url = base_url

    def set_serving_base_url(self, serving_base_url: str) -> None:
        self.serving_base_url = serving_base_url

    @property
    def _is_local(self) -> bool:
        return not str(self.base_url).startswith(("http://", "https://"))

    def _local_transport(self) -> LocalTransport:
        if self._local is None:
            self._local = LocalTransport(home=self._home, device=self._device,
                                         engine_kwargs=self._engine_kwargs)
        return self._local

    # ---- transport (single choke point, reference `sdk.py:103-177`) ----

    def do_request(
        self,
        method: str,
        path: str,
        payload: Optional[dict] = None,
        files: Optional[dict] = None,
        stream: bool = False,
        max_retries: int = 5,
        base_url: Optional[str] = None,
    ):
        if self._is_local and base_url is None:
            return self._local_transport().request(method, path, payload, files,
                                                   stream)
        import requests

        url = (base_url or self.base_url).rstrip("/") + path
        headers = {"Authorization": f"Key {self.api_key}"}
        attempt = 0
        while True:
            try:
                if method.upper() == "GET":
                    resp = requests.get(url, headers=headers, stream=stream,
                                        timeout=600)
                elif method.upper() == "POST":
                    if files:
                        import base64

                        body = dict(payload or {})
                        body["files"] = {
                            name: base64.b64encode(
                                f.read() if hasattr(f, "read") else f
                            ).decode()
                            for name, f in files.items()
                        }
                        resp = requests.post(url, headers=headers, json=body,
                                             timeout=600)
                    else:
                        resp = requests.post(url, headers=headers, json=payload,
                                             stream=stream, timeout=600)
                else:
                    raise ValueError(f"unsupported method {method}")
            except requests.RequestException:
                if attempt >= max_retries:
                    raise
                time.sleep(2 ** attempt)
                attempt += 1
                continue
            # retry ONLY gateway-timeout-style responses (Cloudflare 524)
            if resp.status_code == 524 and attempt < max_retries:
                time.sleep(2 ** attempt)
                attempt += 1
                continue
            return resp

    # ---- job submission ----

    def _run_one_batch_inference(
        self, data, model, column, output_column, job_priority, json_schema,
        sampling_params, system_prompt, dry_run, stay_attached,
        random_seed_per_input, truncate_rows, name, description, id_column,
    ):
        if name is not None and len(name) > JOB_NAME_MAX_LEN:
            raise ValueError(f"name must be <= {JOB_NAME_MAX_LEN} characters")
        if description is not None and len(description) > JOB_DESC_MAX_LEN:
            raise ValueError(f"description must be <= {JOB_DESC_MAX_LEN} characters")
        is_url_input = isinstance(data, str) and data.startswith(("http://", "https://"))
        if id_column is not None and not is_url_input:
            raise ValueError(
                "id_column is only supported for HTTP(S) CSV/Parquet URL inputs")

        inputs = prepare_input_data(data, column)
        payload: Dict[str, Any] = {
            "model": model,
            "inputs": inputs,
            "job_priority": job_priority,
            "json_schema": json_schema,
            "system_prompt": system_prompt,
            "cost_estimate": dry_run,
            "sampling_params": sampling_params,
            "random_seed_per_input": random_seed_per_input,
            "truncate_rows": truncate_rows,
            "name": name,
            "description": description,
        }
        if isinstance(inputs, str):
            payload["column_name"] = column
        if id_column is not None:
            payload["id_column_name"] = id_column

        # submission is not idempotent: never retry (`sdk.py:247-254`)
        resp = self.do_request("POST", "/batch-inference", payload, max_retries=0)
        if resp.status_code != 200:
            raise RuntimeError(f"job submission failed: {resp.status_code} "
                               f"{resp.text}")
        job_id = resp.json()["results"]

        if dry_run:
            self.await_job_completion(job_id, obtain_results=False, quiet=True)
            estimate = self._get_job_cost_estimate(job_id)
            print(to_colored_text(f"Estimated job cost: ${estimate}", "callout"))
            return estimate

        print(to_colored_text(f"✔ Job {job_id} submitted", "success"))
        if stay_attached:
            ok = self._attach_progress(job_id)
            if not ok:
                return job_id
            results = self._fetch_results_with_retry(job_id)
            if results is None:
                return job_id
            df = self._results_to_frame(results, output_column)
            if isinstance(data, pd.DataFrame):
                out = data.copy()
                out[output_column] = df[output_column].values
                print(out.head())
            else:
                print(df.head())
            return job_id
        return job_id

    def infer(
        self,
        data: Union[List, pd.DataFrame, pa.Table, str],
        model: ModelOptions = DEFAULT_MODEL,
        name: Optional[str] = None,
        description: Optional[str] = None,
        column: Union[str, List[str], None] = None,
        output_column: str = "inference_result",
        job_priority: int = 0,
        output_schema: Union[Dict[str, Any], Type[BaseModel], None] = None,
        sampling_params: Optional[dict] = None,
        system_prompt: Optional[str] = None,
        dry_run: bool = False,
        stay_attached: Optional[bool] = None,
        random_seed_per_input: bool = False,
        truncate_rows: bool = True,
        id_column: Optional[str] = None,
    ):
        """Run batch inference. Returns the job ID (or the cost estimate for
        dry runs). See the reference docstring (`sdk.py:465-537`) for argument
        semantics — they are preserved here."""
        if stay_attached is None:
            stay_attached = job_priority == 0
        json_schema = normalize_output_schema(output_schema) if output_schema else None
        return self._run_one_batch_inference(
            data, model, column, output_column, job_priority, json_schema,
            sampling_params, system_prompt, dry_run, stay_attached,
            random_seed_per_input, truncate_rows, name, description, id_column,
        )

    def infer_per_model(
        self,
        data,
        models: List[str],
        **kwargs,
    ) -> List[str]:
        """Submit the same inference across several models; returns job IDs
        (reference `sdk.py:745-851`)."""
        kwargs["stay_attached"] = False
        return [self.infer(data, model=m, **kwargs) for m in models]

    # ---- Functions (online + batch) ----

    def create_function(self, name: str, model: str,
                        system_prompt: Optional[str] = None,
                        output_schema: Union[dict, Type[BaseModel], None] = None):
        """Register a named Function served by `run_function` (local extension;
        the reference's Functions are deployed server-side)."""
        schema = normalize_output_schema(output_schema) if output_schema else None
        resp = self.do_request("POST", "/functions/create", {
            "name": name, "model": model, "system_prompt": system_prompt,
            "output_schema": schema})
        if resp.status_code != 200:
            raise RuntimeError(f"create_function failed: {resp.text}")
        return resp.json()

    def run_function(self, name: str, input_data: Any,
                     stay_attached: bool = True):
        """Run a single Function invocation (reference `sdk.py:539-615`).
        Returns the full response dict {response, confidence, predictions,
        run_id, usage}."""
        t0 = time.time()
        resp = self.do_request("POST", "/functions/run",
                               {"name": name, "input_data": input_data},
                               base_url=None if self._is_local
                               else self.serving_base_url)
        if resp.status_code != 200:
            raise RuntimeError(f"run_function failed: {resp.status_code} {resp.text}")
        out = resp.json()
        self._trace_online_run(name, input_data, out, time.time() - t0)
        return out

    def batch_run_function(
        self,
        data,
        function: str,
        column: Union[str, List[str], None] = None,
        output_column: str = "inference_result",
        job_priority: int = 0,
        stay_attached: Optional[bool] = None,
        name: Optional[str] = None,
        description: Optional[str] = None,
    ):
        """Batch Functions execution (reference `sdk.py:617-743`): rows are
        full records (dicts) when a DataFrame is passed without a column."""
        if isinstance(data, (pd.DataFrame, pa.Table)) and column is None:
            pdf = data.to_pandas() if isinstance(data, pa.Table) else data
            rows: Any = [json.dumps(r) for r in pdf.to_dict(orient="records")]
        else:
            rows = prepare_input_data(data, column)
        job_id = self.infer(
            rows, model=function, output_column=output_column,
            job_priority=job_priority, stay_attached=bool(stay_attached),
            truncate_rows=False, name=name, description=description,
        )
        self._create_batch_traces(job_id, rows)
        return job_id

    # ---- attach / progress ----

    def _attach_progress(self, job_id: str) -> bool:
        """Stream progress updates into a tqdm bar until terminal state."""
        ok = self._await_job_start(job_id)
        if not ok:
            return False
        job = self._fetch_job(job_id)
        total = job.get("num_rows", 0) or 1
        resp = self.do_request("GET", f"/stream-job-progress/{job_id}", stream=True)
        bar = fancy_tqdm(total=total, desc="Processing")
        done = 0
        try:
            for line in resp.iter_lines(decode_unicode=True):
                if not line:
                    continue
                update = json.loads(line)
                if update.get("update_type") == "progress":
                    new_done = int(update["result"])
                    bar.update(max(0, new_done - done))
                    done = new_done
                elif update.get("update_type") == "tokens":
                    r = update["result"]
                    tps = r.get("total_tokens_processed_per_second", 0)
                    bar.set_postfix_str(f"{tps:,.0f} tok/s")
        finally:
            bar.close()
        status = self.get_job_status(job_id)
        if status != JobStatus.SUCCEEDED.value:
            print(to_colored_text(f"Job {job_id} ended with status {status}", "fail"))
            return False
        print(to_colored_text("✔ Job completed", "success"))
        return True

    def attach(self, job_id: str):
        """Reattach to a running job (reference `sdk.py:853-964`)."""
        job = self._fetch_job(job_id)
        status = job.get("status")
        if JobStatus.is_terminal(status):
            print(to_colored_text(f"Job {job_id} already {status}", "callout"))
            return
        self._attach_progress(job_id)

    def _await_job_start(self, job_id: str, timeout: int = 7200) -> bool:
        start = time.time()
        while time.time() - start < timeout:
            status = self.get_job_status(job_id)
            if status in (JobStatus.RUNNING.value, JobStatus.STARTING.value):
                return True
            if JobStatus.is_terminal(status):
                if status == JobStatus.SUCCEEDED.value:
                    return True
                reason = self._get_failure_reason(job_id)
                print(to_colored_text(f"Job {job_id} failed to start: {reason}",
                                      "fail"))
                return False
            time.sleep(POLL_INTERVAL)
        return False

    # ---- job queries ----

    def _list_all_jobs_for_user(self) -> List[dict]:
        resp = self.do_request("GET", "/list-jobs")
        return resp.json().get("jobs", [])

    def list_jobs(self) -> pd.DataFrame:
        return pd.DataFrame(self._list_all_jobs_for_user())

    def _fetch_job(self, job_id: str) -> dict:
        resp = self.do_request("GET", f"/jobs/{job_id}")
        if resp.status_code != 200:
            raise RuntimeError(f"failed to fetch job {job_id}: {resp.text}")
        return resp.json()["job"]

    def _get_job_cost_estimate(self, job_id: str):
        return self._fetch_job(job_id).get("cost_estimate")

    def _get_failure_reason(self, job_id: str) -> str:
        fr = self._fetch_job(job_id).get("failure_reason") or {}
        return fr.get("message", "unknown")

    def get_job_status(self, job_id: str) -> Optional[str]:
        resp = self.do_request("GET", f"/job-status/{job_id}")
        if resp.status_code != 200:
            return None
        return resp.json()["job_status"].get(job_id)

    def cancel_job(self, job_id: str) -> dict:
        resp = self.do_request("GET", f"/job-cancel/{job_id}")
        return resp.json()

    # ---- results ----

    def _cache_dir(self) -> str:
        from .service.jobs import SUTRO_HOME

        d = os.path.join(self._home or SUTRO_HOME, "job-results-cache")
        os.makedirs(d, exist_ok=True)
        return d

    def _fetch_results_with_retry(self, job_id: str, attempts: int = 20,
                                  delay: float = 0.5) -> Optional[dict]:
        """SUCCEEDED may race results materialization; retry (`sdk.py:407-425`)."""
        for _ in range(attempts):
            resp = self.do_request("POST", "/job-results", {
                "job_id": job_id, "include_inputs": False,
                "include_cumulative_logprobs": False})
            if resp.status_code == 200:
                return resp.json()["results"]
            time.sleep(delay)
        return None

    def _results_to_frame(self, results: dict, output_column: str) -> pd.DataFrame:
        cols: Dict[str, Any] = {}
        if "inputs" in results:
            cols["inputs"] = results["inputs"]
        for k, v in results.items():
            if k not in ("outputs", "inputs", "cumulative_logprobs",
                         "confidence_score"):
                cols[k] = v  # id / metadata columns
        cols[output_column] = results["outputs"]
        if "cumulative_logprobs" in results:
            cols["cumulative_logprobs"] = results["cumulative_logprobs"]
        if "confidence_score" in results:
            cols["confidence_score"] = results["confidence_score"]
        return pd.DataFrame(cols)

    def get_job_results(
        self,
        job_id: str,
        include_inputs: bool = False,
        include_cumulative_logprobs: bool = False,
        with_original_df: Union[pd.DataFrame, pa.Table, None] = None,
        output_column: str = "inference_result",
        unpack_json: bool = True,
        disable_cache: bool = False,
    ) -> pd.DataFrame:
        """Materialize job results as a DataFrame, input-ordered, with local
        parquet caching and structured-output JSON unpacking
        (reference `sdk.py:1131-1340`)."""
        cache_path = os.path.join(self._cache_dir(), f"{job_id}.snappy.parquet")
        df: Optional[pd.DataFrame] = None
        expected = [output_column]
        if include_inputs:
            expected.append("inputs")
        if include_cumulative_logprobs:
            expected.append("cumulative_logprobs")
        if not disable_cache and os.path.exists(cache_path):
            try:
                schema_cols = set(pa.parquet.read_schema(cache_path).names)
                if set(expected).issubset(schema_cols):
                    df = pd.read_parquet(cache_path)
            except Exception:
                df = None
        if df is None:
            resp = self.do_request("POST", "/job-results", {
                "job_id": job_id,
                "include_inputs": include_inputs,
                "include_cumulative_logprobs": include_cumulative_logprobs,
            })
            if resp.status_code != 200:
                raise RuntimeError(f"failed to fetch results for {job_id}: "
                                   f"{resp.text}")
            results = resp.json()["results"]
            df = self._results_to_frame(results, output_column)
            if not disable_cache:
                try:
                    df.to_parquet(cache_path, compression="snappy")
                except Exception:
                    pass

        self._complete_batch_traces(job_id, df, output_column)

        if unpack_json:
            df = self._maybe_unpack_json(df, output_column)

        if with_original_df is not None:
            orig = (with_original_df.to_pandas()
                    if isinstance(with_original_df, pa.Table) else
                    with_original_df.reset_index(drop=True))
            add = df.drop(columns=[c for c in ("inputs",) if c in df.columns])
            for c in add.columns:
                if c in orig.columns:
                    raise ValueError(f"column collision joining results: {c!r}")
            df = pd.concat([orig, add.reset_index(drop=True)], axis=1)
        return df

    def _maybe_unpack_json(self, df: pd.DataFrame, output_column: str) -> pd.DataFrame:
        """Explode structured-output JSON strings into top-level columns;
        reasoning outputs ({content, reasoning_content}) unpack content one
        level deeper (reference `sdk.py:1278-1320`)."""
        if output_column not in df.columns or len(df) == 0:
            return df
        sample = df[output_column].dropna()
        if len(sample) == 0:
            return df
        first = sample.iloc[0]
        if not (isinstance(first, str) and first.startswith("{")):
            return df
        try:
            parsed = [json.loads(v) if isinstance(v, str) else None
                      for v in df[output_column]]
        except (json.JSONDecodeError, TypeError):
            return df
        if not all(isinstance(p, dict) or p is None for p in parsed):
            return df
        # reasoning double-unpack
        keysets = {frozenset(p) for p in parsed if p is not None}
        if keysets == {frozenset({"content", "reasoning_content"})}:
            inner = []
            for p in parsed:
                c = p.get("content") if p else None
                obj = None
                if isinstance(c, str) and c.startswith("{"):
                    try:
                        obj = json.loads(c)
                    except json.JSONDecodeError:
                        obj = None
                elif isinstance(c, dict):
                    # engine-side reasoning FSM nests the user schema directly
                    obj = c
                if isinstance(obj, dict):
                    row = dict(obj)
                    row.setdefault("reasoning_content",
                                   p.get("reasoning_content") if p else None)
                    inner.append(row)
                else:
                    inner.append({"content": c,
                                  "reasoning_content": p.get("reasoning_content")
                                  if p else None})
            parsed = inner
        all_keys: List[str] = []
        for p in parsed:
            if p:
                for k in p:
                    if k not in all_keys:
                        all_keys.append(k)
        for k in all_keys:
            if k in df.columns:
                raise ValueError(
                    f"cannot unpack structured output: column {k!r} already "
                    f"exists in the results frame")
        for k in all_keys:
            df[k] = [p.get(k) if p else None for p in parsed]
        return df

    # ---- datasets ----

    def create_dataset(self) -> str:
        resp = self.do_request("GET", "/create-dataset")
        return resp.json()["dataset_id"]

    def upload_to_dataset(
        self,
        dataset_id: Optional[str] = None,
        file_paths: Union[str, List[str], None] = None,
        data: Union[pd.DataFrame, pa.Table, None] = None,
    ) -> str:
        """Upload files or a DataFrame to a dataset (creates one if needed)."""
        if dataset_id is None:
            dataset_id = self.create_dataset()
        files: Dict[str, bytes] = {}
        if data is not None:
            pdf = data.to_pandas() if isinstance(data, pa.Table) else data
            import io

            buf = io.BytesIO()
            pdf.to_parquet(buf)
            files[f"upload-{uuid.uuid4().hex[:8]}.parquet"] = buf.getvalue()
        if file_paths is not None:
            if isinstance(file_paths, str):
                file_paths = ([os.path.join(file_paths, f)
                               for f in os.listdir(file_paths)]
                              if os.path.isdir(file_paths) else [file_paths])
            for p in file_paths:
                with open(p, "rb") as f:
                    files[os.path.basename(p)] = f.read()
        resp = self.do_request("POST", "/upload-to-dataset",
                               {"dataset_id": dataset_id}, files=files)
        if resp.status_code != 200:
            raise RuntimeError(f"upload failed: {resp.text}")
        print(to_colored_text(f"✔ Uploaded {len(files)} file(s) to {dataset_id}",
                              "success"))
        return dataset_id

    def list_datasets(self) -> List[dict]:
        return self.do_request("POST", "/list-datasets").json()["datasets"]

    def list_dataset_files(self, dataset_id: str) -> List[str]:
        return self.do_request("POST", "/list-dataset-files",
                               {"dataset_id": dataset_id}).json()["files"]

    def download_from_dataset(
        self, dataset_id: str, file_name: Optional[str] = None,
        output_path: Optional[str] = None,
    ):
        names = [file_name] if file_name else self.list_dataset_files(dataset_id)
        for n in names:
            resp = self.do_request("POST", "/download-from-dataset",
                                   {"dataset_id": dataset_id, "file_name": n})
            out = os.path.join(output_path or ".", n)
            with open(out, "wb") as f:
                f.write(resp.content)
        return names

    # ---- auth / quotas ----

    def try_authentication(self, api_key: Optional[str] = None) -> dict:
        old = self.api_key
        if api_key is not None:
            self.api_key = api_key
        try:
            resp = self.do_request("GET", "/try-authentication")
            return resp.json() if resp.status_code == 200 else {"authenticated": False}
        finally:
            self.api_key = old if api_key is None else self.api_key

    def get_quotas(self) -> List[dict]:
        return self.do_request("GET", "/get-quotas").json()["quotas"]

    # ---- completion ----

    def await_job_completion(
        self,
        job_id: str,
        timeout: int = 7200,
        obtain_results: bool = True,
        with_original_df=None,
        output_column: str = "inference_result",
        unpack_json: bool = True,
        quiet: bool = False,
    ):
        """Poll until terminal status; fetch results on success
        (reference `sdk.py:1643-1718`)."""
        start = time.time()
        while time.time() - start < timeout:
            status = self.get_job_status(job_id)
            if JobStatus.is_terminal(status):
                if status == JobStatus.SUCCEEDED.value:
                    if not obtain_results:
                        return job_id
                    return self.get_job_results(
                        job_id, with_original_df=with_original_df,
                        output_column=output_column, unpack_json=unpack_json)
                if not quiet:
                    reason = self._get_failure_reason(job_id)
                    print(to_colored_text(
                        f"Job {job_id} finished with status {status}: {reason}",
                        "fail"))
                return None
            if self._is_local:
                # event-driven: the in-process service notifies on terminal
                # transitions (no poll-interval latency for p0 rows)
                self._local_transport().service.wait_terminal(
                    job_id, min(30.0, timeout - (time.time() - start)))
            else:
                time.sleep(POLL_INTERVAL)
        raise TimeoutError(f"job {job_id} did not complete within {timeout}s")

    def fancy_tqdm(self, *args, **kwargs):
        """Styled tqdm factory, also exposed as a client method exactly like
        the reference (`/root/reference/sutro/sdk.py:966-1023` duplicates
        `common.fancy_tqdm` onto the class)."""
        from .common import fancy_tqdm as _fancy

        return _fancy(*args, **kwargs)

    # ---- cache management (CLI) ----

    def _clear_job_results_cache(self) -> int:
        d = self._cache_dir()
        n = 0
        for f in os.listdir(d):
            if f.endswith(".parquet"):
                os.remove(os.path.join(d, f))
                n += 1
        return n

    def _show_cache_contents(self) -> List[dict]:
        d = self._cache_dir()
        out = []
        for f in sorted(os.listdir(d)):
            p = os.path.join(d, f)
            out.append({"file": f, "bytes": os.path.getsize(p)})
        return out

    def shutdown(self) -> None:
        if self._local is not None:
            self._local.service.shutdown()
