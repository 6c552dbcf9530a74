"""HTTP service: the api.sutro.sh endpoint contract served by the local engine.

Every route in SURVEY.md §2.3's endpoint table (reconstructed from
`/root/reference/sutro/sdk.py`) is served here over FastAPI, so the same SDK
can point `base_url` at a remote box running `sutro serve`:
job submission, status/details/list, line-JSON progress streaming, results,
cancel, auth, quotas, datasets, Functions.
"""

from __future__ import annotations

import json
from typing import Dict, Optional

from fastapi import FastAPI, Header, HTTPException, Request, Response
from fastapi.responses import StreamingResponse


def create_app(home: Optional[str] = None, device: str = "auto",
               api_keys: Optional[set] = None,
               engine_kwargs: Optional[dict] = None) -> FastAPI:
    from .datasets import DatasetStore
    from .functions import FunctionStore
    from .jobs import JobService

    service = JobService(home=home, device=device, engine_kwargs=engine_kwargs)
    datasets = DatasetStore(service.home)
    functions = FunctionStore(service.home)
    app = FastAPI(title="sutro-amd", version="0.1.0")
    app.state.service = service

    def check_auth(authorization: Optional[str]) -> None:
        if api_keys is None:
            return
        if not authorization or not authorization.startswith("Key "):
            raise HTTPException(401, "missing Authorization: Key <api_key>")
        if authorization[4:] not in api_keys:
            raise HTTPException(403, "invalid API key")

    @app.post("/batch-inference")
    async def batch_inference(request: Request,
                              authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        try:
            return service.submit_job(payload)
        except (ValueError, KeyError) as e:
            raise HTTPException(400, str(e))

    @app.get("/stream-job-progress/{job_id}")
    def stream_progress(job_id: str,
                        authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        try:
            service.get_job(job_id)
        except KeyError:
            raise HTTPException(404, f"unknown job {job_id}")

        def gen():
            for update in service.stream_progress(job_id):
                yield json.dumps(update) + "\n"

        return StreamingResponse(gen(), media_type="application/jsonl")

    @app.get("/job-status/{job_id}")
    def job_status(job_id: str, authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        try:
            return service.job_status(job_id)
        except KeyError:
            raise HTTPException(404, f"unknown job {job_id}")

    @app.get("/jobs/{job_id}")
    def job_details(job_id: str, authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        try:
            return service.job_details(job_id)
        except KeyError:
            raise HTTPException(404, f"unknown job {job_id}")

    @app.post("/job-results")
    async def job_results(request: Request,
                          authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        try:
            return service.job_results(
                payload["job_id"],
                include_inputs=payload.get("include_inputs", False),
                include_cumulative_logprobs=payload.get(
                    "include_cumulative_logprobs", False))
        except KeyError as e:
            raise HTTPException(404, str(e))
        except RuntimeError as e:
            raise HTTPException(409, str(e))

    @app.get("/list-jobs")
    def list_jobs(authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        return service.list_jobs()

    @app.get("/job-cancel/{job_id}")
    def job_cancel(job_id: str, authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        try:
            return service.cancel_job(job_id)
        except KeyError:
            raise HTTPException(404, f"unknown job {job_id}")

    @app.get("/try-authentication")
    def try_auth(authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        return {"authenticated": True}

    @app.get("/get-quotas")
    def get_quotas(authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        return service.get_quotas()

    @app.get("/create-dataset")
    def create_dataset(authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        return datasets.create()

    @app.post("/upload-to-dataset")
    async def upload_to_dataset(request: Request,
                                authorization: Optional[str] = Header(None)):
        """Files travel as {"dataset_id": ..., "files": {name: base64}}
        (this environment has no python-multipart; both SDK transports use
        the same JSON shape)."""
        check_auth(authorization)
        import base64

        payload = await request.json()
        dataset_id = payload.get("dataset_id")
        if not dataset_id:
            raise HTTPException(400, "dataset_id required")
        n = 0
        for name, b64 in (payload.get("files") or {}).items():
            datasets.upload(dataset_id, name, base64.b64decode(b64))
            n += 1
        return {"uploaded": n}

    @app.post("/list-datasets")
    def list_datasets(authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        return datasets.list_datasets()

    @app.post("/list-dataset-files")
    async def list_dataset_files(request: Request,
                                 authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        try:
            return datasets.list_files(payload["dataset_id"])
        except KeyError as e:
            raise HTTPException(404, str(e))

    @app.post("/download-from-dataset")
    async def download_from_dataset(request: Request,
                                    authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        try:
            data = datasets.download(payload["dataset_id"], payload["file_name"])
        except (KeyError, FileNotFoundError) as e:
            raise HTTPException(404, str(e))
        return Response(content=data, media_type="application/octet-stream")

    @app.post("/functions/run")
    async def functions_run(request: Request,
                            authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        try:
            return functions.run(service, payload["name"],
                                 payload.get("input_data"))
        except KeyError as e:
            raise HTTPException(404, str(e))

    @app.post("/functions/create")
    async def functions_create(request: Request,
                               authorization: Optional[str] = Header(None)):
        check_auth(authorization)
        payload = await request.json()
        return functions.create(payload)

    @app.get("/metrics")
    def metrics():
        """Prometheus text exposition of service/job counters."""
        from prometheus_client import (CONTENT_TYPE_LATEST, CollectorRegistry,
                                       Gauge, generate_latest)

        reg = CollectorRegistry()
        by_status: Dict[str, int] = {}
        in_tok = out_tok = rows_done = rows_total = 0
        for job in service.jobs.values():
            by_status[job.status.value] = by_status.get(job.status.value, 0) + 1
            in_tok += job.input_tokens
            out_tok += job.output_tokens
            rows_done += job.completed_rows
            rows_total += job.num_rows
        g = Gauge("sutro_jobs", "jobs by status", ["status"], registry=reg)
        for st, n in by_status.items():
            g.labels(status=st).set(n)
        Gauge("sutro_input_tokens_total", "input tokens across jobs",
              registry=reg).set(in_tok)
        Gauge("sutro_output_tokens_total", "output tokens across jobs",
              registry=reg).set(out_tok)
        Gauge("sutro_rows_completed_total", "completed rows",
              registry=reg).set(rows_done)
        Gauge("sutro_rows_total", "submitted rows", registry=reg).set(rows_total)
        Gauge("sutro_engine_workers", "live engine workers",
              registry=reg).set(len(service.workers))
        return Response(content=generate_latest(reg),
                        media_type=CONTENT_TYPE_LATEST)

    return app


def run_server(host: str = "127.0.0.1", port: int = 8000, device: str = "auto",
               home: Optional[str] = None, api_keys: Optional[set] = None):
    import uvicorn

    uvicorn.run(create_app(home=home, device=device, api_keys=api_keys),
                host=host, port=port)
