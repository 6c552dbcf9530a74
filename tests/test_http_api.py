"""HTTP service contract tests (FastAPI TestClient, tiny CPU engines)."""

import json
import time

import pandas as pd
import pytest
from fastapi.testclient import TestClient

from sutro_amd.interfaces import JobStatus


@pytest.fixture()
def client(sutro_home):
    from sutro_amd.service.http_api import create_app

    app = create_app(home=sutro_home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    with TestClient(app) as c:
        yield c
    app.state.service.shutdown()


def _submit(client, payload):
    r = client.post("/batch-inference", json=payload)
    assert r.status_code == 200, r.text
    return r.json()["results"]


def _await(client, job_id, timeout=120):
    t0 = time.time()
    while time.time() - t0 < timeout:
        st = client.get(f"/job-status/{job_id}").json()["job_status"][job_id]
        if JobStatus.is_terminal(st):
            return st
        time.sleep(0.05)
    raise TimeoutError


def test_submit_status_results(client):
    job_id = _submit(client, {
        "model": "qwen-3.5-2b", "inputs": ["hello", "world"],
        "job_priority": 0, "sampling_params": {"max_tokens": 6},
    })
    assert _await(client, job_id) == "SUCCEEDED"
    r = client.post("/job-results", json={"job_id": job_id,
                                          "include_inputs": True})
    results = r.json()["results"]
    assert len(results["outputs"]) == 2
    assert results["inputs"] == ["hello", "world"]


def test_job_details_and_list(client):
    job_id = _submit(client, {"model": "qwen-3.5-2b", "inputs": ["x"],
                              "sampling_params": {"max_tokens": 4}})
    _await(client, job_id)
    job = client.get(f"/jobs/{job_id}").json()["job"]
    assert job["num_rows"] == 1 and job["input_tokens"] > 0
    jobs = client.get("/list-jobs").json()["jobs"]
    assert any(j["job_id"] == job_id for j in jobs)


def test_progress_stream_lines(client):
    job_id = _submit(client, {"model": "qwen-3.5-2b", "inputs": ["a", "b"],
                              "sampling_params": {"max_tokens": 4}})
    with client.stream("GET", f"/stream-job-progress/{job_id}") as resp:
        kinds = set()
        for line in resp.iter_lines():
            if not line:
                continue
            u = json.loads(line)
            kinds.add(u["update_type"])
    assert {"progress", "tokens"} <= kinds


def test_unknown_job_404(client):
    assert client.get("/job-status/job-nope").status_code == 404
    assert client.get("/jobs/job-nope").status_code == 404


def test_results_before_success_409(client):
    job_id = _submit(client, {"model": "qwen-3.5-2b",
                              "inputs": [f"r{i}" for i in range(20)],
                              "sampling_params": {"max_tokens": 64}})
    r = client.post("/job-results", json={"job_id": job_id})
    assert r.status_code in (409, 200)  # 200 only if it finished very fast
    client.get(f"/job-cancel/{job_id}")


def test_quotas_and_auth(client):
    q = client.get("/get-quotas").json()["quotas"]
    assert len(q) >= 2
    assert client.get("/try-authentication").json()["authenticated"] is True


def test_auth_enforced(sutro_home):
    from sutro_amd.service.http_api import create_app

    app = create_app(home=sutro_home + "-auth", device="cpu",
                     api_keys={"secret-key"},
                     engine_kwargs={"num_kv_blocks": 64})
    with TestClient(app) as c:
        assert c.get("/list-jobs").status_code == 401
        assert c.get("/list-jobs",
                     headers={"Authorization": "Key wrong"}).status_code == 403
        assert c.get("/list-jobs",
                     headers={"Authorization": "Key secret-key"}).status_code == 200
    app.state.service.shutdown()


def test_datasets_over_http(client, tmp_path):
    import base64

    ds = client.get("/create-dataset").json()["dataset_id"]
    p = tmp_path / "d.csv"
    pd.DataFrame({"t": ["u", "v"]}).to_csv(p, index=False)
    r = client.post("/upload-to-dataset", json={
        "dataset_id": ds,
        "files": {"d.csv": base64.b64encode(p.read_bytes()).decode()}})
    assert r.status_code == 200
    files = client.post("/list-dataset-files", json={"dataset_id": ds}).json()["files"]
    assert files == ["d.csv"]
    raw = client.post("/download-from-dataset",
                      json={"dataset_id": ds, "file_name": "d.csv"}).content
    assert b"u" in raw
    # dataset-backed job over HTTP
    job_id = _submit(client, {"model": "qwen-3.5-2b", "inputs": ds,
                              "column_name": "t",
                              "sampling_params": {"max_tokens": 4}})
    assert _await(client, job_id) == "SUCCEEDED"


def test_functions_over_http(client):
    client.post("/functions/create", json={"name": "f1", "model": "qwen-3.5-2b"})
    out = client.post("/functions/run",
                      json={"name": "f1", "input_data": "hi"}).json()
    assert {"response", "confidence", "run_id", "usage"} <= set(out)


def test_sdk_against_http_server(client, sutro_home, monkeypatch):
    """The SDK's HTTP transport against the real app (requests patched to the
    TestClient)."""
    from sutro_amd.sdk import Sutro

    so = Sutro(api_key="k", base_url="http://testserver", home=sutro_home)

    class _Shim:
        def get(self, url, headers=None, stream=False, timeout=None):
            return client.get(url.replace("http://testserver", ""),
                              headers=headers)

        def post(self, url, headers=None, json=None, data=None, files=None,
                 stream=False, timeout=None):
            path = url.replace("http://testserver", "")
            if files:
                return client.post(path, data=data, files=files, headers=headers)
            return client.post(path, json=json, headers=headers)

        RequestException = Exception

    import sutro_amd.sdk as sdk_mod
    shim = _Shim()
    monkeypatch.setattr("requests.get", shim.get)
    monkeypatch.setattr("requests.post", shim.post)
    job_id = so.infer(["one", "two"], model="qwen-3.5-2b", stay_attached=False,
                      sampling_params={"max_tokens": 4})
    df = so.await_job_completion(job_id)
    assert len(df) == 2


def test_metrics_endpoint(client):
    job_id = _submit(client, {"model": "qwen-3.5-2b", "inputs": ["m1", "m2"],
                              "sampling_params": {"max_tokens": 4}})
    _await(client, job_id)
    r = client.get("/metrics")
    assert r.status_code == 200
    body = r.text
    assert "sutro_jobs" in body
    assert "sutro_output_tokens_total" in body
    assert 'status="SUCCEEDED"' in body
