"""Data-parallel service pool: 2 engine worker processes on CPU."""

import os
import time

import pytest

from sutro_amd.interfaces import JobStatus


@pytest.fixture()
def dp_service(sutro_home, monkeypatch):
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "2")
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    yield svc
    svc.shutdown()


def _await(svc, job_id, timeout=240):
    t0 = time.time()
    while time.time() - t0 < timeout:
        st = svc.job_status(job_id)["job_status"][job_id]
        if JobStatus.is_terminal(st):
            return st
        time.sleep(0.1)
    raise TimeoutError


def test_dp_job_sharded_and_ordered(dp_service):
    rows = [f"row number {i}" for i in range(9)]
    out = dp_service.submit_job({
        "model": "qwen-3.5-2b", "inputs": rows,
        "sampling_params": {"max_tokens": 6, "temperature": 0.8},
    })
    job_id = out["results"]
    assert _await(dp_service, job_id) == "SUCCEEDED"
    job = dp_service.get_job(job_id)
    assert isinstance(dp_service.workers["qwen-3.5-2b"].procs, list)
    assert len(dp_service.workers["qwen-3.5-2b"].procs) == 2
    results = dp_service.job_results(job_id)["results"]
    assert len(results["outputs"]) == 9
    assert all(o is not None for o in results["outputs"])
    assert job.input_tokens > 0 and job.output_tokens > 0


def test_dp_seeded_rows_deterministic(dp_service):
    """random_seed_per_input keys the stream on the GLOBAL row index, so the
    result for a row does not depend on which worker ran it."""
    rows = ["same prompt text"] * 4
    payload = {"model": "qwen-3.5-2b", "inputs": rows,
               "random_seed_per_input": True,
               "sampling_params": {"max_tokens": 8, "temperature": 1.0}}
    ids = [dp_service.submit_job(payload)["results"] for _ in range(2)]
    for job_id in ids:
        assert _await(dp_service, job_id) == "SUCCEEDED"
    r0 = dp_service.job_results(ids[0])["results"]["outputs"]
    r1 = dp_service.job_results(ids[1])["results"]["outputs"]
    assert r0 == r1


def test_dp_guided_job(dp_service):
    schema = {"type": "object", "properties": {"label": {"enum": ["A", "B"]}}}
    out = dp_service.submit_job({
        "model": "qwen-3.5-2b", "inputs": ["x", "y", "z"],
        "json_schema": schema})
    job_id = out["results"]
    assert _await(dp_service, job_id) == "SUCCEEDED"
    import json

    results = dp_service.job_results(job_id)["results"]
    for o in results["outputs"]:
        assert json.loads(o)["label"] in ("A", "B")


def test_dp_pool_with_tp_groups(sutro_home, monkeypatch):
    """2 workers forming ONE tp=2 replica over gloo: lockstep admission via
    group broadcast; results complete and ordered."""
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "2")
    monkeypatch.setenv("SUTRO_AMD_TP", "2")
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home + "-tp", device="cpu",
                     engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    try:
        out = svc.submit_job({
            "model": "qwen-3.5-2b", "inputs": ["alpha", "beta", "gamma"],
            "sampling_params": {"max_tokens": 6, "temperature": 0.7},
        })
        job_id = out["results"]
        assert _await(svc, job_id) == "SUCCEEDED"
        w = svc.workers["qwen-3.5-2b"]
        assert w.tp == 2 and w.n_replicas == 1
        results = svc.job_results(job_id)["results"]
        assert len(results["outputs"]) == 3
        assert all(o is not None for o in results["outputs"])
    finally:
        svc.shutdown()


def test_worker_death_fails_job(sutro_home, monkeypatch):
    """Failure detection: a dead engine worker marks in-flight jobs FAILED."""
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "2")
    monkeypatch.delenv("SUTRO_AMD_TP", raising=False)
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home + "-fault", device="cpu",
                     engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    try:
        out = svc.submit_job({
            "model": "qwen-3.5-2b",
            "inputs": [f"long running row {i}" for i in range(40)],
            "sampling_params": {"max_tokens": 500, "temperature": 0.9},
        })
        job_id = out["results"]
        w = svc.workers["qwen-3.5-2b"]
        time.sleep(1.0)
        # simulate a GPU/worker crash
        for p in w.procs:
            p.terminate()
        for p in w.procs:
            p.join(timeout=30)
        w.out_q.put(("worker_error", 0, None, "simulated worker crash"))
        t0 = time.time()
        while time.time() - t0 < 60:
            st = svc.job_status(job_id)["job_status"][job_id]
            if JobStatus.is_terminal(st):
                break
            time.sleep(0.2)
        assert st == "FAILED"
        job = svc.get_job(job_id)
        assert "crash" in job.failure_reason["message"]
    finally:
        svc.shutdown()


def test_engine_restart_resume(sutro_home, monkeypatch):
    """A job mid-flight when the service dies resumes from persisted shards:
    the restarted service re-runs only the missing rows."""
    import json as _json
    import os

    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "1")
    from sutro_amd.service.jobs import JobService

    home = sutro_home + "-resume"
    os.makedirs(os.path.join(home, "jobs"), exist_ok=True)
    os.makedirs(os.path.join(home, "job-results"), exist_ok=True)
    # craft a half-finished persisted job (as the periodic persist writes it)
    jid = "job-resume-test"
    with open(os.path.join(home, "jobs", f"{jid}.json"), "w") as f:
        _json.dump({"job_id": jid, "model": "qwen-3.5-2b", "status": "RUNNING",
                    "num_rows": 4, "input_tokens": 10, "output_tokens": 5,
                    "datetime_created": "2026-01-01T00:00:00"}, f)
    with open(os.path.join(home, "job-results", f"{jid}.json"), "w") as f:
        _json.dump({"outputs": ["done-a", "done-b", None, None],
                    "embeddings": [None] * 4,
                    "cumulative_logprobs": [-1.0, -1.0, None, None],
                    "confidence_score": [0.5, 0.5, None, None],
                    "inputs": ["r0", "r1", "r2", "r3"],
                    "sampling_params": {"max_tokens": 6},
                    "json_schema": None, "system_prompt": None}, f)
    svc = JobService(home=home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    try:
        assert _await(svc, jid) == "SUCCEEDED"
        job = svc.get_job(jid)
        # previously completed rows untouched; missing rows filled
        assert job.outputs[0] == "done-a" and job.outputs[1] == "done-b"
        assert job.outputs[2] is not None and job.outputs[3] is not None
    finally:
        svc.shutdown()


def test_dp_embedding_job(dp_service):
    out = dp_service.submit_job({
        "model": "qwen-3-embedding-0.6b",
        "inputs": ["vector me", "and me", "me too"]})
    job_id = out["results"]
    assert _await(dp_service, job_id) == "SUCCEEDED"
    results = dp_service.job_results(job_id)["results"]
    assert len(results["outputs"]) == 3
    assert all(isinstance(v, list) and len(v) > 0 for v in results["outputs"])


def test_dp_pool_with_async_decode(sutro_home, monkeypatch):
    """The multi-process DP pool works with async_decode engines (lagged
    finishes stream back through the collector like any other)."""
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "2")
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 128,
                                    "max_model_len": 1024,
                                    "async_decode": True})
    try:
        rows = [f"async row {i}" for i in range(6)]
        out = svc.submit_job({
            "model": "qwen-3.5-2b", "inputs": rows,
            "random_seed_per_input": True,
            "sampling_params": {"max_tokens": 6, "temperature": 0.9},
        })
        job_id = out["results"]
        assert _await(svc, job_id) == "SUCCEEDED"
        results = svc.job_results(job_id)["results"]
        assert len(results["outputs"]) == 6
        assert all(o is not None for o in results["outputs"])
    finally:
        svc.shutdown()


def test_restart_resume_embedding_job(sutro_home, monkeypatch):
    """Service restart mid-embedding-job resumes by the EMBEDDINGS column
    (outputs stay None for embedding rows by design)."""
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "1")
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 128,
                                    "max_model_len": 1024})
    try:
        out = svc.submit_job({
            "model": "qwen-3-embedding-0.6b",
            "inputs": [f"doc {i}" for i in range(4)],
        })
        job_id = out["results"]
        assert _await(svc, job_id) == "SUCCEEDED"
        job = svc.get_job(job_id)
        # persist results (normally every 256 rows / at completion)
        svc.persist_job(job, with_results=True)
    finally:
        svc.shutdown()

    # a fresh service on the same home must see the job complete, not re-run
    svc2 = JobService(home=sutro_home, device="cpu",
                      engine_kwargs={"num_kv_blocks": 128,
                                     "max_model_len": 1024})
    try:
        job2 = svc2.get_job(job_id)
        assert job2.status.value == "SUCCEEDED"
        res = svc2.job_results(job_id)["results"]
        assert len(res["outputs"]) == 4
    finally:
        svc2.shutdown()


def test_dp8_soak_sharded_and_ordered(sutro_home, monkeypatch):
    """SCALE-readiness soak: 8 engine worker processes (the full-node DP
    shape the driver benches sight-unseen), 64 rows, ordered results,
    per-row seeds deterministic."""
    monkeypatch.setenv("SUTRO_AMD_NUM_WORKERS", "8")
    from sutro_amd.service.jobs import JobService

    svc = JobService(home=sutro_home, device="cpu",
                     engine_kwargs={"num_kv_blocks": 64,
                                    "max_model_len": 256})
    try:
        rows = [f"soak row {i} " + "x" * (i % 17) for i in range(64)]
        out = svc.submit_job({
            "model": "qwen-3.5-2b", "inputs": rows,
            "random_seed_per_input": True,
            "sampling_params": {"max_tokens": 5, "temperature": 0.9},
        })
        job_id = out["results"]
        assert _await(svc, job_id, timeout=600) == "SUCCEEDED"
        assert len(svc.workers["qwen-3.5-2b"].procs) == 8
        res = svc.job_results(job_id, include_inputs=True)["results"]
        assert len(res["outputs"]) == 64
        assert all(o is not None for o in res["outputs"])
        assert res["inputs"] == rows  # input-ordered merge across 8 shards
        # determinism: resubmit, same seeded outputs
        out2 = svc.submit_job({
            "model": "qwen-3.5-2b", "inputs": rows,
            "random_seed_per_input": True,
            "sampling_params": {"max_tokens": 5, "temperature": 0.9},
        })
        jid2 = out2["results"]
        assert _await(svc, jid2, timeout=600) == "SUCCEEDED"
        res2 = svc.job_results(jid2)["results"]
        assert res2["outputs"] == res["outputs"]
    finally:
        svc.shutdown()
