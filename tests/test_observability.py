"""Tracing: per-row deterministic-UUID batch traces (reference
`observability.py` ported to a local JSONL sink)."""

import json
import os


from sutro_amd.observability import _row_run_id, tracing_enabled


def test_row_run_ids_deterministic():
    a = _row_run_id("job-x", 0)
    b = _row_run_id("job-x", 0)
    c = _row_run_id("job-x", 1)
    d = _row_run_id("job-y", 0)
    assert a == b
    assert len({a, c, d}) == 3


def test_tracing_disabled_by_default(monkeypatch):
    monkeypatch.delenv("SUTRO_TRACING", raising=False)
    monkeypatch.delenv("LANGSMITH_TRACING", raising=False)
    assert not tracing_enabled()


def test_batch_traces_created_and_completed(sutro_home, monkeypatch):
    monkeypatch.setenv("SUTRO_TRACING", "true")
    from sutro_amd.sdk import Sutro

    client = Sutro(home=sutro_home, device="cpu",
                   engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    try:
        job_id = client.infer(["trace row a", "trace row b"],
                              model="qwen-3.5-2b", stay_attached=False,
                              sampling_params={"max_tokens": 4})
        client._create_batch_traces(job_id, ["trace row a", "trace row b"])
        assert client._has_open_batch_traces(job_id)
        client.await_job_completion(job_id)  # completes traces on fetch
        assert not client._has_open_batch_traces(job_id)
        path = os.path.join(sutro_home, "traces.jsonl")
        events = [json.loads(l) for l in open(path)]
        kinds = {e["event"] for e in events if e.get("job_id") == job_id}
        assert kinds == {"create", "complete"}
        completes = [e for e in events if e.get("event") == "complete"
                     and e["job_id"] == job_id]
        assert len(completes) == 2
        assert all("usage" in e for e in completes)
        # deterministic ids line up between create and complete
        created = {e["run_id"] for e in events if e["event"] == "create"
                   and e["job_id"] == job_id}
        assert {e["run_id"] for e in completes} <= created
    finally:
        client.shutdown()


def test_online_run_traced(sutro_home, monkeypatch):
    monkeypatch.setenv("SUTRO_TRACING", "true")
    from sutro_amd.sdk import Sutro

    client = Sutro(home=sutro_home, device="cpu",
                   engine_kwargs={"num_kv_blocks": 128, "max_model_len": 1024})
    try:
        client.create_function("traced-fn", model="qwen-3.5-2b")
        client.run_function("traced-fn", "hello")
        path = os.path.join(sutro_home, "traces.jsonl")
        events = [json.loads(l) for l in open(path)]
        online = [e for e in events if e["event"] == "online_run"]
        assert online and online[0]["name"] == "traced-fn"
        assert "latency_s" in online[0]
    finally:
        client.shutdown()
