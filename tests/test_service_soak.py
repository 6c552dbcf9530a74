"""Concurrency soak of the job service: several jobs of different kinds
in flight at once (generative, schema-guided, embedding, cancelled) must all
reach consistent terminal states with input-ordered results."""

import pandas as pd

from sutro_amd.interfaces import JobStatus


def test_concurrent_mixed_jobs(local_client):
    so = local_client

    df_gen = pd.DataFrame({"text": [f"row {i}" for i in range(12)]})
    df_cls = pd.DataFrame({"text": ["good", "bad", "meh", "fine"]})
    df_emb = pd.DataFrame({"text": [f"doc {i}" for i in range(6)]})
    df_cancel = pd.DataFrame({"text": [f"slow {i}" for i in range(16)]})

    schema = {"type": "object",
              "properties": {"label": {"enum": ["pos", "neg"]}}}

    j_gen = so.infer(df_gen, column="text", model="qwen-3.5-2b",
                     sampling_params={"max_tokens": 8}, stay_attached=False)
    j_cls = so.infer(df_cls, column="text", model="qwen-3.5-2b",
                     output_schema=schema, stay_attached=False)
    j_emb = so.infer(df_emb, column="text", model="qwen-3-embedding-0.6b",
                     stay_attached=False)
    j_cancel = so.infer(df_cancel, column="text", model="qwen-3.5-2b",
                        sampling_params={"max_tokens": 512},
                        stay_attached=False)
    so.cancel_job(j_cancel)

    for j in (j_gen, j_cls, j_emb):
        so.await_job_completion(j, timeout=120, quiet=True)
        assert so.get_job_status(j) == JobStatus.SUCCEEDED.value

    # cancelled job terminal (either cancelled before finishing, or — tiny
    # rows — already succeeded by the time the cancel landed)
    st = so.get_job_status(j_cancel)
    assert st in (JobStatus.CANCELLED.value, JobStatus.CANCELLING.value,
                  JobStatus.SUCCEEDED.value)

    r_gen = so.get_job_results(j_gen, include_inputs=True)
    assert len(r_gen) == 12
    assert list(r_gen["inputs"]) == list(df_gen["text"])  # input order

    r_cls = so.get_job_results(j_cls)
    assert len(r_cls) == 4
    assert set(r_cls["label"]) <= {"pos", "neg"}  # FSM-enforced + unpacked

    r_emb = so.get_job_results(j_emb)
    assert len(r_emb) == 6
    col = "inference_result" if "inference_result" in r_emb else "outputs"
    emb0 = r_emb[col].iloc[0]
    assert hasattr(emb0, "__len__") and len(emb0) > 4


def test_single_proc_service_with_async_decode(sutro_home):
    """In-process EngineWorker with async_decode: lagged finishes flow
    through the worker loop and results stay input-ordered."""
    from sutro_amd.sdk import Sutro

    so = Sutro(home=sutro_home, device="cpu",
               engine_kwargs={"num_kv_blocks": 256, "max_model_len": 2048,
                              "async_decode": True})
    try:
        rows = [f"async svc row {i}" for i in range(8)]
        job_id = so.infer(rows, model="qwen-3.5-2b", stay_attached=False,
                          sampling_params={"max_tokens": 6,
                                           "temperature": 0.8})
        df = so.await_job_completion(job_id, timeout=120)
        assert len(df) == 8
        assert df["inference_result"].notna().all()
    finally:
        so.shutdown()
