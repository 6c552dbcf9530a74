"""Randomized SERVICE soak (CPU): concurrent jobs across models/priorities,
schemas (incl. patterns/bounds), embeddings, cancellation, results
integrity. `python tools/soak_service.py [--minutes 5]`."""

from __future__ import annotations

import argparse
import json
import random
import tempfile
import time

SCHEMAS = [
    None,
    {"type": "object", "properties": {
        "label": {"enum": ["a", "b"]},
        "score": {"type": "integer", "minimum": 0, "maximum": 10**6}}},
    {"type": "object", "properties": {
        "sku": {"type": "string", "pattern": r"^[A-Z]{2}-\d{3}$"},
        "w": {"type": "number", "minimum": 0, "maximum": 1}}},
]


def run_epoch(seed: int) -> int:
    from sutro_amd.sdk import Sutro

    rng = random.Random(seed)
    home = tempfile.mkdtemp(prefix="sutro-soak-")
    client = Sutro(home=home, device="cpu",
                   engine_kwargs={"num_kv_blocks": 128,
                                  "max_model_len": 512})
    jobs = []
    for j in range(rng.randint(2, 5)):
        model = rng.choice(["qwen-3.5-2b", "qwen-3.5-2b-thinking",
                            "qwen-3-embedding-0.6b"])
        rows = [f"row {j}.{i} " + "y" * rng.randint(0, 40)
                for i in range(rng.randint(2, 10))]
        if "embedding" in model:
            df = client.embed(rows, model=model)
            assert len(df) == len(rows)
            continue
        schema = rng.choice(SCHEMAS)
        jid = client.infer(
            rows, model=model, stay_attached=False,
            job_priority=rng.randint(0, 1), output_schema=schema,
            random_seed_per_input=rng.random() < 0.5,
            sampling_params={"max_tokens": rng.randint(4, 64)
                             if schema is None else rng.randint(256, 512),
                             "temperature": rng.choice([0.0, 0.9])})
        jobs.append((jid, rows, schema, model))
    # cancel one occasionally
    if jobs and rng.random() < 0.3:
        client.cancel_job(jobs[-1][0])
    done = 0
    for jid, rows, schema, model in jobs:
        res = client.await_job_completion(jid, timeout=600, quiet=True)
        if res is None:
            st = client.get_job_status(jid)
            assert st in ("CANCELLED", "CANCELLING", "FAILED"), st
            continue
        assert len(res) == len(rows)
        if schema is not None and "thinking" not in model:
            for v in res["inference_result"]:
                obj = json.loads(v)
                if "score" in obj:
                    assert 0 <= obj["score"] <= 10**6
                if "sku" in obj:
                    import re

                    assert re.fullmatch(r"[A-Z]{2}-\d{3}", obj["sku"])
                if "w" in obj:
                    assert 0 <= obj["w"] <= 1
        done += 1
    client.shutdown()
    return done


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=5.0)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    t0 = time.time()
    epochs = 0
    master = random.Random(args.seed)
    while time.time() - t0 < args.minutes * 60:
        run_epoch(master.randrange(1 << 30))
        epochs += 1
        print(f"[{time.time()-t0:6.1f}s] epoch {epochs} ok", flush=True)
    print(f"SERVICE SOAK PASS: {epochs} epochs")


if __name__ == "__main__":
    main()
