"""Functions: named, pre-configured inference endpoints.

The reference client calls a serving base URL (`POST /functions/run`,
`/root/reference/sutro/sdk.py:539-615`) and submits batch jobs whose `model`
field is a Function name (`sdk.py:710-721`). Locally a Function is a stored
{model, system_prompt, output_schema} record; `run` executes one synchronous
generation and returns the reference's response shape
{response, confidence, predictions, run_id, usage}.
"""

from __future__ import annotations

import json
import math
import os
import uuid
from typing import Any, Dict, Optional


class FunctionStore:
    def __init__(self, home: str):
        self.root = os.path.join(home, "functions")
        os.makedirs(self.root, exist_ok=True)

    def _path(self, name: str) -> str:
        safe = "".join(c for c in name if c.isalnum() or c in "-_.")
        return os.path.join(self.root, f"{safe}.json")

    def create(self, payload: Dict[str, Any]) -> Dict[str, Any]:
        name = payload["name"]
        record = {
            "name": name,
            "model": payload["model"],
            "system_prompt": payload.get("system_prompt"),
            "output_schema": payload.get("output_schema"),
        }
        with open(self._path(name), "w") as f:
            json.dump(record, f, indent=2)
        return {"name": name, "created": True}

    def get(self, name: str) -> Optional[Dict[str, Any]]:
        try:
            with open(self._path(name)) as f:
                return json.load(f)
        except FileNotFoundError:
            return None

    def run(self, job_service, name: str, input_data: Any) -> Dict[str, Any]:
        fn = self.get(name)
        if fn is None:
            raise KeyError(f"unknown function {name!r}")
        worker = job_service._get_local_worker(fn["model"])
        eng = worker.engine
        from ..engine.request import SamplingParams

        tok = eng.tokenizer
        text = input_data if isinstance(input_data, str) else json.dumps(input_data)
        ids = tok.render_prompt(text, fn.get("system_prompt"))
        fsm_id = None
        if fn.get("output_schema"):
            key = json.dumps(fn["output_schema"], sort_keys=True)
            cache = getattr(worker, "_fn_fsm_cache", None)
            if cache is None:
                cache = worker._fn_fsm_cache = {}
            if key not in cache:
                cache[key] = eng.register_fsm(fn["output_schema"])
            fsm_id = cache[key]
        # add_request only touches python queues (GIL-atomic appends); the
        # worker thread owns all KV/tensor state and picks the request up.
        req = eng.add_request(ids, SamplingParams(max_tokens=512, temperature=0.7),
                              fsm_id=fsm_id, priority=0)
        worker._wake.set()
        import time

        t0 = time.time()
        while not req.finished and time.time() - t0 < 600:
            time.sleep(0.01)
        out_text = eng.output_text(req)
        n = max(1, len(req.output_token_ids))
        confidence = float(min(1.0, max(0.0, math.exp(req.cumulative_logprob / n))))
        return {
            "response": out_text,
            "confidence": confidence,
            "predictions": [{"label": out_text[:64], "confidence": confidence}],
            "run_id": f"run-{uuid.uuid4().hex[:12]}",
            "usage": {
                "input_tokens": len(req.prompt_token_ids),
                "output_tokens": len(req.output_token_ids),
            },
        }
