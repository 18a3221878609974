"""AIM metrics driver: receives scheduler metric posts on POST /status.

Parity with /root/reference/drivers/aim-driver/main.py (FastAPI -> aim.Run):
the `aim` package is not installed offline, so metrics are appended to a
JSONL run log (and to aim when importable). Start with:
    uvicorn drivers.aim_driver:app --port 43800
"""

from __future__ import annotations

import json
import os
import time

from fastapi import FastAPI
from pydantic import BaseModel

app = FastAPI()
LOG = os.environ.get("HYPHA_AIM_LOG", "/tmp/hypha-aim-run.jsonl")

try:  # pragma: no cover - aim not installed in this environment
    import aim

    _run = aim.Run()
except Exception:
    _run = None


class Status(BaseModel):
    worker_id: str
    round: int
    metric_name: str
    value: float


@app.post("/status")
def status(s: Status):
    if _run is not None:  # pragma: no cover
        _run.track(name=f"{s.worker_id}_{s.metric_name}", epoch=s.round, value=s.value)
    with open(LOG, "a") as f:
        f.write(json.dumps({**s.dict(), "ts": time.time()}) + "\n")
    return {"ok": True}
