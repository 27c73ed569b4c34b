"""Step logging: stderr INFO lines + machine-readable JSONL.

Replaces the reference's ``tf.logging`` / ``log_step_count_steps`` cadence
(01:76,105; another-example.py:284,330) and adds the observability SURVEY.md
section 5.5 calls for: loss, lr, step time, samples/sec, peak HBM.
"""

from __future__ import annotations

import json
import os
import sys
import time
from typing import Optional


class StepLogger:
    def __init__(self, model_dir: Optional[str] = None, jsonl_name: str = "steps.jsonl",
                 rank: int = 0):
        self.rank = rank
        self.path = None
        self._f = None
        if model_dir and rank == 0:
            os.makedirs(model_dir, exist_ok=True)
            self.path = os.path.join(model_dir, jsonl_name)
            self._f = open(self.path, "a")

    def log(self, **fields) -> None:
        if self.rank != 0:
            return
        fields.setdefault("time", time.time())
        if self._f:
            self._f.write(json.dumps(fields) + "\n")
            self._f.flush()
        step = fields.get("step")
        loss = fields.get("loss")
        msg = ", ".join(
            f"{k}={v:.6g}" if isinstance(v, float) else f"{k}={v}"
            for k, v in fields.items() if k != "time"
        )
        print(f"INFO:ga_amd: {msg}", file=sys.stderr)

    def close(self):
        if self._f:
            self._f.close()
            self._f = None
