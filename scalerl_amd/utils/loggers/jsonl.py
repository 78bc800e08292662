"""JSONL metrics backend — the always-available logger.

One line per write: ``{"ns": ..., "step": ..., "ts": ..., **scalars}``.
Resume metadata is re-read from the tail of the file (parity with the
reference's TB-event-scalar resume mechanism, logger/tensorboard.py:47-82).
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict

from .base import BaseLogger


class JsonlLogger(BaseLogger):
    def __init__(self, log_dir: str, filename: str = "metrics.jsonl", **kw):
        super().__init__(**kw)
        os.makedirs(log_dir, exist_ok=True)
        self.path = os.path.join(log_dir, filename)
        self._fh = open(self.path, "a", buffering=1)

    def write(self, namespace: str, step: int, data: Dict[str, Any]) -> None:
        rec = {"ns": namespace, "step": int(step), "ts": time.time()}
        for k, v in data.items():
            try:
                rec[k] = float(v)
            except (TypeError, ValueError):
                rec[k] = str(v)
        self._fh.write(json.dumps(rec) + "\n")

    def restore_data(self) -> Dict[str, int]:
        out = {"epoch": 0, "env_step": 0, "gradient_step": 0}
        if not os.path.exists(self.path):
            return out
        with open(self.path) as fh:
            for line in fh:
                try:
                    rec = json.loads(line)
                except json.JSONDecodeError:
                    continue
                if rec.get("ns") == "save":
                    for k in out:
                        if k in rec:
                            out[k] = int(rec[k])
        return out

    def close(self) -> None:
        self._fh.close()
