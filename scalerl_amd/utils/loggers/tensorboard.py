"""TensorBoard backend (activates only when the tensorboard package is
installed; parity with ``scalerl/utils/logger/tensorboard.py``)."""

from __future__ import annotations

from typing import Any, Dict

from torch.utils.tensorboard import SummaryWriter  # noqa: F401 — ImportError gates backend

from .base import BaseLogger


class TensorboardLogger(BaseLogger):
    def __init__(self, log_dir: str, **kw):
        super().__init__(**kw)
        self.writer = SummaryWriter(log_dir)

    def write(self, namespace: str, step: int, data: Dict[str, Any]) -> None:
        for k, v in data.items():
            try:
                self.writer.add_scalar(f"{namespace}/{k}", float(v), step)
            except (TypeError, ValueError):
                pass
        self.writer.flush()

    def restore_data(self) -> Dict[str, int]:
        # Event-file re-read requires the tensorboard EventAccumulator.
        try:
            from tensorboard.backend.event_processing.event_accumulator import (
                EventAccumulator)
            acc = EventAccumulator(self.writer.log_dir)
            acc.Reload()
            out = {}
            for k in ("epoch", "env_step", "gradient_step"):
                tag = f"save/{k}"
                out[k] = int(acc.Scalars(tag)[-1].value) if tag in acc.Tags().get(
                    "scalars", []) else 0
            return out
        except Exception:
            return {"epoch": 0, "env_step": 0, "gradient_step": 0}

    def close(self) -> None:
        self.writer.close()
