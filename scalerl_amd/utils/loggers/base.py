"""Metrics logger layer (parity with ``scalerl/utils/logger/base.py:46-90``:
interval-gated ``train/ test/ update/`` scalar namespaces and
save/restore of resume metadata).

The default backend writes JSONL (this image ships no tensorboard/wandb);
TensorBoard and Weights & Biases adapters activate when their packages are
importable, keeping the reference's logger surface.
"""

from __future__ import annotations

import abc
from typing import Any, Dict


class BaseLogger(abc.ABC):
    """Interval-gated scalar namespaces: train/ test/ update/."""

    def __init__(self, train_interval: int = 1, test_interval: int = 1,
                 update_interval: int = 1):
        self.train_interval = train_interval
        self.test_interval = test_interval
        self.update_interval = update_interval
        self._last = {"train": None, "test": None, "update": None}

    @abc.abstractmethod
    def write(self, namespace: str, step: int, data: Dict[str, Any]) -> None:
        ...

    def _gated(self, ns: str, step: int, data: Dict[str, Any], interval: int):
        last = self._last[ns]
        if last is None or step - last >= interval:
            self.write(ns, step, data)
            self._last[ns] = step

    def log_train_data(self, data: Dict[str, Any], step: int) -> None:
        self._gated("train", step, data, self.train_interval)

    def log_test_data(self, data: Dict[str, Any], step: int) -> None:
        self._gated("test", step, data, self.test_interval)

    def log_update_data(self, data: Dict[str, Any], step: int) -> None:
        self._gated("update", step, data, self.update_interval)

    def save_data(self, epoch: int, env_step: int, gradient_step: int) -> None:
        """Persist resume metadata (reference: logger/tensorboard.py:47-82)."""
        self.write("save", env_step,
                   {"epoch": epoch, "env_step": env_step,
                    "gradient_step": gradient_step})

    def restore_data(self) -> Dict[str, int]:
        return {"epoch": 0, "env_step": 0, "gradient_step": 0}

    def close(self) -> None:
        pass


def make_logger(backend: str, log_dir: str, **kw) -> "BaseLogger":
    from .jsonl import JsonlLogger
    if backend in ("jsonl", "", None):
        return JsonlLogger(log_dir, **kw)
    if backend == "tensorboard":
        try:
            from .tensorboard import TensorboardLogger
            return TensorboardLogger(log_dir, **kw)
        except ImportError:
            return JsonlLogger(log_dir, **kw)
    if backend == "wandb":
        try:
            from .wandb import WandbLogger
            return WandbLogger(log_dir, **kw)
        except ImportError:
            return JsonlLogger(log_dir, **kw)
    raise ValueError(f"unknown logger backend: {backend}")
