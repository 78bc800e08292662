"""Weights & Biases backend (activates only when wandb is installed;
parity with ``scalerl/utils/logger/wandb.py:19-160``)."""

from __future__ import annotations

from typing import Any, Dict

import wandb  # noqa: F401 — ImportError gates backend

from .base import BaseLogger


class WandbLogger(BaseLogger):
    def __init__(self, log_dir: str, project: str = "scalerl-amd",
                 name: str = None, config: Dict[str, Any] = None, **kw):
        super().__init__(**kw)
        self.run = wandb.init(project=project, name=name, dir=log_dir,
                              config=config or {}, resume="allow")

    def write(self, namespace: str, step: int, data: Dict[str, Any]) -> None:
        wandb.log({f"{namespace}/{k}": v for k, v in data.items()}, step=step)

    def restore_data(self) -> Dict[str, int]:
        s = dict(self.run.summary) if self.run.resumed else {}
        return {"epoch": int(s.get("save/epoch", 0)),
                "env_step": int(s.get("save/env_step", 0)),
                "gradient_step": int(s.get("save/gradient_step", 0))}

    def close(self) -> None:
        self.run.finish()
