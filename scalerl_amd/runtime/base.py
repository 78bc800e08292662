"""Agent interface (parity with ``scalerl/algorithms/base.py:7-124``:
get_action / predict / learn / get_weights / set_weights /
save_checkpoint / load_checkpoint / name)."""

from __future__ import annotations

import abc
from typing import Any, Dict

import torch


class BaseAgent(abc.ABC):
    def __init__(self, args):
        self.args = args
        self.global_update_step = 0

    @property
    def name(self) -> str:
        return getattr(self.args, "algo_name", type(self).__name__)

    @abc.abstractmethod
    def get_action(self, obs) -> Any:
        """Exploration action (training)."""

    @abc.abstractmethod
    def predict(self, obs) -> Any:
        """Greedy action (evaluation)."""

    @abc.abstractmethod
    def learn(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        ...

    def get_weights(self) -> Dict[str, torch.Tensor]:
        return {k: v.cpu() for k, v in self.model.state_dict().items()}

    def set_weights(self, weights: Dict[str, torch.Tensor]) -> None:
        self.model.load_state_dict(weights)

    def save_checkpoint(self, path: str) -> None:
        raise NotImplementedError

    def load_checkpoint(self, path: str) -> None:
        raise NotImplementedError
