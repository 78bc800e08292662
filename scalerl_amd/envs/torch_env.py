"""Dict-tensor env protocol — the actor↔buffer wire format.

Parity with the reference's MonoBeast-heritage ``TorchEnvWrapper``
(``scalerl/envs/torch_envwrapper.py:16-88``): the env output is a dict of
tensors ``{obs, reward, done, episode_return, episode_step, last_action}``,
auto-reset on done.  In this framework the same field set is what actor
processes write into rollout slots (see
:mod:`scalerl_amd.parallel.rollout`); here it is shaped [1,1,...] for
single-env agents, matching the reference's unbatched actor loop.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch

from .base import Env


class TorchEnvWrapper:
    def __init__(self, env: Env):
        self.env = env
        self.episode_return: Optional[torch.Tensor] = None
        self.episode_step: Optional[torch.Tensor] = None

    def _obs(self, obs: np.ndarray) -> torch.Tensor:
        return torch.from_numpy(np.ascontiguousarray(obs)).view(
            1, 1, *obs.shape)

    def initial(self, seed: Optional[int] = None) -> Dict[str, torch.Tensor]:
        obs, _ = self.env.reset(seed=seed)
        self.episode_return = torch.zeros(1, 1)
        self.episode_step = torch.zeros(1, 1, dtype=torch.int32)
        return {
            "obs": self._obs(obs),
            "reward": torch.zeros(1, 1),
            "done": torch.ones(1, 1, dtype=torch.bool),
            "episode_return": self.episode_return.clone(),
            "episode_step": self.episode_step.clone(),
            "last_action": torch.zeros(1, 1, dtype=torch.int64),
        }

    def step(self, action: torch.Tensor) -> Dict[str, torch.Tensor]:
        obs, reward, term, trunc, _ = self.env.step(int(action.item()))
        done = term or trunc
        self.episode_step += 1
        self.episode_return += reward
        ep_ret = self.episode_return.clone()
        ep_step = self.episode_step.clone()
        if done:
            obs, _ = self.env.reset()
            self.episode_return.zero_()
            self.episode_step.zero_()
        return {
            "obs": self._obs(obs),
            "reward": torch.tensor(reward).view(1, 1),
            "done": torch.tensor(done).view(1, 1),
            "episode_return": ep_ret,
            "episode_step": ep_step,
            "last_action": action.view(1, 1),
        }

    def close(self):
        self.env.close()
