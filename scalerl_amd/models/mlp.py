"""MLP heads (parity with ``scalerl/algorithms/utils/network.py:5-95``:
QNet / ActorNet / CriticNet / ActorCriticNet — with the reference's broken
``network_init`` (iterating the unbound ``self.modules`` method, :87)
implemented correctly).  Plain GEMMs — rocBLAS via torch, no custom kernels
(SURVEY.md §2.1)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


def _init_linear(m: nn.Module) -> None:
    if isinstance(m, nn.Linear):
        nn.init.orthogonal_(m.weight, gain=2 ** 0.5)
        nn.init.zeros_(m.bias)


class QNet(nn.Module):
    """obs → Q-values; optional dueling decomposition and NoisyNet heads
    (noisy layers per Fortunato et al. 2018 — declared but unimplemented in
    the reference, rl_args.py:163-315)."""

    def __init__(self, obs_dim: int, action_dim: int, hidden_dim: int = 128,
                 dueling: bool = False, noisy: bool = False):
        super().__init__()
        from .noisy import NoisyLinear
        self.dueling = dueling
        self.noisy = noisy
        lin = NoisyLinear if noisy else nn.Linear
        self.body = nn.Sequential(
            nn.Linear(obs_dim, hidden_dim), nn.ReLU(),
            lin(hidden_dim, hidden_dim), nn.ReLU())
        if dueling:
            self.value_head = lin(hidden_dim, 1)
            self.adv_head = lin(hidden_dim, action_dim)
        else:
            self.head = lin(hidden_dim, action_dim)
        self.apply(_init_linear)

    def reset_noise(self):
        from .noisy import reset_noise
        reset_noise(self)

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        h = self.body(obs)
        if self.dueling:
            v = self.value_head(h)
            a = self.adv_head(h)
            return v + a - a.mean(dim=-1, keepdim=True)
        return self.head(h)


class ActorNet(nn.Module):
    def __init__(self, obs_dim: int, action_dim: int, hidden_dim: int = 128):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(obs_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, action_dim))
        self.apply(_init_linear)

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        return self.net(obs)  # logits


class CriticNet(nn.Module):
    def __init__(self, obs_dim: int, hidden_dim: int = 128):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(obs_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, 1))
        self.apply(_init_linear)

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        return self.net(obs)


class ActorCriticNet(nn.Module):
    """Shared body, policy-logits + value heads."""

    def __init__(self, obs_dim: int, action_dim: int, hidden_dim: int = 128):
        super().__init__()
        self.body = nn.Sequential(
            nn.Linear(obs_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU())
        self.policy = nn.Linear(hidden_dim, action_dim)
        self.value = nn.Linear(hidden_dim, 1)
        self.apply(_init_linear)

    def forward(self, obs: torch.Tensor):
        h = self.body(obs)
        return self.policy(h), self.value(h).squeeze(-1)

    def get_action(self, obs: torch.Tensor, greedy: bool = False):
        logits, value = self(obs)
        if greedy:
            action = logits.argmax(dim=-1)
        else:
            action = torch.multinomial(F.softmax(logits, dim=-1), 1).squeeze(-1)
        return action, logits, value
