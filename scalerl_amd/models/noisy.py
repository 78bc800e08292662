"""NoisyNet linear layer + C51 categorical Q-network.

The reference declares ``noisy_dqn`` / ``categorical_dqn`` flags with
v_min/v_max/atoms hyperparams (rl_args.py:163-315) but ships no
implementation; these implement the published semantics (Fortunato et al.
2018 factorised-Gaussian NoisyNet; Bellemare et al. 2017 C51)."""

from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F


class NoisyLinear(nn.Module):
    """Factorised-Gaussian noisy linear: w = μ_w + σ_w·(ε_out ⊗ ε_in)."""

    def __init__(self, in_features: int, out_features: int,
                 sigma0: float = 0.5):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight_mu = nn.Parameter(torch.empty(out_features, in_features))
        self.weight_sigma = nn.Parameter(torch.empty(out_features, in_features))
        self.bias_mu = nn.Parameter(torch.empty(out_features))
        self.bias_sigma = nn.Parameter(torch.empty(out_features))
        self.register_buffer("eps_in", torch.zeros(in_features))
        self.register_buffer("eps_out", torch.zeros(out_features))
        bound = 1.0 / math.sqrt(in_features)
        nn.init.uniform_(self.weight_mu, -bound, bound)
        nn.init.uniform_(self.bias_mu, -bound, bound)
        nn.init.constant_(self.weight_sigma, sigma0 * bound)
        nn.init.constant_(self.bias_sigma, sigma0 * bound)
        self.reset_noise()

    @staticmethod
    def _f(x: torch.Tensor) -> torch.Tensor:
        return x.sign() * x.abs().sqrt()

    @torch.no_grad()
    def reset_noise(self) -> None:
        self.eps_in.copy_(self._f(torch.randn_like(self.eps_in)))
        self.eps_out.copy_(self._f(torch.randn_like(self.eps_out)))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.training:
            w = self.weight_mu + self.weight_sigma * torch.outer(
                self.eps_out, self.eps_in)
            b = self.bias_mu + self.bias_sigma * self.eps_out
        else:
            w, b = self.weight_mu, self.bias_mu
        return F.linear(x, w, b)


def reset_noise(module: nn.Module) -> None:
    for m in module.modules():
        if isinstance(m, NoisyLinear):
            m.reset_noise()


class CategoricalQNet(nn.Module):
    """C51 distributional Q-network over ``num_atoms`` support points."""

    def __init__(self, obs_dim: int, action_dim: int, hidden_dim: int = 128,
                 num_atoms: int = 51, v_min: float = -10.0,
                 v_max: float = 10.0, noisy: bool = False):
        super().__init__()
        self.action_dim = action_dim
        self.num_atoms = num_atoms
        lin = NoisyLinear if noisy else nn.Linear
        self.body = nn.Sequential(
            nn.Linear(obs_dim, hidden_dim), nn.ReLU(),
            nn.Linear(hidden_dim, hidden_dim), nn.ReLU())
        self.head = lin(hidden_dim, action_dim * num_atoms)
        self.register_buffer(
            "support", torch.linspace(v_min, v_max, num_atoms))
        self.v_min, self.v_max = v_min, v_max
        self.delta_z = (v_max - v_min) / (num_atoms - 1)

    def dist(self, obs: torch.Tensor) -> torch.Tensor:
        """→ log-probabilities [B, A, atoms]."""
        h = self.body(obs)
        logits = self.head(h).view(-1, self.action_dim, self.num_atoms)
        return F.log_softmax(logits, dim=-1)

    def forward(self, obs: torch.Tensor) -> torch.Tensor:
        """Expected Q-values [B, A]."""
        return (self.dist(obs).exp() * self.support).sum(-1)


def c51_loss(net: CategoricalQNet, target_net: CategoricalQNet, obs, actions,
             rewards, discounts, next_obs, double: bool = True,
             weights=None):
    """Categorical projection TD loss (Bellemare et al. 2017, alg. 1).

    Returns (loss scalar, |TD|-style priorities = per-sample KL)."""
    B = obs.shape[0]
    with torch.no_grad():
        next_q = (net if double else target_net)(next_obs)
        astar = next_q.argmax(dim=1)
        next_dist = target_net.dist(next_obs).exp()[
            torch.arange(B), astar]                       # [B, atoms]
        tz = (rewards.unsqueeze(1)
              + discounts.unsqueeze(1) * net.support.unsqueeze(0))
        tz = tz.clamp(net.v_min, net.v_max)
        b = (tz - net.v_min) / net.delta_z
        lo = b.floor().long().clamp(0, net.num_atoms - 1)
        hi = b.ceil().long().clamp(0, net.num_atoms - 1)
        proj = torch.zeros_like(next_dist)
        # distribute mass to the two neighbouring atoms
        lo_w = (hi.float() - b).where(hi != lo, torch.ones_like(b))
        hi_w = (b - lo.float())
        proj.scatter_add_(1, lo, next_dist * lo_w)
        proj.scatter_add_(1, hi, next_dist * hi_w)
    log_p = net.dist(obs)[torch.arange(B), actions]       # [B, atoms]
    kl = -(proj * log_p).sum(dim=1)
    loss = (kl * weights).mean() if weights is not None else kl.mean()
    return loss, kl.detach()
