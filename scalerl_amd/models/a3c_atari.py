"""A3C Atari policy (parity with ``scalerl/algorithms/a3c/utils/
atari_model.py:57-144``: 4×(3×3/2 conv, ELU) on 42×42 input → LSTM(288,256)
→ actor/critic heads, with normalized-columns init for the heads).  The
LSTMCell is this framework's MaskedLSTM stepped with T=1."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import MaskedLSTM


def normalized_columns_init_(weight: torch.Tensor, std: float = 1.0) -> None:
    out = torch.randn_like(weight)
    out *= std / torch.sqrt(out.pow(2).sum(dim=1, keepdim=True))
    with torch.no_grad():
        weight.copy_(out)


class A3CAtariNet(nn.Module):
    def __init__(self, in_channels: int = 1, num_actions: int = 6):
        super().__init__()
        self.num_actions = num_actions
        self.conv1 = nn.Conv2d(in_channels, 32, 3, stride=2, padding=1)
        self.conv2 = nn.Conv2d(32, 32, 3, stride=2, padding=1)
        self.conv3 = nn.Conv2d(32, 32, 3, stride=2, padding=1)
        self.conv4 = nn.Conv2d(32, 32, 3, stride=2, padding=1)
        self.core = MaskedLSTM(32 * 3 * 3, 256, num_layers=1)
        self.actor = nn.Linear(256, num_actions)
        self.critic = nn.Linear(256, 1)
        normalized_columns_init_(self.actor.weight, 0.01)
        normalized_columns_init_(self.critic.weight, 1.0)
        nn.init.zeros_(self.actor.bias)
        nn.init.zeros_(self.critic.bias)

    def initial_state(self, batch_size: int, device=None):
        return self.core.initial_state(batch_size, device=device)

    def forward(self, obs: torch.Tensor, state, notdone=None):
        """obs [B,C,42,42] float; one step.  Returns logits, value, state."""
        B = obs.shape[0]
        x = F.elu(self.conv1(obs))
        x = F.elu(self.conv2(x))
        x = F.elu(self.conv3(x))
        x = F.elu(self.conv4(x))
        x = x.flatten(1).unsqueeze(0)  # [1,B,288]
        if notdone is None:
            notdone = torch.ones(1, B, device=obs.device)
        out, state = self.core(x, notdone, state)
        h = out.squeeze(0)
        return self.actor(h), self.critic(h).squeeze(-1), state

    def unroll(self, obs: torch.Tensor, notdone: torch.Tensor, state):
        """Learner-side unroll: obs [T,B,C,42,42], notdone [T,B] →
        (logits [T,B,A], values [T,B], state).  Convs batched over T*B,
        recurrence through the MaskedLSTM core (done-masked)."""
        T, B = obs.shape[:2]
        x = obs.flatten(0, 1)
        x = F.elu(self.conv1(x))
        x = F.elu(self.conv2(x))
        x = F.elu(self.conv3(x))
        x = F.elu(self.conv4(x))
        x = x.flatten(1).view(T, B, -1)
        out, state = self.core(x, notdone, state)
        h = out.flatten(0, 1).float()
        if h.is_cuda:
            # heads fp32 under autocast (value/logit noise sensitivity)
            with torch.autocast(device_type="cuda", enabled=False):
                logits = self.actor(h)
                values = self.critic(h).squeeze(-1)
        else:
            logits = self.actor(h)
            values = self.critic(h).squeeze(-1)
        return (logits.view(T, B, self.num_actions), values.view(T, B),
                state)
