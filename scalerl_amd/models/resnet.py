"""ResNet-18 + LSTM policy for DD-PPO (benchmark config 5: synthetic
128×128 RGB-D PointGoal).  Not present in the reference's code (README
citation only) — implemented fresh per SURVEY.md §7 step 9, following the
DD-PPO paper's architecture sketch (Wijmans et al. 2020): ResNet visual
encoder → LSTM → actor/critic heads.  torchvision is not in this image, so
the ResNet-18 trunk is defined here (GroupNorm instead of BatchNorm, as
DD-PPO uses, because per-rank batch stats don't sync)."""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import MaskedLSTM


def _gn(ch: int) -> nn.GroupNorm:
    return nn.GroupNorm(min(32, ch), ch)


class BasicBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, stride=stride, padding=1,
                               bias=False)
        self.n1 = _gn(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, padding=1, bias=False)
        self.n2 = _gn(out_ch)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                _gn(out_ch))

    def forward(self, x):
        idn = x if self.down is None else self.down(x)
        x = F.relu(self.n1(self.conv1(x)))
        x = self.n2(self.conv2(x))
        return F.relu(x + idn)


class ResNet18Encoder(nn.Module):
    def __init__(self, in_channels: int = 4, base: int = 64):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(in_channels, base, 7, stride=2, padding=3, bias=False),
            _gn(base), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1))
        chs = [base, base * 2, base * 4, base * 8]
        layers = []
        in_ch = base
        for i, ch in enumerate(chs):
            stride = 1 if i == 0 else 2
            layers += [BasicBlock(in_ch, ch, stride), BasicBlock(ch, ch)]
            in_ch = ch
        self.layers = nn.Sequential(*layers)
        self.out_dim = chs[-1]

    def forward(self, x):
        x = self.stem(x)
        x = self.layers(x)
        return F.adaptive_avg_pool2d(x, 1).flatten(1)


class ResNetLSTMPolicy(nn.Module):
    """[T,B,4,128,128] uint8 → policy logits / value, LSTM core over T."""

    def __init__(self, num_actions: int = 4, in_channels: int = 4,
                 hidden: int = 512):
        super().__init__()
        self.num_actions = num_actions
        self.encoder = ResNet18Encoder(in_channels)
        self.fc = nn.Linear(self.encoder.out_dim, hidden)
        self.core = MaskedLSTM(hidden, hidden, num_layers=1)
        self.policy = nn.Linear(hidden, num_actions)
        self.value = nn.Linear(hidden, 1)

    def initial_state(self, batch_size: int, device=None):
        return self.core.initial_state(batch_size, device=device)

    def forward(self, obs: torch.Tensor, notdone: torch.Tensor, state):
        """obs [T,B,C,H,W] u8; notdone [T,B].  → logits [T,B,A], value [T,B]."""
        T, B = obs.shape[:2]
        x = obs.flatten(0, 1).float() / 255.0
        feat = F.relu(self.fc(self.encoder(x)))
        out, state = self.core(feat.view(T, B, -1), notdone, state)
        out = out.flatten(0, 1).float()
        return (self.policy(out).view(T, B, -1),
                self.value(out).view(T, B), state)
