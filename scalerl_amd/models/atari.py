"""IMPALA Atari policy network.

Reference semantics (algorithms/utils/atari_model.py:8-143): Nature-CNN
3-conv encoder (8×8/4→32, 4×4/2→64, 3×3/1→64) + FC 3136→512, core input =
[features, clipped reward, one-hot last action], optional 2-layer LSTM with
per-step done masking, policy-logits + baseline heads, multinomial sample in
train / argmax in eval.  I/O is the [T,B,...] dict wire format.

MI355X design differences from the reference:
- the LSTM unroll is :class:`scalerl_amd.ops.MaskedLSTM` (hoisted input
  GEMM + one fused HIP pointwise kernel per step) instead of a Python loop
  over nn.LSTM;
- frames stay uint8 until the normalize-on-device divide;
- the conv/FC stack runs in bf16 under autocast on the learner (heads and
  LSTM stay fp32).
"""

from __future__ import annotations

from typing import Dict, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import MaskedLSTM


class AtariNet(nn.Module):
    def __init__(self, observation_shape=(4, 84, 84), num_actions: int = 6,
                 use_lstm: bool = True, native_conv: bool = None):
        super().__init__()
        self.observation_shape = tuple(observation_shape)
        self.num_actions = num_actions
        self.use_lstm = use_lstm
        # Hand-written MFMA encoder convs (ops/conv.py) — ALL kernels
        # hardware-validated (r2); opt-in because MIOpen still wins the
        # per-op A/B (profiles/README.md; the panel/stride-decomposed v3
        # kernels queued for r3 close the gap).  Requires (4,84,84).
        if native_conv is None:
            import os
            native_conv = bool(os.environ.get("SCALERL_NATIVE_CONV"))
        self.native_conv = native_conv and tuple(observation_shape) == (4, 84, 84)

        c = observation_shape[0]
        self.conv1 = nn.Conv2d(c, 32, kernel_size=8, stride=4)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=4, stride=2)
        self.conv3 = nn.Conv2d(64, 64, kernel_size=3, stride=1)
        conv_out = 64 * 7 * 7  # 84x84 input
        self.fc = nn.Linear(conv_out, 512)

        core_output_size = 512 + num_actions + 1
        if use_lstm:
            self.core = MaskedLSTM(core_output_size, core_output_size,
                                   num_layers=2)
        self.policy = nn.Linear(core_output_size, num_actions)
        self.baseline = nn.Linear(core_output_size, 1)

    def initial_state(self, batch_size: int, device=None):
        if not self.use_lstm:
            return tuple()
        return self.core.initial_state(batch_size, device=device)

    def encode(self, x: torch.Tensor) -> torch.Tensor:
        """uint8 [N,C,H,W] → fp feature [N,512]."""
        if self.native_conv and x.is_cuda:
            from ..ops.conv import native_conv
            h = native_conv(1, x, self.conv1.weight, self.conv1.bias)
            h = native_conv(2, h, self.conv2.weight, self.conv2.bias)
            h = native_conv(3, h, self.conv3.weight, self.conv3.bias)
            return F.relu(self.fc(torch.flatten(h, 1)))
        x = x.float() / 255.0
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.relu(self.conv3(x))
        x = torch.flatten(x, 1)
        return F.relu(self.fc(x))

    def forward(self, inputs: Dict[str, torch.Tensor],
                core_state: Tuple = (), greedy: bool = False):
        """inputs: obs [T,B,C,H,W] u8, reward [T,B], done [T,B] bool,
        last_action [T,B].  Returns (out dict, new core state)."""
        obs = inputs["obs"]
        T, B = obs.shape[:2]
        feat = self.encode(obs.flatten(0, 1))

        one_hot_action = F.one_hot(
            inputs["last_action"].view(T * B), self.num_actions).float()
        clipped_reward = torch.clamp(inputs["reward"].view(T * B, 1), -1, 1)
        core_input = torch.cat(
            [feat.float(), clipped_reward, one_hot_action], dim=-1)

        if self.use_lstm:
            notdone = (~inputs["done"].view(T, B)).float()
            core_output, core_state = self.core(
                core_input.view(T, B, -1), notdone, core_state)
            core_output = core_output.flatten(0, 1)
        else:
            core_output = core_input

        # Heads stay fp32 even under autocast: the V-trace ratio
        # exp(log pi - log mu) is noise-sensitive and behavior logits come
        # from the fp32 CPU actor.
        if core_output.is_cuda:
            with torch.autocast(device_type="cuda", enabled=False):
                policy_logits = self.policy(core_output.float())
                baseline = self.baseline(core_output.float())
        else:
            policy_logits = self.policy(core_output.float())
            baseline = self.baseline(core_output.float())
        if greedy or not self.training:
            action = torch.argmax(policy_logits, dim=-1)
        else:
            action = torch.multinomial(
                F.softmax(policy_logits, dim=-1), num_samples=1).squeeze(-1)
        return ({"policy_logits": policy_logits.view(T, B, self.num_actions),
                 "baseline": baseline.view(T, B),
                 "action": action.view(T, B)}, core_state)


class AtariQNet(nn.Module):
    """Nature-CNN Q-network for Ape-X (apex/network.py parity): uint8
    [N,C,84,84] → Q [N,A], with dueling and NoisyNet head options
    (Rainbow-style exploration for the image agent)."""

    def __init__(self, observation_shape=(4, 84, 84), num_actions: int = 6,
                 dueling: bool = True, noisy: bool = False):
        super().__init__()
        from .noisy import NoisyLinear
        c = observation_shape[0]
        self.num_actions = num_actions
        self.dueling = dueling
        self.noisy = noisy
        lin = NoisyLinear if noisy else nn.Linear
        self.conv1 = nn.Conv2d(c, 32, kernel_size=8, stride=4)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=4, stride=2)
        self.conv3 = nn.Conv2d(64, 64, kernel_size=3, stride=1)
        self.fc = nn.Linear(64 * 7 * 7, 512)
        if dueling:
            self.value_head = lin(512, 1)
            self.adv_head = lin(512, num_actions)
        else:
            self.head = lin(512, num_actions)

    def reset_noise(self):
        from .noisy import reset_noise
        reset_noise(self)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x.float() / 255.0
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.relu(self.conv3(x))
        h = F.relu(self.fc(torch.flatten(x, 1)))
        if self.dueling:
            v = self.value_head(h)
            a = self.adv_head(h)
            return v + a - a.mean(dim=-1, keepdim=True)
        return self.head(h)


class CategoricalAtariQNet(nn.Module):
    """C51 distributional Q-network on the Nature-CNN body (the reference
    declares v_min/v_max/atoms for the Atari-scale DQN, rl_args.py:221-260,
    but ships no image implementation).  Same ``dist``/``forward``/support
    surface as :class:`scalerl_amd.models.noisy.CategoricalQNet`, so
    ``c51_loss`` works against either."""

    def __init__(self, observation_shape=(4, 84, 84), num_actions: int = 6,
                 num_atoms: int = 51, v_min: float = -10.0,
                 v_max: float = 10.0, noisy: bool = False):
        super().__init__()
        from .noisy import NoisyLinear
        c = observation_shape[0]
        self.action_dim = self.num_actions = num_actions
        self.num_atoms = num_atoms
        self.noisy = noisy
        lin = NoisyLinear if noisy else nn.Linear
        self.conv1 = nn.Conv2d(c, 32, kernel_size=8, stride=4)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=4, stride=2)
        self.conv3 = nn.Conv2d(64, 64, kernel_size=3, stride=1)
        self.fc = nn.Linear(64 * 7 * 7, 512)
        self.head = lin(512, num_actions * num_atoms)
        self.register_buffer("support", torch.linspace(v_min, v_max, num_atoms))
        self.v_min, self.v_max = v_min, v_max
        self.delta_z = (v_max - v_min) / (num_atoms - 1)

    def reset_noise(self):
        from .noisy import reset_noise
        reset_noise(self)

    def dist(self, x: torch.Tensor) -> torch.Tensor:
        """→ log-probabilities [B, A, atoms]."""
        x = x.float() / 255.0
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.relu(self.conv3(x))
        h = F.relu(self.fc(torch.flatten(x, 1)))
        logits = self.head(h).view(-1, self.num_actions, self.num_atoms)
        return F.log_softmax(logits, dim=-1)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """Expected Q-values [B, A]."""
        return (self.dist(x).exp() * self.support).sum(-1)
