"""Sampler facade (parity with ``scalerl/data/sampler.py:10-71``): strategy
dispatch over {uniform, n-step, prioritized, distributed} sampling, hiding
the buffer type from the trainer loop."""

from __future__ import annotations

from typing import Dict

import torch

from .replay import MultiStepReplayBuffer, PrioritizedReplayBuffer, ReplayBuffer


class Sampler:
    def __init__(self, buffer: ReplayBuffer, batch_size: int,
                 distributed: bool = False):
        self.buffer = buffer
        self.batch_size = batch_size
        self.distributed = distributed
        self.is_per = isinstance(buffer, PrioritizedReplayBuffer)
        self.is_nstep = isinstance(buffer, MultiStepReplayBuffer)

    def sample(self) -> Dict[str, torch.Tensor]:
        """Per-rank independent draws: with per-rank device-resident buffers
        there is no cross-rank sampling to coordinate (the reference's
        Accelerate-sharded DataLoader path, replay_data.py:8-26, exists for
        buffers shared across ranks — each rank here owns its shard)."""
        return self.buffer.sample(self.batch_size)

    def update_priorities(self, idx, td_abs) -> None:
        if self.is_per:
            self.buffer.update_priorities(idx, td_abs)
