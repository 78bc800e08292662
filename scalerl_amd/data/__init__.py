from .replay import (MultiStepReplayBuffer, PrioritizedReplayBuffer,
                     ReplayBuffer, Transition)
from .sampler import Sampler
from .replay_dataset import ReplayDataset

__all__ = ["ReplayBuffer", "MultiStepReplayBuffer", "PrioritizedReplayBuffer",
           "Transition", "Sampler", "ReplayDataset"]
