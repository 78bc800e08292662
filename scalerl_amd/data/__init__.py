from .replay import (MultiStepReplayBuffer, PrioritizedReplayBuffer,
                     ReplayBuffer, Transition)
from .sampler import Sampler
from .replay_dataset import ReplayDataset
from .segment_tree import MinSegmentTree, SegmentTree, SumSegmentTree

__all__ = ["ReplayBuffer", "MultiStepReplayBuffer", "PrioritizedReplayBuffer",
           "Transition", "Sampler", "ReplayDataset",
           "SegmentTree", "SumSegmentTree", "MinSegmentTree"]
