"""IterableDataset over a replay buffer (parity with
``scalerl/data/replay_data.py:8-26``): lets a torch DataLoader drive replay
sampling, e.g. to put CPU-side sampling on worker processes.  The MI355X
learner path samples device-resident buffers directly and does not need
this; it is kept for API parity and CPU-trainer use."""

from __future__ import annotations

from torch.utils.data import IterableDataset


class ReplayDataset(IterableDataset):
    def __init__(self, buffer, batch_size: int):
        self.buffer = buffer
        self.batch_size = batch_size

    def __iter__(self):
        while True:
            yield self.buffer.sample(self.batch_size)
