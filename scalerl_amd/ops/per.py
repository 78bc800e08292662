"""Device-resident prioritized-replay sum tree.

GPU: tree lives in HBM, updated/sampled by kernels (csrc/per_tree.hip).
CPU: numpy-free torch implementation with identical semantics (the oracle).
Reference: data/segment_tree.py:7-197, replay_buffer.py:276-381.
"""

from __future__ import annotations

import ctypes
from typing import Tuple

import torch

from . import _backend

_c = ctypes.c_void_p


def _next_pow2(n: int) -> int:
    m = 1
    while m < n:
        m <<= 1
    return m


class SumTree:
    """Batched sum tree on either device.  Leaves hold p^alpha directly."""

    def __init__(self, capacity: int, device="cpu"):
        self.capacity = capacity
        self.M = _next_pow2(capacity)
        self.device = torch.device(device)
        self.tree = torch.zeros(2 * self.M, dtype=torch.float32,
                                device=self.device)
        self.size = 0  # live leaves (monotone up to capacity)
        self._min_bits = torch.zeros(1, dtype=torch.int32, device=self.device)

    @property
    def total(self) -> torch.Tensor:
        return self.tree[1]

    def update(self, idx: torch.Tensor, prio: torch.Tensor,
               max_idx: int = None) -> None:
        """Set leaves idx (long [B]) to prio (float [B]) and fix ancestors.

        ``max_idx``: highest index in ``idx`` + 1, if the caller knows it
        (avoids a device→host sync on the GPU path; replay buffers track
        their own cursor so they always do).
        """
        idx = idx.to(self.device, torch.long).contiguous()
        prio = prio.to(self.device, torch.float32).contiguous()
        if max_idx is None:
            max_idx = int(idx.max()) + 1
        self.size = max(self.size, max_idx)
        if self.tree.is_cuda:
            ret = _backend.lib().per_update(
                _c(self.tree.data_ptr()), self.M, _c(idx.data_ptr()),
                _c(prio.data_ptr()), idx.numel(), _backend.current_stream())
            _backend.check(ret, "per_update")
            return
        for k in range(idx.numel()):
            node = self.M + int(idx[k])
            self.tree[node] = prio[k]
            node >>= 1
            while node >= 1:
                self.tree[node] = self.tree[2 * node] + self.tree[2 * node + 1]
                node >>= 1

    def sample(self, batch: int, generator=None) -> Tuple[torch.Tensor, torch.Tensor]:
        """Stratified proportional sample → (idx [B] long, prio [B])."""
        u = torch.rand(batch, device=self.device, generator=generator)
        if self.tree.is_cuda:
            idx = torch.empty(batch, dtype=torch.long, device=self.device)
            prio = torch.empty(batch, dtype=torch.float32, device=self.device)
            ret = _backend.lib().per_sample(
                _c(self.tree.data_ptr()), self.M, self.size, _c(u.data_ptr()),
                batch, _c(idx.data_ptr()), _c(prio.data_ptr()),
                _backend.current_stream())
            _backend.check(ret, "per_sample")
            return idx, prio
        total = float(self.tree[1])
        idx = torch.empty(batch, dtype=torch.long)
        prio = torch.empty(batch, dtype=torch.float32)
        for k in range(batch):
            mass = (k + float(u[k])) / batch * total
            node = 1
            while node < self.M:
                left = 2 * node
                if mass <= float(self.tree[left]):
                    node = left
                else:
                    mass -= float(self.tree[left])
                    node = left + 1
            i = min(node - self.M, self.size - 1)
            idx[k] = i
            prio[k] = self.tree[self.M + i]
        return idx, prio

    def min_leaf(self) -> torch.Tensor:
        """Min priority among live leaves (device scalar)."""
        if self.tree.is_cuda:
            self._min_bits.fill_(0x7f7fffff)  # +FLT_MAX bits
            ret = _backend.lib().per_leaf_min(
                _c(self.tree.data_ptr()), self.M, self.size,
                _c(self._min_bits.data_ptr()), _backend.current_stream())
            _backend.check(ret, "per_leaf_min")
            return self._min_bits.view(torch.float32)[0]
        return self.tree[self.M:self.M + self.size].min()

    def state_dict(self):
        return {"tree": self.tree, "size": self.size}

    def load_state_dict(self, sd):
        self.tree.copy_(sd["tree"])
        self.size = int(sd["size"])
