"""Array-backed segment trees — reference-API parity
(``scalerl/data/segment_tree.py:7-197``: SegmentTree / SumSegmentTree with
``find_prefixsum_idx`` / MinSegmentTree).

The PRODUCTION prioritized replay uses the device-resident
:class:`scalerl_amd.ops.SumTree` (HIP update/sample kernels); this module
is the CPU reference surface for code written against the reference's
classes, and doubles as an oracle in tests.
"""

from __future__ import annotations

import operator
from typing import Callable


class SegmentTree:
    """Fixed-capacity (power of two) tree supporting O(log n) point
    updates and range reductions with an arbitrary associative op."""

    def __init__(self, capacity: int, operation: Callable,
                 neutral_element: float):
        assert capacity > 0 and capacity & (capacity - 1) == 0, \
            "capacity must be a positive power of 2"
        self._capacity = capacity
        self._op = operation
        self._neutral = neutral_element
        self._value = [neutral_element] * (2 * capacity)

    def _reduce(self, start: int, end: int, node: int, node_start: int,
                node_end: int):
        if start == node_start and end == node_end:
            return self._value[node]
        mid = (node_start + node_end) // 2
        if end <= mid:
            return self._reduce(start, end, 2 * node, node_start, mid)
        if start > mid:
            return self._reduce(start, end, 2 * node + 1, mid + 1, node_end)
        return self._op(
            self._reduce(start, mid, 2 * node, node_start, mid),
            self._reduce(mid + 1, end, 2 * node + 1, mid + 1, node_end))

    def reduce(self, start: int = 0, end: int = None):
        """Reduce op over [start, end) (reference semantics)."""
        if end is None:
            end = self._capacity
        if end <= 0:
            end += self._capacity
        return self._reduce(start, end - 1, 1, 0, self._capacity - 1)

    def __setitem__(self, idx: int, val: float) -> None:
        idx += self._capacity
        self._value[idx] = val
        idx //= 2
        while idx >= 1:
            self._value[idx] = self._op(self._value[2 * idx],
                                        self._value[2 * idx + 1])
            idx //= 2

    def __getitem__(self, idx: int) -> float:
        assert 0 <= idx < self._capacity
        return self._value[self._capacity + idx]


class SumSegmentTree(SegmentTree):
    def __init__(self, capacity: int):
        super().__init__(capacity, operator.add, 0.0)

    def sum(self, start: int = 0, end: int = None) -> float:
        return self.reduce(start, end)

    def find_prefixsum_idx(self, prefixsum: float) -> int:
        """Largest i such that sum(arr[0..i-1]) <= prefixsum — the
        root-to-leaf descent PER sampling uses."""
        assert 0 <= prefixsum <= self.sum() + 1e-5
        idx = 1
        while idx < self._capacity:  # descend to a leaf
            left = 2 * idx
            if self._value[left] > prefixsum:
                idx = left
            else:
                prefixsum -= self._value[left]
                idx = left + 1
        return idx - self._capacity


class MinSegmentTree(SegmentTree):
    def __init__(self, capacity: int):
        super().__init__(capacity, min, float("inf"))

    def min(self, start: int = 0, end: int = None) -> float:
        return self.reduce(start, end)
