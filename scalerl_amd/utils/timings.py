"""Section timing.

:class:`Timings` — online mean/variance per named section (parity with
``scalerl/utils/profile.py:10-65``; same section taxonomy is used by the
IMPALA actor — ``model/step/write`` — and learner —
``dequeue/batch/device/learn``).  :class:`Timer` — context-manager wall
timer (parity with ``scalerl/utils/timer.py``).  On a GPU, device-side
section timing uses hipEvents via :class:`CudaTimings` (HIP events through
the torch.cuda.Event API on ROCm).
"""

from __future__ import annotations

import collections
import time
from typing import Dict


class Timings:
    """Welford online mean/std of deltas between ``time()`` calls."""

    def __init__(self):
        self._means: Dict[str, float] = collections.defaultdict(float)
        self._vars: Dict[str, float] = collections.defaultdict(float)
        self._counts: Dict[str, int] = collections.defaultdict(int)
        self.reset()

    def reset(self):
        self.last_time = time.perf_counter()

    def time(self, name: str):
        now = time.perf_counter()
        x = now - self.last_time
        self.last_time = now
        n = self._counts[name]
        mean = self._means[name]
        self._means[name] = (n * mean + x) / (n + 1)
        self._vars[name] = (n / (n + 1)) * self._vars[name] + (x - mean) * (
            x - self._means[name]) / (n + 1)
        self._counts[name] = n + 1

    def means(self) -> Dict[str, float]:
        return dict(self._means)

    def summary(self, prefix: str = "") -> str:
        means = self.means()
        total = sum(means.values()) or 1.0
        rows = [f"{prefix}{k}: {1000*v:.2f} ms ({100*v/total:.1f}%)"
                for k, v in sorted(means.items(), key=lambda kv: -kv[1])]
        rows.append(f"{prefix}total: {1000*total:.2f} ms")
        return "\n".join(rows)


class Timer:
    """``with Timer() as t: ...; t.elapsed``"""

    def __init__(self):
        self.elapsed = 0.0

    def __enter__(self):
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.elapsed = time.perf_counter() - self._t0
        return False


_TIMER_REGISTRY = {}


def check_time(name: str) -> "Timer":
    """Named global timers (parity with utils/timer.py's check_time
    registry): ``with check_time("phase"): ...`` accumulates per name."""
    t = _TIMER_REGISTRY.get(name)
    if t is None:
        t = _AccumTimer(name)
        _TIMER_REGISTRY[name] = t
    return t


def timer_report() -> str:
    return "\n".join(f"{n}: {t.total:.4f}s over {t.count} calls"
                      for n, t in sorted(_TIMER_REGISTRY.items()))


class _AccumTimer:
    def __init__(self, name: str):
        self.name = name
        self.total = 0.0
        self.count = 0

    def __enter__(self):
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.total += time.perf_counter() - self._t0
        self.count += 1
        return False


class CudaTimings:
    """hipEvent-based per-section device timings (lazy sync).

    Usage: ``ct.start(); ...; ct.mark("fwd"); ...; ct.mark("bwd")``;
    ``ct.means()`` synchronizes and folds all recorded deltas.
    """

    def __init__(self, max_pending: int = 64):
        import torch
        self._torch = torch
        self._pending = []  # list of (name, ev_start, ev_end)
        self._means: Dict[str, float] = collections.defaultdict(float)
        self._counts: Dict[str, int] = collections.defaultdict(int)
        self._max_pending = max_pending
        self._last_event = None

    def start(self):
        ev = self._torch.cuda.Event(enable_timing=True)
        ev.record()
        self._last_event = ev

    def mark(self, name: str):
        ev = self._torch.cuda.Event(enable_timing=True)
        ev.record()
        self._pending.append((name, self._last_event, ev))
        self._last_event = ev
        if len(self._pending) > self._max_pending:
            self._drain()

    def _drain(self):
        for name, e0, e1 in self._pending:
            e1.synchronize()
            ms = e0.elapsed_time(e1)
            n = self._counts[name]
            self._means[name] = (n * self._means[name] + ms / 1000.0) / (n + 1)
            self._counts[name] = n + 1
        self._pending.clear()

    def means(self) -> Dict[str, float]:
        self._drain()
        return dict(self._means)
