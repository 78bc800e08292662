"""Value schedulers (parity with ``scalerl/utils/lr_scheduler.py:7-118``:
PiecewiseScheduler / LinearDecayScheduler / MultiStepScheduler)."""

from __future__ import annotations

from typing import List, Sequence, Tuple


class PiecewiseScheduler:
    """Piecewise-linear interpolation over (step, value) breakpoints."""

    def __init__(self, endpoints: Sequence[Tuple[int, float]]):
        assert len(endpoints) >= 1
        assert all(e1[0] < e2[0] for e1, e2 in zip(endpoints, endpoints[1:]))
        self.endpoints = list(endpoints)
        self.cur_step = 0

    def value(self, step: int) -> float:
        pts = self.endpoints
        if step <= pts[0][0]:
            return pts[0][1]
        for (s0, v0), (s1, v1) in zip(pts, pts[1:]):
            if s0 <= step < s1:
                frac = (step - s0) / (s1 - s0)
                return v0 + frac * (v1 - v0)
        return pts[-1][1]

    def step(self, step_num: int = 1) -> float:
        self.cur_step += step_num
        return self.value(self.cur_step)


class LinearDecayScheduler:
    """Linear decay from start to end over ``decay_steps`` (used for
    ε-greedy exploration like the reference's dqn_agent.py:84-88)."""

    def __init__(self, start_value: float, end_value: float, decay_steps: int):
        self.start_value = float(start_value)
        self.end_value = float(end_value)
        self.decay_steps = max(1, int(decay_steps))
        self.cur_step = 0

    def value(self, step: int) -> float:
        frac = min(max(step / self.decay_steps, 0.0), 1.0)
        return self.start_value + frac * (self.end_value - self.start_value)

    def step(self, step_num: int = 1) -> float:
        self.cur_step += step_num
        return self.value(self.cur_step)


class MultiStepScheduler:
    """Multiplies by ``gamma`` at each milestone step."""

    def __init__(self, start_value: float, milestones: List[int], gamma: float = 0.1):
        assert sorted(milestones) == list(milestones)
        self.start_value = float(start_value)
        self.milestones = milestones
        self.gamma = gamma
        self.cur_step = 0

    def value(self, step: int) -> float:
        v = self.start_value
        for m in self.milestones:
            if step >= m:
                v *= self.gamma
        return v

    def step(self, step_num: int = 1) -> float:
        self.cur_step += step_num
        return self.value(self.cur_step)
