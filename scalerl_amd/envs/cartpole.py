"""CartPole-v1 with the standard published dynamics — the benchmark's
config-1 environment (DQN plumbing config), implemented locally because
gym is not in this image.  Physics constants and termination thresholds
follow the classic formulation (Barto, Sutton & Anderson 1983) used by
gym's CartPole-v1 (max 500 steps, reward 1 per step)."""

from __future__ import annotations

import math
from typing import Optional

import numpy as np

from .base import Box, Discrete, Env


class CartPoleEnv(Env):
    spec_id = "CartPole-v1"

    GRAVITY = 9.8
    MASSCART = 1.0
    MASSPOLE = 0.1
    TOTAL_MASS = MASSCART + MASSPOLE
    LENGTH = 0.5  # half pole length
    POLEMASS_LENGTH = MASSPOLE * LENGTH
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_THRESHOLD = 12 * 2 * math.pi / 360
    X_THRESHOLD = 2.4
    MAX_STEPS = 500

    def __init__(self, seed: Optional[int] = None):
        high = np.array([self.X_THRESHOLD * 2, np.inf,
                         self.THETA_THRESHOLD * 2, np.inf], dtype=np.float32)
        self.observation_space = Box(-high, high)
        self.action_space = Discrete(2)
        self._rng = np.random.default_rng(seed)
        self._state = np.zeros(4, dtype=np.float64)
        self._steps = 0

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self._state = self._rng.uniform(-0.05, 0.05, size=4)
        self._steps = 0
        return self._state.astype(np.float32), {}

    def step(self, action):
        x, x_dot, theta, theta_dot = self._state
        force = self.FORCE_MAG if int(action) == 1 else -self.FORCE_MAG
        costheta, sintheta = math.cos(theta), math.sin(theta)
        temp = (force + self.POLEMASS_LENGTH * theta_dot**2 * sintheta) / self.TOTAL_MASS
        thetaacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASSPOLE * costheta**2 / self.TOTAL_MASS))
        xacc = temp - self.POLEMASS_LENGTH * thetaacc * costheta / self.TOTAL_MASS
        x += self.TAU * x_dot
        x_dot += self.TAU * xacc
        theta += self.TAU * theta_dot
        theta_dot += self.TAU * thetaacc
        self._state = np.array([x, x_dot, theta, theta_dot])
        self._steps += 1
        terminated = bool(abs(x) > self.X_THRESHOLD or abs(theta) > self.THETA_THRESHOLD)
        truncated = bool(self._steps >= self.MAX_STEPS)
        return self._state.astype(np.float32), 1.0, terminated, truncated, {}
