"""Synthetic benchmark environments ("a synthetic env
(random 84×84×4 frames — exactly the BASELINE.json benchmark config) as the
fake backend for deterministic actor-learner integration tests without ALE").

Design goals:
- *Cheap*: observations come from a pre-generated frame bank, so an env
  step is an index update + a view — the benchmark measures the learner
  pipeline, not numpy RNG.
- *Deterministic*: state transition is a pure function of (state, action).
- *Learnable*: reward = 1 iff ``action == state % num_actions``; the frame
  bank encodes the state, so integration tests can verify that the policy
  actually improves.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from .base import Box, Discrete, Env


class SyntheticAtariEnv(Env):
    """84×84×4 uint8 frames, Atari-like action space (benchmark config 3)."""

    spec_id = "synthetic-atari"

    def __init__(self, num_actions: int = 6, bank_size: int = 64,
                 episode_length: int = 1000, frame_shape: Tuple[int, int, int] = (4, 84, 84),
                 seed: Optional[int] = None):
        self.observation_space = Box(0, 255, shape=frame_shape, dtype=np.uint8)
        self.action_space = Discrete(num_actions)
        self.num_actions = num_actions
        self.bank_size = bank_size
        self.episode_length = episode_length
        rng = np.random.default_rng(1234)  # bank is identical across envs
        self.bank = rng.integers(0, 256, size=(bank_size, *frame_shape), dtype=np.uint8)
        self._rng = np.random.default_rng(seed)
        self._state = 0
        self._steps = 0

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self._state = int(self._rng.integers(self.bank_size))
        self._steps = 0
        return self.bank[self._state], {}

    def step(self, action):
        action = int(action)
        reward = 1.0 if action == (self._state % self.num_actions) else 0.0
        noise = int(self._rng.integers(4))  # stochastic transitions keep
        # state coverage broad even under a (near-)deterministic policy
        self._state = (self._state * 5 + action + 1 + noise) % self.bank_size
        self._steps += 1
        truncated = self._steps >= self.episode_length
        return self.bank[self._state], reward, False, truncated, {}


class SyntheticAtariVecEnv:
    """Natively vectorized synthetic Atari: steps E envs with numpy batch ops.

    This is what the IMPALA/Ape-X actors run — per-env Python loops would
    bottleneck the 8-core CPU side long before the MI355X learner saturates.
    API: :meth:`reset` → obs [E,C,H,W] u8; :meth:`step(actions[E])` →
    (obs, reward[E] f32, done[E] bool).  Episodes auto-reset (done reports
    the *pre-reset* terminal, as IMPALA's TorchEnvWrapper expects).
    """

    def __init__(self, num_envs: int, num_actions: int = 6, bank_size: int = 64,
                 episode_length: int = 1000,
                 frame_shape: Tuple[int, int, int] = (4, 84, 84),
                 seed: Optional[int] = None):
        self.num_envs = num_envs
        self.num_actions = num_actions
        self.bank_size = bank_size
        self.episode_length = episode_length
        rng = np.random.default_rng(1234)
        self.bank = rng.integers(0, 256, size=(bank_size, *frame_shape), dtype=np.uint8)
        self._rng = np.random.default_rng(seed)
        self.observation_space = Box(0, 255, shape=frame_shape, dtype=np.uint8)
        self.action_space = Discrete(num_actions)
        self._state = np.zeros(num_envs, dtype=np.int64)
        self._steps = np.zeros(num_envs, dtype=np.int64)

    def reset(self) -> np.ndarray:
        self._state = self._rng.integers(self.bank_size, size=self.num_envs)
        self._steps[:] = 0
        return self.bank[self._state]

    def step(self, actions: np.ndarray):
        actions = np.asarray(actions, dtype=np.int64)
        reward = (actions == (self._state % self.num_actions)).astype(np.float32)
        noise = self._rng.integers(4, size=self.num_envs)
        self._state = (self._state * 5 + actions + 1 + noise) % self.bank_size
        self._steps += 1
        done = self._steps >= self.episode_length
        if done.any():
            n = int(done.sum())
            self._state[done] = self._rng.integers(self.bank_size, size=n)
            self._steps[done] = 0
        return self.bank[self._state], reward, done


class SyntheticPointGoalEnv(Env):
    """128×128 RGB-D PointGoal navigation stand-in (benchmark config 5,
    DD-PPO).  Obs = 4×128×128 uint8 (RGB + depth) plus a 2-d goal vector
    appended by the model from ``info['pointgoal']``-free design: the goal
    is folded into channel statistics so the obs is a single tensor."""

    spec_id = "synthetic-pointgoal"

    def __init__(self, num_actions: int = 4, bank_size: int = 32,
                 episode_length: int = 500, seed: Optional[int] = None):
        shape = (4, 128, 128)
        self.observation_space = Box(0, 255, shape=shape, dtype=np.uint8)
        self.action_space = Discrete(num_actions)
        self.num_actions = num_actions
        self.bank_size = bank_size
        self.episode_length = episode_length
        rng = np.random.default_rng(4321)
        self.bank = rng.integers(0, 256, size=(bank_size, *shape), dtype=np.uint8)
        self._rng = np.random.default_rng(seed)
        self._state = 0
        self._steps = 0

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self._state = int(self._rng.integers(self.bank_size))
        self._steps = 0
        return self.bank[self._state], {}

    def step(self, action):
        action = int(action)
        reward = 1.0 if action == (self._state % self.num_actions) else -0.01
        self._state = (self._state * 3 + action + 2) % self.bank_size
        self._steps += 1
        truncated = self._steps >= self.episode_length
        return self.bank[self._state], reward, False, truncated, {}
