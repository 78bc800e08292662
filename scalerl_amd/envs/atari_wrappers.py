"""DeepMind Atari preprocessing stack (capability parity with
``scalerl/envs/atari_wrapper.py:19-311``: NoopReset / MaxAndSkip /
EpisodicLife / FireReset / WarpFrame / ScaledFloat / ClipReward /
FrameStack, composed by :func:`wrap_deepmind`).

Implemented against this package's Env API and cv2-free: WarpFrame uses a
numpy bilinear resample.  These wrappers run host-side in actor processes
(frame preprocessing is CPU work in the MI355X design; frames ship to the
learner as uint8).
"""

from __future__ import annotations

import collections
from typing import Optional

import numpy as np

from .base import Box, Env, Wrapper


def _rgb_to_gray(frame: np.ndarray) -> np.ndarray:
    # ITU-R 601 luma, matching cv2.cvtColor(RGB2GRAY) coefficients.
    return (frame[..., 0] * 0.299 + frame[..., 1] * 0.587 +
            frame[..., 2] * 0.114).astype(np.float32)


def _resize_bilinear(img: np.ndarray, out_h: int, out_w: int) -> np.ndarray:
    """cv2.resize(INTER_LINEAR)-style bilinear resample of a 2-D array."""
    in_h, in_w = img.shape
    y = (np.arange(out_h) + 0.5) * in_h / out_h - 0.5
    x = (np.arange(out_w) + 0.5) * in_w / out_w - 0.5
    y0 = np.clip(np.floor(y).astype(np.int64), 0, in_h - 1)
    x0 = np.clip(np.floor(x).astype(np.int64), 0, in_w - 1)
    y1 = np.clip(y0 + 1, 0, in_h - 1)
    x1 = np.clip(x0 + 1, 0, in_w - 1)
    wy = np.clip(y - y0, 0.0, 1.0)[:, None]
    wx = np.clip(x - x0, 0.0, 1.0)[None, :]
    a = img[np.ix_(y0, x0)]
    b = img[np.ix_(y0, x1)]
    c = img[np.ix_(y1, x0)]
    d = img[np.ix_(y1, x1)]
    return a * (1 - wy) * (1 - wx) + b * (1 - wy) * wx + c * wy * (1 - wx) + d * wy * wx


class NoopResetEnv(Wrapper):
    """Random number of no-ops after reset (atari_wrapper.py:19-51)."""

    def __init__(self, env: Env, noop_max: int = 30, noop_action: int = 0):
        super().__init__(env)
        self.noop_max = noop_max
        self.noop_action = noop_action
        self._rng = np.random.default_rng()

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        obs, info = self.env.reset(seed=seed)
        noops = int(self._rng.integers(1, self.noop_max + 1))
        for _ in range(noops):
            obs, _, term, trunc, info = self.env.step(self.noop_action)
            if term or trunc:
                obs, info = self.env.reset()
        return obs, info


class MaxAndSkipEnv(Wrapper):
    """Repeat action ``skip`` times; obs = max of last two frames
    (atari_wrapper.py:54-89)."""

    def __init__(self, env: Env, skip: int = 4):
        super().__init__(env)
        self._skip = skip
        shape = env.observation_space.shape
        self._buf = np.zeros((2, *shape), dtype=env.observation_space.dtype)

    def step(self, action):
        total = 0.0
        term = trunc = False
        info = {}
        for i in range(self._skip):
            obs, r, term, trunc, info = self.env.step(action)
            if i == self._skip - 2:
                self._buf[0] = obs
            if i == self._skip - 1:
                self._buf[1] = obs
            total += r
            if term or trunc:
                break
        return self._buf.max(axis=0), total, term, trunc, info


class EpisodicLifeEnv(Wrapper):
    """End episode on life loss, reset only on true game over
    (atari_wrapper.py:92-147).  Uses ``info['lives']`` when present."""

    def __init__(self, env: Env):
        super().__init__(env)
        self.lives = 0
        self.was_real_done = True

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self.was_real_done = term or trunc
        lives = info.get("lives", 0)
        if 0 < lives < self.lives:
            term = True
        self.lives = lives
        return obs, r, term, trunc, info

    def reset(self, seed: Optional[int] = None):
        if self.was_real_done:
            obs, info = self.env.reset(seed=seed)
        else:
            obs, _, _, _, info = self.env.step(0)
        self.lives = info.get("lives", 0)
        return obs, info


class FireResetEnv(Wrapper):
    """Press FIRE after reset for envs that need it (atari_wrapper.py:150-166).
    ``fire_action`` defaults to 1 (ALE FIRE)."""

    def __init__(self, env: Env, fire_action: int = 1):
        super().__init__(env)
        self.fire_action = fire_action

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        obs, _, term, trunc, info = self.env.step(self.fire_action)
        if term or trunc:
            obs, info = self.env.reset()
        return obs, info


class WarpFrame(Wrapper):
    """Grayscale + resize to 84×84 (atari_wrapper.py:169-189)."""

    def __init__(self, env: Env, width: int = 84, height: int = 84):
        super().__init__(env)
        self.width, self.height = width, height
        self.observation_space = Box(0, 255, shape=(height, width), dtype=np.uint8)

    def _warp(self, obs: np.ndarray) -> np.ndarray:
        if obs.ndim == 3 and obs.shape[-1] == 3:
            obs = _rgb_to_gray(obs)
        return np.clip(_resize_bilinear(obs.astype(np.float32), self.height,
                                        self.width), 0, 255).astype(np.uint8)

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        return self._warp(obs), info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return self._warp(obs), r, term, trunc, info


class ScaledFloatFrame(Wrapper):
    """uint8 → float32 / 255 (atari_wrapper.py:192-212).  NOT used in the
    hot path — the MI355X design ships uint8 and normalizes on-device."""

    def __init__(self, env: Env):
        super().__init__(env)
        self.observation_space = Box(0.0, 1.0, shape=env.observation_space.shape,
                                     dtype=np.float32)

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        return np.asarray(obs, dtype=np.float32) / 255.0, info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return np.asarray(obs, dtype=np.float32) / 255.0, r, term, trunc, info


class ClipRewardEnv(Wrapper):
    """Reward → sign(reward) (atari_wrapper.py:215-230)."""

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return obs, float(np.sign(r)), term, trunc, info


class FrameStack(Wrapper):
    """Stack last ``n`` frames channel-first → [n,H,W] (atari_wrapper.py:233-274)."""

    def __init__(self, env: Env, n: int = 4):
        super().__init__(env)
        self.n = n
        shape = env.observation_space.shape
        self.frames = collections.deque(maxlen=n)
        self.observation_space = Box(0, 255, shape=(n, *shape),
                                     dtype=env.observation_space.dtype)

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        for _ in range(self.n):
            self.frames.append(obs)
        return np.stack(self.frames), info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self.frames.append(obs)
        return np.stack(self.frames), r, term, trunc, info


def wrap_deepmind(env: Env, *, episode_life: bool = True, clip_rewards: bool = True,
                  frame_stack: int = 4, scale: bool = False, warp: bool = True,
                  noop_max: int = 30, skip: int = 4, fire_reset: bool = False) -> Env:
    """Compose the DeepMind stack (atari_wrapper.py:277-311)."""
    if noop_max > 0:
        env = NoopResetEnv(env, noop_max=noop_max)
    if skip > 1:
        env = MaxAndSkipEnv(env, skip=skip)
    if episode_life:
        env = EpisodicLifeEnv(env)
    if fire_reset:
        env = FireResetEnv(env)
    if warp:
        env = WarpFrame(env)
    if scale:
        env = ScaledFloatFrame(env)
    if clip_rewards:
        env = ClipRewardEnv(env)
    if frame_stack > 1:
        env = FrameStack(env, frame_stack)
    return env
