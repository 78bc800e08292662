"""A3C Atari preprocessing (parity with
``scalerl/algorithms/a3c/utils/atari_env.py:9-122``): 42×42 grayscale
rescale (crop + double downscale + channel mean) and a running mean/std
observation normalizer.  cv2-free (numpy bilinear resample)."""

from __future__ import annotations

from typing import Optional

import numpy as np

from .atari_wrappers import _resize_bilinear, _rgb_to_gray
from .base import Box, Env, Wrapper
from .registry import make_env


class AtariRescale42x42(Wrapper):
    """Crop playfield, downscale 80×80 → 42×42, mean-normalize channels
    → float32 [1, 42, 42] (atari_env.py:9-47 semantics)."""

    def __init__(self, env: Env):
        super().__init__(env)
        self.observation_space = Box(0.0, 1.0, shape=(1, 42, 42),
                                     dtype=np.float32)

    @staticmethod
    def _process(frame: np.ndarray) -> np.ndarray:
        if frame.ndim == 3 and frame.shape[-1] == 3:
            frame = _rgb_to_gray(frame)
        elif frame.ndim == 3:  # stacked frames → use the newest
            frame = frame[-1].astype(np.float32)
        f = frame.astype(np.float32)
        if f.shape[0] > 160:  # ALE 210×160: crop score bars
            f = f[34:194]
        f = _resize_bilinear(f, 80, 80)
        f = _resize_bilinear(f, 42, 42)
        f = f * (1.0 / 255.0)
        return f.reshape(1, 42, 42)

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        return self._process(obs), info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return self._process(obs), r, term, trunc, info


class NormalizedEnv(Wrapper):
    """Running mean/std observation normalizer with debiased warmup
    (atari_env.py:83-122 semantics)."""

    def __init__(self, env: Env):
        super().__init__(env)
        self.state_mean = 0.0
        self.state_std = 0.0
        self.alpha = 0.9999
        self.num_steps = 0

    def _normalize(self, obs: np.ndarray) -> np.ndarray:
        self.num_steps += 1
        self.state_mean = (self.state_mean * self.alpha
                           + obs.mean() * (1 - self.alpha))
        self.state_std = (self.state_std * self.alpha
                          + obs.std() * (1 - self.alpha))
        unbias = 1 - self.alpha ** self.num_steps
        mean = self.state_mean / unbias
        std = self.state_std / unbias
        return (obs - mean) / (std + 1e-8)

    def reset(self, seed: Optional[int] = None):
        obs, info = self.env.reset(seed=seed)
        return self._normalize(np.asarray(obs, dtype=np.float32)), info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        return (self._normalize(np.asarray(obs, dtype=np.float32)),
                r, term, trunc, info)


def create_atari_env(env_id: str, seed: Optional[int] = None) -> Env:
    """42×42 normalized Atari env for A3C (atari_env.py create_atari_env)."""
    env = make_env(env_id, seed=seed)
    return NormalizedEnv(AtariRescale42x42(env))
