"""Env factory (parity with ``scalerl/envs/gym_env.py:6-33`` make_gym_env).

Resolution order for an env id:
1. built-in envs (CartPole-v1, synthetic-atari, synthetic-pointgoal);
2. gymnasium (with ALE), adapted to this package's API, if importable.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np

from .base import Box, Discrete, Env
from .cartpole import CartPoleEnv
from .synthetic import SyntheticAtariEnv, SyntheticPointGoalEnv


class GymAdapter(Env):
    """Wraps a gymnasium env into this package's Env API."""

    def __init__(self, gym_env):
        self._env = gym_env
        self.spec_id = getattr(gym_env.spec, "id", "gym")
        obs_sp, act_sp = gym_env.observation_space, gym_env.action_space
        if hasattr(act_sp, "n"):
            self.action_space = Discrete(act_sp.n)
        else:
            self.action_space = Box(act_sp.low, act_sp.high, dtype=act_sp.dtype)
        self.observation_space = Box(
            getattr(obs_sp, "low", 0), getattr(obs_sp, "high", 255),
            shape=obs_sp.shape, dtype=obs_sp.dtype)

    def reset(self, seed: Optional[int] = None):
        obs, info = self._env.reset(seed=seed)
        return np.asarray(obs), dict(info)

    def step(self, action):
        obs, r, term, trunc, info = self._env.step(action)
        return np.asarray(obs), float(r), bool(term), bool(trunc), dict(info)

    def close(self):
        self._env.close()


_BUILTIN = {
    "CartPole-v1": CartPoleEnv,
    "synthetic-atari": SyntheticAtariEnv,
    "synthetic-pointgoal": SyntheticPointGoalEnv,
}


def make_env(env_id: str, seed: Optional[int] = None,
             env_kwargs: Optional[Dict[str, Any]] = None,
             deepmind_wrap: bool = False) -> Env:
    env_kwargs = env_kwargs or {}
    if env_id in _BUILTIN:
        env = _BUILTIN[env_id](seed=seed, **env_kwargs)
    else:
        try:
            import gymnasium as gym
        except ImportError as e:
            raise ValueError(
                f"env id {env_id!r} is not a built-in and gymnasium is not "
                f"installed (built-ins: {sorted(_BUILTIN)})") from e
        env = GymAdapter(gym.make(env_id, **env_kwargs))
    if deepmind_wrap:
        from .atari_wrappers import wrap_deepmind
        env = wrap_deepmind(env)
    if seed is not None:
        env.action_space.seed(seed)
    return env


def make_gym_env(env_id: str, seed: int = 42, capture_video: bool = False,
                 save_video_dir: str = "work_dir",
                 save_video_name: str = "test",
                 deepmind_wrap: bool = False):
    """Reference ``make_gym_env`` parity (scalerl/envs/gym_env.py:6-33):
    optional video capture + episode-statistics wrapper + action-space
    seeding.  Video is written as animated GIFs (no encoder in the image);
    the directory layout ``{save_video_dir}/{save_video_name}`` matches."""
    from .recording import RecordEpisodeStatistics, RecordVideo
    env = make_env(env_id, seed=seed, deepmind_wrap=deepmind_wrap)
    if capture_video:
        import os
        env = RecordVideo(env, os.path.join(save_video_dir, save_video_name),
                          name_prefix=save_video_name)
    env = RecordEpisodeStatistics(env)
    env.action_space.seed(seed)
    return env
