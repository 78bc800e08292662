"""Episode statistics + video recording wrappers.

Parity with the reference's ``make_gym_env`` (scalerl/envs/gym_env.py:6-33),
which composes ``gym.wrappers.RecordVideo`` and
``gym.wrappers.RecordEpisodeStatistics``.  This image ships no gymnasium,
no cv2 and no video encoder, so the wrappers are implemented against this
package's Env API:

- :class:`RecordEpisodeStatistics` puts ``info["episode"] = {"r","l","t"}``
  on the terminal step (gymnasium's dict contract);
- :class:`RecordVideo` collects frames from ``env.render()`` (falling back
  to image observations) and writes animated GIFs via PIL — the same
  episode-trigger semantics as gym's wrapper, a different container.
"""

from __future__ import annotations

import os
import time
from typing import Callable, Optional

import numpy as np

from .base import Env


class Wrapper(Env):
    """Pass-through base wrapper."""

    def __init__(self, env: Env):
        self.env = env
        self.observation_space = env.observation_space
        self.action_space = env.action_space
        self.spec_id = getattr(env, "spec_id", "Env")

    def reset(self, seed=None):
        return self.env.reset(seed=seed)

    def step(self, action):
        return self.env.step(action)

    def render(self):
        return self.env.render()

    def close(self):
        self.env.close()

    @property
    def unwrapped(self):
        return self.env.unwrapped


class RecordEpisodeStatistics(Wrapper):
    """gymnasium.wrappers.RecordEpisodeStatistics contract: on the step
    that ends an episode, ``info["episode"] = {"r": return, "l": length,
    "t": elapsed wall seconds}``."""

    def __init__(self, env: Env):
        super().__init__(env)
        self._ret = 0.0
        self._len = 0
        self._t0 = time.perf_counter()

    def reset(self, seed=None):
        obs, info = self.env.reset(seed=seed)
        self._ret, self._len = 0.0, 0
        self._t0 = time.perf_counter()
        return obs, info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        self._ret += float(r)
        self._len += 1
        if term or trunc:
            info = dict(info)
            info["episode"] = {
                "r": self._ret, "l": self._len,
                "t": round(time.perf_counter() - self._t0, 6)}
            self._ret, self._len = 0.0, 0
            self._t0 = time.perf_counter()
        return obs, r, term, trunc, info


def _default_episode_trigger(ep: int) -> bool:
    """gym's capped_cubic_video_schedule: cubes until 1000, then every
    1000th episode."""
    if ep < 1000:
        r = round(ep ** (1.0 / 3.0))
        return r ** 3 == ep
    return ep % 1000 == 0


class RecordVideo(Wrapper):
    """Record episodes as animated GIFs under ``video_dir``.

    Frames come from ``env.render()`` when it returns an array, else from
    image-shaped observations ([H,W], [H,W,C] or [C,H,W] uint8).  Files:
    ``{name_prefix}-episode-{n}.gif``.
    """

    def __init__(self, env: Env, video_dir: str,
                 episode_trigger: Optional[Callable[[int], bool]] = None,
                 name_prefix: str = "rl-video", fps: int = 30):
        super().__init__(env)
        self.video_dir = video_dir
        self.episode_trigger = episode_trigger or _default_episode_trigger
        self.name_prefix = name_prefix
        self.fps = fps
        self.episode_id = -1
        self._frames = []
        self._recording = False
        self.recorded_files = []
        os.makedirs(video_dir, exist_ok=True)

    # -- frame capture -----------------------------------------------------
    def _grab(self, obs) -> Optional[np.ndarray]:
        frame = self.env.render()
        if frame is None:
            a = np.asarray(obs)
            if a.ndim == 2:
                frame = a
            elif a.ndim == 3:
                # [C,H,W] → [H,W,C] for small leading channel dims
                frame = a.transpose(1, 2, 0) if a.shape[0] <= 4 else a
            else:
                return None
        f = np.asarray(frame)
        if f.dtype != np.uint8:
            lo, hi = float(f.min()), float(f.max())
            f = ((f - lo) / (hi - lo + 1e-8) * 255).astype(np.uint8)
        if f.ndim == 3 and f.shape[2] not in (1, 3):
            f = f[..., :1]  # stacked frames: keep the newest-ish plane
        if f.ndim == 3 and f.shape[2] == 1:
            f = f[..., 0]
        return f

    def _flush(self):
        if not self._frames:
            return
        from PIL import Image
        imgs = [Image.fromarray(f) for f in self._frames]
        path = os.path.join(
            self.video_dir, f"{self.name_prefix}-episode-{self.episode_id}.gif")
        imgs[0].save(path, save_all=True, append_images=imgs[1:],
                     duration=max(1, int(1000 / self.fps)), loop=0)
        self.recorded_files.append(path)
        self._frames = []

    # -- env API -----------------------------------------------------------
    def reset(self, seed=None):
        if self._recording:
            self._flush()
        obs, info = self.env.reset(seed=seed)
        self.episode_id += 1
        self._recording = self.episode_trigger(self.episode_id)
        if self._recording:
            f = self._grab(obs)
            self._frames = [f] if f is not None else []
        return obs, info

    def step(self, action):
        obs, r, term, trunc, info = self.env.step(action)
        if self._recording:
            f = self._grab(obs)
            if f is not None:
                self._frames.append(f)
            if term or trunc:
                self._flush()
                self._recording = False
        return obs, r, term, trunc, info

    def close(self):
        if self._recording:
            self._flush()
        self.env.close()
