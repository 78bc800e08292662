"""Multi-agent environment layer.

Capability parity with the reference's PettingZoo stack
(``scalerl/envs/vector/pz_vec_env.py:4-92`` dict-of-agent batched
reshaping, ``pettingzoo_wrappers.py:9-64`` reset-when-all-done), defined
against a self-contained protocol because pettingzoo is not in this image;
:class:`PettingZooAdapter` bridges real PettingZoo parallel envs when the
package is importable.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np

from .base import Box, Discrete


class MultiAgentEnv:
    """Parallel multi-agent API: dict-keyed obs/action/reward per agent."""

    agents: List[str]
    observation_spaces: Dict[str, Any]
    action_spaces: Dict[str, Any]

    def reset(self, seed: Optional[int] = None):
        raise NotImplementedError

    def step(self, actions: Dict[str, Any]):
        """→ obs, rewards, terminations, truncations, infos (all dicts)."""
        raise NotImplementedError

    def close(self):
        pass


class SyntheticMultiAgentEnv(MultiAgentEnv):
    """Two-player synthetic game with legal-action masks: each agent is
    rewarded for matching the shared state modulo its action count; some
    actions are masked each step (the hpc Generator's legal-action
    machinery, generation.py:109-121, needs masks to be exercised)."""

    def __init__(self, num_agents: int = 2, num_actions: int = 4,
                 obs_dim: int = 8, episode_length: int = 100,
                 seed: Optional[int] = None):
        self.agents = [f"player_{i}" for i in range(num_agents)]
        self.num_actions = num_actions
        self.obs_dim = obs_dim
        self.episode_length = episode_length
        self.observation_spaces = {a: Box(-1, 1, shape=(obs_dim,))
                                   for a in self.agents}
        self.action_spaces = {a: Discrete(num_actions) for a in self.agents}
        self._rng = np.random.default_rng(seed)
        self._state = 0
        self._steps = 0

    def _obs(self) -> Dict[str, np.ndarray]:
        base = np.zeros(self.obs_dim, dtype=np.float32)
        base[self._state % self.obs_dim] = 1.0
        return {a: base.copy() for a in self.agents}

    def legal_actions(self, agent: str) -> np.ndarray:
        mask = np.ones(self.num_actions, dtype=bool)
        mask[(self._state + self.agents.index(agent)) % self.num_actions] = \
            self._steps % 2 == 0  # alternate masking pattern
        return mask

    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self._rng = np.random.default_rng(seed)
        self._state = int(self._rng.integers(64))
        self._steps = 0
        infos = {a: {"legal_actions": self.legal_actions(a)}
                 for a in self.agents}
        return self._obs(), infos

    def step(self, actions: Dict[str, int]):
        rewards = {a: float(int(actions[a]) == self._state % self.num_actions)
                   for a in self.agents}
        self._state = int((self._state * 3 + sum(actions.values()) + 1) % 64)
        self._steps += 1
        done = self._steps >= self.episode_length
        obs = self._obs()
        terms = {a: False for a in self.agents}
        truncs = {a: done for a in self.agents}
        infos = {a: {"legal_actions": self.legal_actions(a)}
                 for a in self.agents}
        return obs, rewards, terms, truncs, infos


class MultiAgentAutoReset(MultiAgentEnv):
    """Reset when ALL agents are done (pettingzoo_wrappers.py:32-44)."""

    def __init__(self, env: MultiAgentEnv):
        self.env = env
        self.agents = env.agents
        self.observation_spaces = env.observation_spaces
        self.action_spaces = env.action_spaces

    def reset(self, seed: Optional[int] = None):
        return self.env.reset(seed=seed)

    def step(self, actions):
        obs, rew, term, trunc, info = self.env.step(actions)
        if all(term[a] or trunc[a] for a in self.env.agents):
            obs, info = self.env.reset()
        return obs, rew, term, trunc, info

    def close(self):
        self.env.close()


class MultiAgentVecEnv:
    """Batches N multi-agent envs: actions arrive as
    ``{agent: [N]}``, observations return as ``{agent: [N, obs...]}``
    (pz_vec_env.py:53-68 reshaping semantics, in-process)."""

    def __init__(self, env_fns):
        self.envs: List[MultiAgentEnv] = [MultiAgentAutoReset(fn())
                                          for fn in env_fns]
        self.num_envs = len(self.envs)
        self.agents = self.envs[0].agents
        self.observation_spaces = self.envs[0].observation_spaces
        self.action_spaces = self.envs[0].action_spaces

    def reset(self, seed: Optional[int] = None):
        per_env = [e.reset(None if seed is None else seed + i)[0]
                   for i, e in enumerate(self.envs)]
        return {a: np.stack([o[a] for o in per_env]) for a in self.agents}

    def step(self, actions: Dict[str, np.ndarray]):
        obs_l, rew_l, done_l = [], [], []
        for i, e in enumerate(self.envs):
            acts = {a: actions[a][i] for a in self.agents}
            obs, rew, term, trunc, _ = e.step(acts)
            obs_l.append(obs)
            rew_l.append(rew)
            done_l.append({a: term[a] or trunc[a] for a in self.agents})
        batch = lambda key, rows: {a: np.stack([r[a] for r in rows])
                                   for a in self.agents}
        return (batch("obs", obs_l),
                {a: np.array([r[a] for r in rew_l], dtype=np.float32)
                 for a in self.agents},
                {a: np.array([d[a] for d in done_l]) for a in self.agents})

    def close(self):
        for e in self.envs:
            e.close()


class PettingZooAdapter(MultiAgentEnv):
    """Bridge for real PettingZoo parallel envs (import-gated)."""

    def __init__(self, pz_env):
        self._env = pz_env
        self.agents = list(pz_env.possible_agents)
        self.observation_spaces = {a: pz_env.observation_space(a)
                                   for a in self.agents}
        self.action_spaces = {a: pz_env.action_space(a) for a in self.agents}

    def reset(self, seed: Optional[int] = None):
        return self._env.reset(seed=seed)

    def step(self, actions):
        return self._env.step(actions)

    def close(self):
        self._env.close()
