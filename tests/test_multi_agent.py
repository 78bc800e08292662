"""Multi-agent env layer + episode generator."""

import pytest
import numpy as np
import torch

from scalerl_amd.envs.multi_agent import (MultiAgentVecEnv,
                                          SyntheticMultiAgentEnv)
from scalerl_amd.parallel.generation import (EpisodeGenerator,
                                             masked_action_probs)


def test_multi_agent_env_api():
    env = SyntheticMultiAgentEnv(num_agents=2, seed=0)
    obs, infos = env.reset(seed=0)
    assert set(obs) == {"player_0", "player_1"}
    assert infos["player_0"]["legal_actions"].shape == (4,)
    acts = {a: 0 for a in env.agents}
    obs, rew, term, trunc, infos = env.step(acts)
    assert set(rew) == set(obs)


def test_multi_agent_vec_env_batching():
    v = MultiAgentVecEnv([lambda: SyntheticMultiAgentEnv(seed=i)
                          for i in range(3)])
    obs = v.reset(seed=0)
    assert obs["player_0"].shape == (3, 8)
    actions = {a: np.zeros(3, dtype=np.int64) for a in v.agents}
    obs, rew, done = v.step(actions)
    assert rew["player_1"].shape == (3,)
    v.close()


def test_masked_softmax_zeroes_illegal():
    logits = torch.zeros(4)
    mask = np.array([True, False, True, False])
    p = masked_action_probs(logits, mask)
    assert p[1] == 0 and p[3] == 0
    assert abs(float(p.sum()) - 1.0) < 1e-6


def test_episode_generator_roundtrip():
    env = SyntheticMultiAgentEnv(num_agents=2, episode_length=20, seed=0)
    policies = {a: (lambda o: torch.zeros(4)) for a in env.agents}
    gen = EpisodeGenerator(env, policies, gamma=0.5, compress_steps=8)
    ep = gen.execute()
    assert ep["length"] == 20
    assert len(ep["chunks"]) == 3  # 8 + 8 + 4
    steps = EpisodeGenerator.decode_chunks(ep["chunks"])
    assert len(steps) == 20
    # discounted fold sanity: R_0 = r_0 + 0.5 R_1
    r = [s["rewards"]["player_0"] for s in steps]
    R = ep["returns"]["player_0"]
    assert abs(R[0] - (r[0] + 0.5 * R[1])) < 1e-5


def _have_pettingzoo() -> bool:
    try:
        import pettingzoo  # noqa: F401
        return True
    except ImportError:
        return False


@pytest.mark.skipif(not _have_pettingzoo(), reason="pettingzoo not in this "
                    "image (no network); runs where it is installed")
def test_real_pettingzoo_adapter_smoke():
    """PettingZooAdapter over a real parallel PZ env (the in-image tests
    cover the same API against SyntheticMultiAgentEnv fakes)."""
    from pettingzoo.butterfly import pistonball_v6
    from scalerl_amd.envs.multi_agent import PettingZooAdapter
    env = PettingZooAdapter(pistonball_v6.parallel_env())
    obs, _ = env.reset(seed=0)
    assert set(obs) == set(env.agents)
    for _ in range(5):
        actions = {a: env.action_space(a).sample() for a in env.agents}
        obs, rew, term, trunc, _ = env.step(actions)
        assert set(rew) <= set(env.agents) | set(obs)
    env.close()
