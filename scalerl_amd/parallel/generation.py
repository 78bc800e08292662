"""Episode generator for remote/multi-agent rollout collection.

Capability parity with the reference's HandyRL-heritage generator
(``scalerl/hpc/generation.py:16-183``): per-player policy inference with
legal-action masking (:109-121), discounted-return folding (:142-147), and
bz2+pickle chunked episode encoding for the TCP control plane (:150-162).
The return fold runs through :func:`scalerl_amd.ops.discounted_returns`
(the same scan kernel family the learners use)."""

from __future__ import annotations

import bz2
import pickle
from typing import Any, Callable, Dict, List, Optional

import numpy as np
import torch

from ..envs.multi_agent import MultiAgentEnv
from ..ops import discounted_returns


def masked_action_probs(logits: torch.Tensor,
                        legal_mask: Optional[np.ndarray]) -> torch.Tensor:
    """Softmax restricted to legal actions (generation.py:109-121)."""
    if legal_mask is not None:
        mask = torch.as_tensor(legal_mask, dtype=torch.bool,
                               device=logits.device)
        logits = logits.masked_fill(~mask, float("-inf"))
    return torch.softmax(logits, dim=-1)


class EpisodeGenerator:
    """Rolls out one episode per call with per-player policies."""

    def __init__(self, env: MultiAgentEnv,
                 policies: Dict[str, Callable[[torch.Tensor], torch.Tensor]],
                 gamma: float = 0.99, compress_steps: int = 64,
                 seed: Optional[int] = None):
        self.env = env
        self.policies = policies
        self.gamma = gamma
        self.compress_steps = compress_steps
        self._rng = np.random.default_rng(seed)

    @torch.no_grad()
    def execute(self, max_steps: int = 10_000) -> Dict[str, Any]:
        env = self.env
        obs, infos = env.reset()
        steps: List[Dict[str, Any]] = []
        rewards: Dict[str, List[float]] = {a: [] for a in env.agents}
        t = 0
        while t < max_steps:
            actions = {}
            for agent in env.agents:
                logits = self.policies[agent](
                    torch.as_tensor(obs[agent], dtype=torch.float32))
                probs = masked_action_probs(
                    logits, infos.get(agent, {}).get("legal_actions"))
                actions[agent] = int(torch.multinomial(probs, 1).item())
            obs, rew, term, trunc, infos = env.step(actions)
            steps.append({"actions": actions, "rewards": dict(rew)})
            for a in env.agents:
                rewards[a].append(float(rew[a]))
            t += 1
            if all(term[a] or trunc[a] for a in env.agents):
                break

        # discounted return fold per agent (generation.py:142-147)
        returns = {}
        for a in env.agents:
            r = torch.tensor(rewards[a]).unsqueeze(1)
            d = torch.full_like(r, self.gamma)
            returns[a] = discounted_returns(r, d).squeeze(1).tolist()

        # bz2-compressed chunks of `compress_steps` steps (:150-162)
        chunks = []
        for i in range(0, len(steps), self.compress_steps):
            blob = pickle.dumps(steps[i:i + self.compress_steps],
                                protocol=pickle.HIGHEST_PROTOCOL)
            chunks.append(bz2.compress(blob))
        return {"length": len(steps), "returns": returns,
                "chunks": chunks, "agents": list(env.agents)}

    @staticmethod
    def decode_chunks(chunks: List[bytes]) -> List[Dict[str, Any]]:
        out: List[Dict[str, Any]] = []
        for c in chunks:
            out.extend(pickle.loads(bz2.decompress(c)))
        return out
