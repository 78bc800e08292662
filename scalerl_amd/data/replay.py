"""Replay buffers.

Reimplements the semantics of the reference's replay layer
(``scalerl/data/replay_buffer.py:10-381``: uniform deque buffer, n-step
fold-at-insert buffer, proportional prioritized buffer over segment trees)
with an MI355X-first storage design:

- Structure-of-Arrays ring storage in preallocated torch tensors — on the
  LEARNER'S DEVICE when one is given (288 GB HBM3E holds a 1M-transition
  Atari PER buffer ~30 GB with room to spare), CPU
  otherwise;
- batched vectorized inserts (the actor→learner ingest path hands whole
  chunks, not single transitions);
- the prioritized variant keeps its sum tree on the same device and
  updates/samples it with the HIP kernels (:class:`scalerl_amd.ops.SumTree`);
  min-priority for the IS normalizer is a flat reduce, not a second tree;
- n-step folding at ingest uses :func:`scalerl_amd.ops.nstep_fold` on
  device.

Single-transition ``add`` (the reference's API) is kept for parity and
tests; it simply wraps the batched path.
"""

from __future__ import annotations

import collections
from typing import Dict, Optional, Tuple

import torch

Transition = collections.namedtuple(
    "Transition", ["obs", "action", "reward", "next_obs", "done"])


class ReplayBuffer:
    """Uniform ring replay with SoA tensor storage."""

    def __init__(self, capacity: int, obs_shape: Tuple[int, ...],
                 obs_dtype=torch.float32, action_dtype=torch.int64,
                 device: str = "cpu", seed: Optional[int] = None,
                 gamma: float = 0.99):
        self.capacity = capacity
        self.gamma = gamma
        self.device = torch.device(device)
        self.obs = torch.empty((capacity, *obs_shape), dtype=obs_dtype,
                               device=self.device)
        self.next_obs = torch.empty_like(self.obs)
        self.action = torch.empty(capacity, dtype=action_dtype,
                                  device=self.device)
        self.reward = torch.empty(capacity, dtype=torch.float32,
                                  device=self.device)
        self.done = torch.empty(capacity, dtype=torch.float32,
                                device=self.device)
        # bootstrap multiplier for the TD target: gamma^m * (1 - done_m)
        # (m = 1 for 1-step; the n-step buffer stores its fold's m)
        self.discount = torch.empty(capacity, dtype=torch.float32,
                                    device=self.device)
        self.cursor = 0
        self.size = 0
        self.generator = None
        if seed is not None:
            self.generator = torch.Generator(device=self.device)
            self.generator.manual_seed(seed)

    def __len__(self) -> int:
        return self.size

    def _slots_for(self, n: int) -> torch.Tensor:
        idx = (torch.arange(n) + self.cursor) % self.capacity
        self.cursor = int((self.cursor + n) % self.capacity)
        self.size = min(self.size + n, self.capacity)
        return idx.to(self.device)

    def add_batch(self, obs, action, reward, next_obs, done,
                  discount=None) -> torch.Tensor:
        """Insert a chunk of transitions; returns the slot indices used."""
        n = obs.shape[0]
        idx = self._slots_for(n)
        self.obs[idx] = obs.to(self.device, self.obs.dtype)
        self.next_obs[idx] = next_obs.to(self.device, self.obs.dtype)
        self.action[idx] = action.to(self.device, self.action.dtype)
        self.reward[idx] = reward.to(self.device, torch.float32)
        done = done.to(self.device, torch.float32)
        self.done[idx] = done
        if discount is None:
            discount = self.gamma * (1.0 - done)
        self.discount[idx] = discount.to(self.device, torch.float32)
        return idx

    def add(self, obs, action, reward, next_obs, done) -> torch.Tensor:
        """Single transition (reference API, replay_buffer.py:98-116)."""
        def up(x, dtype=None):
            t = torch.as_tensor(x)
            return t.unsqueeze(0)
        return self.add_batch(up(obs), torch.as_tensor([action]),
                              torch.as_tensor([float(reward)]),
                              up(next_obs),
                              torch.as_tensor([float(done)]))

    def sample_indices(self, batch_size: int) -> torch.Tensor:
        return torch.randint(0, self.size, (batch_size,),
                             generator=self.generator, device=self.device)

    def gather(self, idx: torch.Tensor) -> Dict[str, torch.Tensor]:
        return {"obs": self.obs[idx], "action": self.action[idx],
                "reward": self.reward[idx], "next_obs": self.next_obs[idx],
                "done": self.done[idx], "discount": self.discount[idx]}

    def sample(self, batch_size: int) -> Dict[str, torch.Tensor]:
        return self.gather(self.sample_indices(batch_size))

    def state_dict(self) -> Dict:
        return {k: getattr(self, k) for k in
                ("obs", "next_obs", "action", "reward", "done", "discount",
                 "cursor", "size")}

    def load_state_dict(self, sd: Dict) -> None:
        for k in ("obs", "next_obs", "action", "reward", "done", "discount"):
            getattr(self, k).copy_(sd[k])
        self.cursor = int(sd["cursor"])
        self.size = int(sd["size"])


class MultiStepReplayBuffer(ReplayBuffer):
    """n-step returns folded at insert (replay_buffer.py:132-273 semantics):
    the stored transition for time t carries sum_{k<m} gamma^k r_{t+k}, the
    observation m steps ahead, and the done flag of the window; m stops at
    the first terminal.

    Chunked ingest: callers hand [T,B] trajectory chunks;
    :func:`scalerl_amd.ops.nstep_fold` does the window math on device.
    Single-env ``add`` keeps per-env pending deques (the reference's way).
    """

    def __init__(self, capacity: int, obs_shape, n_steps: int = 3,
                 gamma: float = 0.99, num_envs: int = 1, **kw):
        super().__init__(capacity, obs_shape, **kw)
        self.n_steps = n_steps
        self.gamma = gamma
        self._pending = [collections.deque(maxlen=n_steps)
                         for _ in range(num_envs)]

    def add(self, obs, action, reward, next_obs, done, env_id: int = 0):
        """Single-transition insert with per-env n-step folding."""
        q = self._pending[env_id]
        q.append((obs, action, float(reward), next_obs, float(done)))
        out = None

        def fold():
            folded_r, g = 0.0, 1.0
            for (_, _, r, no, d) in q:
                folded_r += g * r
                g *= self.gamma
                last_next, last_done = no, d
                if d:
                    break
            disc = g * (1.0 - last_done)
            return folded_r, last_next, last_done, disc

        def insert_front():
            folded_r, last_next, last_done, disc = fold()
            o0, a0 = q[0][0], q[0][1]
            i = self.add_batch(
                torch.as_tensor(o0).unsqueeze(0), torch.as_tensor([a0]),
                torch.as_tensor([folded_r]),
                torch.as_tensor(last_next).unsqueeze(0),
                torch.as_tensor([float(last_done)]),
                discount=torch.as_tensor([disc]))
            q.popleft()
            return i

        if len(q) == self.n_steps or done:
            out = insert_front()
            if done:
                while q:  # flush remaining shorter windows
                    insert_front()
        return out

    def add_chunk(self, obs, action, reward, done, bootstrap_obs):
        """Vectorized ingest of a [T,B] trajectory chunk (device path).

        obs [T,B,...]; bootstrap_obs [n_steps,B,...] = the observations
        following the chunk (for next_obs of the tail rows).
        """
        from ..ops import nstep_fold
        T, B = reward.shape
        folded_r, folded_d, steps = nstep_fold(
            reward.to(self.device), done.to(self.device, torch.float32),
            self.gamma, self.n_steps)
        # next_obs index for (t, b): t + m (within chunk or bootstrap rows)
        all_obs = torch.cat([obs.to(self.device), bootstrap_obs.to(self.device)], 0)
        t_idx = (torch.arange(T, device=self.device).unsqueeze(1)
                 + steps.to(self.device).long())
        b_idx = torch.arange(B, device=self.device).unsqueeze(0).expand(T, B)
        next_obs = all_obs[t_idx.reshape(-1), b_idx.reshape(-1)]
        disc = (self.gamma ** steps.to(self.device).float()) * (1.0 - folded_d)
        return self.add_batch(
            obs.reshape(T * B, *obs.shape[2:]),
            action.reshape(-1), folded_r.reshape(-1), next_obs,
            folded_d.reshape(-1), discount=disc.reshape(-1))


class PrioritizedReplayBuffer(MultiStepReplayBuffer):
    """Proportional PER (replay_buffer.py:276-381 semantics) over the
    device sum tree.  Priorities stored as p^alpha; new transitions get
    max_priority^alpha; stratified sampling; IS weights computed in the
    fused TD-loss kernel from (prio, total, min).

    Inherits the n-step fold-at-insert front end (n_steps=1 → plain 1-step)
    so PER and n-step compose, as in the reference where
    PrioritizedReplayBuffer subclasses MultiStepReplayBuffer."""

    def __init__(self, capacity: int, obs_shape, alpha: float = 0.6,
                 n_steps: int = 1, **kw):
        super().__init__(capacity, obs_shape, n_steps=n_steps, **kw)
        from ..ops import SumTree
        self.alpha = alpha
        self.tree = SumTree(capacity, device=self.device)
        self.max_priority = 1.0

    def add_batch(self, obs, action, reward, next_obs, done,
                  discount=None,
                  priorities: Optional[torch.Tensor] = None) -> torch.Tensor:
        idx = super().add_batch(obs, action, reward, next_obs, done,
                                discount=discount)
        if priorities is None:
            p = torch.full((idx.numel(),), self.max_priority ** self.alpha,
                           device=self.device)
        else:
            p = priorities.to(self.device) ** self.alpha
        self.tree.update(idx, p, max_idx=self.size)
        return idx

    def sample_with_priorities(self, batch_size: int):
        """→ (batch dict, idx, prio, p_total, p_min) — all device-resident,
        ready for the fused TD-loss kernel (no host round trip)."""
        idx, prio = self.tree.sample(batch_size, generator=self.generator)
        batch = self.gather(idx)
        return batch, idx, prio, self.tree.total, self.tree.min_leaf()

    def update_priorities(self, idx: torch.Tensor, td_abs: torch.Tensor,
                          eps: float = 1e-6) -> None:
        p = (td_abs.to(self.device) + eps)
        self.tree.update(idx, p ** self.alpha, max_idx=self.size)
        m = float(td_abs.max()) + eps
        self.max_priority = max(self.max_priority, m)

    def sample(self, batch_size: int) -> Dict[str, torch.Tensor]:
        batch, idx, prio, total, pmin = self.sample_with_priorities(batch_size)
        batch.update(indices=idx, priorities=prio)
        return batch

    def state_dict(self) -> Dict:
        sd = super().state_dict()
        sd["tree"] = self.tree.state_dict()
        sd["max_priority"] = self.max_priority
        return sd

    def load_state_dict(self, sd: Dict) -> None:
        super().load_state_dict(sd)
        self.tree.load_state_dict(sd["tree"])
        self.max_priority = float(sd["max_priority"])
