"""DQN agent on the MI355X learner stack.

Reference semantics (dqn_agent.py:19-233): double-DQN target, (the
reference's) MSE TD loss, grad clip, soft target update every
``target_update_frequency`` learner steps, linear ε-decay, Accelerate DDP.
MI355X design:

- online/target nets are flat-param pairs; the TD target + loss + IS
  weighting + |TD| priorities run in ONE fused HIP kernel
  (:func:`scalerl_amd.ops.fused_td_loss`);
- gradient sync is one RCCL all-reduce on the flat grad;
- soft target update is the fused polyak kernel over the flat pairs;
- optional PER / n-step via the device-resident buffers in
  :mod:`scalerl_amd.data`.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

from ..config import DQNArguments
from ..models.mlp import QNet
from ..ops import FusedAdam, clip_grad_norm_, fused_polyak_, fused_td_loss
from ..parallel import FlatParams, all_reduce_flat
from ..utils.checkpoint import load_agent_checkpoint, save_agent_checkpoint
from ..utils.schedulers import LinearDecayScheduler
from .base import BaseAgent


class DQNAgent(BaseAgent):
    def __init__(self, args: DQNArguments, obs_dim: int, action_dim: int,
                 device: str = "cpu"):
        super().__init__(args)
        self.device = torch.device(device)
        self.action_dim = action_dim
        self.categorical = args.categorical_dqn
        self.noisy = args.noisy_dqn

        def build():
            if self.categorical:
                from ..models.noisy import CategoricalQNet
                return CategoricalQNet(obs_dim, action_dim, args.hidden_dim,
                                       num_atoms=args.num_atoms,
                                       v_min=args.v_min, v_max=args.v_max,
                                       noisy=self.noisy)
            return QNet(obs_dim, action_dim, args.hidden_dim,
                        dueling=args.dueling_dqn, noisy=self.noisy)

        self.model = build().to(self.device)
        self.target_model = build().to(self.device)
        self.target_model.load_state_dict(self.model.state_dict())
        self.flat = FlatParams(self.model, device=self.device)
        self.target_flat = FlatParams(self.target_model, device=self.device)
        self.optimizer = FusedAdam(self.flat.flat, lr=args.learning_rate)
        self.eps_sched = LinearDecayScheduler(
            args.eps_greedy_start, args.eps_greedy_end, args.eps_decay_steps)
        self.eps = args.eps_greedy_start
        self.per_beta = args.per_beta
        self.rng = np.random.default_rng(args.seed)

    @torch.no_grad()
    def get_action(self, obs: np.ndarray) -> np.ndarray:
        """ε-greedy batch action (dqn_agent.py:90-112); with NoisyNet the
        exploration comes from resampled parameter noise instead."""
        obs = np.atleast_2d(obs)
        n = obs.shape[0]
        if self.noisy:
            self.model.train()
            from ..models.noisy import reset_noise
            reset_noise(self.model)
            t = torch.as_tensor(obs, dtype=torch.float32, device=self.device)
            return self.model(t).argmax(dim=-1).cpu().numpy()
        self.eps = self.eps_sched.step(n)
        greedy = self.predict(obs)
        explore = self.rng.random(n) < self.eps
        random_actions = self.rng.integers(self.action_dim, size=n)
        return np.where(explore, random_actions, greedy)

    @torch.no_grad()
    def predict(self, obs: np.ndarray) -> np.ndarray:
        self.model.eval()
        t = torch.as_tensor(np.atleast_2d(obs), dtype=torch.float32,
                            device=self.device)
        return self.model(t).argmax(dim=-1).cpu().numpy()

    def learn(self, batch: Dict[str, torch.Tensor],
              replay_size: int = 0) -> Dict[str, float]:
        args = self.args
        obs = batch["obs"].to(self.device, torch.float32)
        next_obs = batch["next_obs"].to(self.device, torch.float32)
        actions = batch["action"].to(self.device)
        rewards = batch["reward"].to(self.device)
        discounts = batch["discount"].to(self.device)

        self.flat.flat_grad.zero_()
        if not self.categorical:
            self.model.train()
            q = self.model(obs)
            with torch.no_grad():
                q_next_target = self.target_model(next_obs)
                q_next_online = self.model(next_obs) if args.double_dqn else None

        prios = batch.get("priorities")
        p_total = p_min = None
        if prios is not None and hasattr(self, "_per_stats"):
            p_total, p_min = self._per_stats
        if self.categorical:
            from ..models.noisy import c51_loss
            from ..ops import per_is_weights
            weights = None
            if prios is not None:
                weights = per_is_weights(prios, p_total, p_min, replay_size,
                                         self.per_beta)
            self.model.train()
            loss, td_abs = c51_loss(self.model, self.target_model, obs,
                                    actions, rewards, discounts, next_obs,
                                    double=args.double_dqn, weights=weights)
        else:
            loss, td_abs = fused_td_loss(
                q, q_next_online, q_next_target, actions, rewards, discounts,
                prios=prios, p_total=p_total, p_min=p_min, beta=self.per_beta,
                replay_size=replay_size, huber=False)
        loss.backward()
        all_reduce_flat(self.flat.flat_grad, average=True)
        if args.max_grad_norm > 0:
            clip_grad_norm_(self.flat.flat_grad, args.max_grad_norm)
        self.optimizer.step(self.flat.flat_grad)

        self.global_update_step += 1
        if args.soft_update_tau > 0:
            if self.global_update_step % args.target_update_frequency == 0:
                fused_polyak_(self.target_flat.flat, self.flat.flat,
                              args.soft_update_tau)
        elif self.global_update_step % args.target_update_frequency == 0:
            self.target_flat.flat.copy_(self.flat.flat)

        return {"loss": loss.detach(), "td_abs": td_abs, "eps": self.eps}

    def set_per_stats(self, p_total, p_min) -> None:
        self._per_stats = (p_total, p_min)

    def save_checkpoint(self, path: str) -> None:
        """Agent-format checkpoint (dqn_agent.py:210-222 keys)."""
        save_agent_checkpoint(path, actor=self.model,
                              actor_target=self.target_model,
                              optimizer=None,
                              extra={"optimizer_state_dict":
                                     self.optimizer.state_dict()})

    def load_checkpoint(self, path: str) -> None:
        ckpt = load_agent_checkpoint(path, actor=self.model,
                                     actor_target=self.target_model,
                                     map_location=self.device)
        if ckpt.get("optimizer_state_dict"):
            self.optimizer.load_state_dict(ckpt["optimizer_state_dict"])
