"""Generic off-policy training loop (parity with
``scalerl/trainer/off_policy.py:21-323``: replay setup incl. PER/n-step,
warmup gating, train_frequency, episodic stats, periodic eval, final
checkpoint), re-homed on this framework's vectorized envs + device-resident
replay."""

from __future__ import annotations

import time
from typing import Dict, Optional

import numpy as np
import torch

from ..data import MultiStepReplayBuffer, PrioritizedReplayBuffer, ReplayBuffer
from ..envs.vec_env import EpisodeMetrics, make_vect_envs
from .base import BaseTrainer


class OffPolicyTrainer(BaseTrainer):
    def __init__(self, args, agent, device: str = "cpu"):
        env = make_vect_envs(args.env_id, args.num_envs, seed=args.seed)
        test_env = make_vect_envs(args.env_id, 1, seed=args.seed + 10_000)
        super().__init__(args, env, test_env, agent)
        self.device = device
        obs_shape = env.observation_space.shape
        use_per = getattr(args, "use_per", False)
        n_steps = getattr(args, "n_steps", 1)
        if use_per:
            # PER composes with n-step (reference: PrioritizedReplayBuffer
            # subclasses MultiStepReplayBuffer)
            self.buffer = PrioritizedReplayBuffer(
                args.buffer_size, obs_shape, alpha=args.per_alpha,
                n_steps=n_steps, num_envs=args.num_envs,
                device=device, gamma=args.gamma, seed=args.seed)
        elif n_steps > 1:
            self.buffer = MultiStepReplayBuffer(
                args.buffer_size, obs_shape, n_steps=n_steps,
                gamma=args.gamma, num_envs=args.num_envs, device=device,
                seed=args.seed)
        else:
            self.buffer = ReplayBuffer(args.buffer_size, obs_shape,
                                       device=device, gamma=args.gamma,
                                       seed=args.seed)
        self.use_per = use_per
        self.n_steps = n_steps
        self.metrics = EpisodeMetrics(args.num_envs)
        self.global_step = 0
        self.grad_steps = 0

    def store_experience(self, obs, action, reward, next_obs, done) -> None:
        if self.n_steps > 1:  # n-step folds per env (PER or uniform)
            for i in range(len(action)):
                self.buffer.add(obs[i], action[i], float(reward[i]),
                                next_obs[i], float(done[i]), env_id=i)
        else:
            self.buffer.add_batch(
                torch.as_tensor(obs), torch.as_tensor(action),
                torch.as_tensor(reward, dtype=torch.float32),
                torch.as_tensor(next_obs),
                torch.as_tensor(done, dtype=torch.float32))

    def train_step(self) -> Optional[Dict[str, float]]:
        args = self.args
        if len(self.buffer) < args.warmup_learn_steps:
            return None
        stats = None
        for _ in range(args.learner_update_times):
            if self.use_per:
                batch, idx, prio, p_total, p_min = \
                    self.buffer.sample_with_priorities(args.batch_size)
                batch = dict(batch, priorities=prio)
                self.agent.set_per_stats(p_total, p_min)
                stats = self.agent.learn(batch, replay_size=len(self.buffer))
                self.buffer.update_priorities(idx, stats["td_abs"])
            else:
                batch = self.buffer.sample(args.batch_size)
                stats = self.agent.learn(batch, replay_size=len(self.buffer))
            self.grad_steps += 1
        return stats

    def run_evaluate_episodes(self, n_episodes: int) -> Dict[str, float]:
        returns = []
        for _ in range(n_episodes):
            env = self.test_env.envs[0]
            obs, _ = env.reset()
            total, done = 0.0, False
            while not done:
                action = int(self.agent.predict(obs[None])[0])
                obs, r, term, trunc, _ = env.step(action)
                total += r
                done = term or trunc
            returns.append(total)
        return {"reward_mean": float(np.mean(returns)),
                "reward_std": float(np.std(returns)),
                "length_mean": 0.0}

    def run(self) -> Dict[str, float]:
        args = self.args
        obs = self.train_env.reset(seed=args.seed)
        t0 = time.time()
        last_stats: Dict[str, float] = {}
        episodes = 0
        while self.global_step < args.max_train_steps:
            actions = self.agent.get_action(obs)
            next_obs, rewards, dones = self.train_env.step(actions)
            self.store_experience(obs, actions, rewards, next_obs, dones)
            self.metrics.add(rewards, dones)
            obs = next_obs
            self.global_step += args.num_envs
            if self.global_step % max(args.train_frequency, args.num_envs) < args.num_envs:
                s = self.train_step()
                if s is not None:
                    last_stats = {k: float(v) for k, v in s.items()
                                  if k != "td_abs"}
            rets, lens = self.metrics.pop()
            if rets:
                episodes += len(rets)
                fps = self.global_step / (time.time() - t0)
                data = dict(last_stats, reward=float(np.mean(rets)),
                            fps=fps, rpm=len(self.buffer))
                self.log_train(data, self.global_step)
                if self.is_main_process and episodes % args.train_log_interval == 0:
                    self.text_logger.info(
                        f"step {self.global_step} ep {episodes} "
                        f"ret {np.mean(rets):.1f} fps {fps:,.0f} "
                        + " ".join(f"{k}={v:.4f}" for k, v in last_stats.items()))
                if self.is_main_process and episodes % args.test_log_interval == 0:
                    ev = self.run_evaluate_episodes(args.eval_episodes)
                    self.log_test(ev, self.global_step)
        if args.save_model and self.is_main_process:
            import os
            self.agent.save_checkpoint(os.path.join(self.model_dir,
                                                    "checkpoint.pth"))
        return last_stats
