"""A3C: asynchronous shared-memory actor-critic (hogwild).

Reference semantics reimplemented (parallel_a3c.py:71-513, parallel_ac.py,
share_optim.py): N CPU worker processes around a shared-memory model +
SharedAdam; each worker syncs its local model from the shared one, rolls
out up to ``rollout_steps``, computes the A2C loss (GAE advantages, value
MSE, entropy bonus), writes its gradients into the shared parameters and
steps the shared optimizer; an eval process tracks greedy returns.  The
reference's races (per-process ε schedules, default-arg locks —
SURVEY.md "Broken-as-shipped") are not reproduced: gradient publication
uses a lock, and the step counter is a shared Value.

This is the CPU-parity path (benchmark config 1/2 small scale).  The
GPU-scale A3C/A2C config (16 CPU actors + 1 MI355X learner) runs on the
shared actor-learner runtime: ImpalaTrainer with clip_rho/clip_c at their
defaults degenerates to importance-corrected A2C on near-on-policy data.

Scan math (returns/GAE) goes through :mod:`scalerl_amd.ops` (the same
kernels the GPU learners use; CPU reference path here).
"""

from __future__ import annotations

import time

import numpy as np
import torch
import torch.multiprocessing as mp
import torch.nn.functional as F

from ..config import A3CArguments
from ..envs.registry import make_env
from ..models.mlp import ActorCriticNet
from ..ops import gae as gae_op
from ..utils.checkpoint import save_agent_checkpoint


class SharedAdam(torch.optim.Adam):
    """Adam whose state tensors live in shared memory
    (share_optim.py:9-122 parity; the update itself is torch's)."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0):
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay)
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state[p]
                state["step"] = torch.zeros(1)
                state["exp_avg"] = torch.zeros_like(p.data)
                state["exp_avg_sq"] = torch.zeros_like(p.data)
                state["step"].share_memory_()
                state["exp_avg"].share_memory_()
                state["exp_avg_sq"].share_memory_()

    @torch.no_grad()
    def step(self, closure=None):
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                state["step"] += 1
                b1, b2 = group["betas"]
                state["exp_avg"].mul_(b1).add_(p.grad, alpha=1 - b1)
                state["exp_avg_sq"].mul_(b2).addcmul_(p.grad, p.grad,
                                                      value=1 - b2)
                step = float(state["step"].item())
                bc1 = 1 - b1 ** step
                bc2 = 1 - b2 ** step
                denom = (state["exp_avg_sq"] / bc2).sqrt().add_(group["eps"])
                p.data.addcdiv_(state["exp_avg"] / bc1, denom,
                                value=-group["lr"])


def ensure_shared_grads(local_model, shared_model) -> None:
    """Copy local grads to the shared params (parallel_a3c.py:221-233)."""
    for lp, sp in zip(local_model.parameters(), shared_model.parameters()):
        sp._grad = lp.grad


def _compute_a3c_loss(model, obs_seq, actions, rewards, dones, last_obs,
                      gamma: float, gae_lambda: float,
                      value_loss_coef: float, entropy_coef: float):
    """Batched-forward A2C loss over one rollout (parallel_a3c.py:235-288
    semantics, advantages via the GAE scan op)."""
    T = len(actions)
    obs = torch.stack(obs_seq)                      # [T, obs]
    logits, values = model(obs)
    with torch.no_grad():
        _, v_last = model(last_obs.unsqueeze(0))
        bootstrap = v_last.reshape(1) * (1.0 - dones[-1])
    r = rewards.unsqueeze(1)
    d = (gamma * (1.0 - dones)).unsqueeze(1)
    adv, ret = gae_op(r, values.detach().unsqueeze(1), bootstrap, d,
                      lam=gae_lambda)
    logp = F.log_softmax(logits, dim=-1)
    logp_a = logp.gather(1, actions.unsqueeze(1)).squeeze(1)
    entropy = -(logp.exp() * logp).sum(dim=-1)
    pg_loss = -(logp_a * adv.squeeze(1)).sum()
    value_loss = F.mse_loss(values, ret.squeeze(1), reduction="sum")
    loss = (pg_loss + value_loss_coef * value_loss
            - entropy_coef * entropy.sum())
    return loss, pg_loss.detach(), value_loss.detach(), entropy.mean().detach()


def a3c_worker(worker_id: int, args_dict: dict, shared_model, optimizer,
               global_step, stop_event, lock, result_q=None):
    """Hogwild worker main (parallel_a3c.py:327-389)."""
    args = A3CArguments(**args_dict)
    torch.manual_seed(args.seed + worker_id)
    torch.set_num_threads(1)
    env = make_env(args.env_id, seed=args.seed + worker_id)
    obs_dim = env.observation_space.shape[0]
    act_dim = env.action_space.n
    local = ActorCriticNet(obs_dim, act_dim)
    obs, _ = env.reset()
    ep_ret, ep_len = 0.0, 0
    while not stop_event.is_set() and global_step.value < args.max_train_steps:
        local.load_state_dict(shared_model.state_dict())
        obs_seq, act_seq, rew_seq, done_seq = [], [], [], []
        for _ in range(args.rollout_steps):
            t = torch.as_tensor(obs, dtype=torch.float32)
            with torch.no_grad():
                action, _, _ = local.get_action(t.unsqueeze(0))
            a = int(action.item())
            next_obs, r, term, trunc, _ = env.step(a)
            done = term or trunc
            obs_seq.append(t)
            act_seq.append(a)
            rew_seq.append(float(r))
            done_seq.append(float(done))
            ep_ret += r
            ep_len += 1
            obs = next_obs
            if done:
                if result_q is not None:
                    try:
                        result_q.put_nowait((worker_id, ep_ret, ep_len))
                    except Exception:
                        pass
                obs, _ = env.reset()
                ep_ret, ep_len = 0.0, 0
                break
        loss, *_ = _compute_a3c_loss(
            local, obs_seq, torch.as_tensor(act_seq),
            torch.as_tensor(rew_seq), torch.as_tensor(done_seq),
            torch.as_tensor(obs, dtype=torch.float32), args.gamma,
            args.gae_lambda, args.value_loss_coef, args.entropy_coef)
        local.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(local.parameters(), 50.0)
        with lock:
            ensure_shared_grads(local, shared_model)
            optimizer.step()
        with global_step.get_lock():
            global_step.value += len(act_seq)
    env.close()


class A3CTrainer:
    """Owner of the hogwild topology (parallel_a3c.py:468-507)."""

    def __init__(self, args: A3CArguments):
        self.args = args
        env = make_env(args.env_id, seed=args.seed)
        self.obs_dim = env.observation_space.shape[0]
        self.act_dim = env.action_space.n
        env.close()
        self.shared_model = ActorCriticNet(self.obs_dim, self.act_dim)
        self.shared_model.share_memory()
        opt_params = self.shared_model.parameters()
        self.optimizer = SharedAdam(opt_params, lr=args.learning_rate)
        ctx = mp.get_context("fork")
        self.global_step = ctx.Value("l", 0)
        self.stop_event = ctx.Event()
        self.lock = ctx.Lock()
        self.result_q = ctx.Queue(maxsize=512)
        self.workers = []
        self._ctx = ctx

    def start(self) -> None:
        args_dict = {f: getattr(self.args, f)
                     for f in self.args.__dataclass_fields__}
        for i in range(self.args.num_workers):
            p = self._ctx.Process(
                target=a3c_worker,
                args=(i, args_dict, self.shared_model, self.optimizer,
                      self.global_step, self.stop_event, self.lock,
                      self.result_q),
                daemon=True)
            p.start()
            self.workers.append(p)

    def train(self, log_interval_s: float = 5.0) -> None:
        self.start()
        try:
            while self.global_step.value < self.args.max_train_steps:
                time.sleep(min(log_interval_s, 0.2))
        finally:
            self.shutdown()

    @torch.no_grad()
    def evaluate(self, n_episodes: int = 3) -> float:
        env = make_env(self.args.env_id, seed=self.args.seed + 777)
        rets = []
        for _ in range(n_episodes):
            obs, _ = env.reset()
            done, total = False, 0.0
            while not done:
                t = torch.as_tensor(obs, dtype=torch.float32).unsqueeze(0)
                action, _, _ = self.shared_model.get_action(t, greedy=True)
                obs, r, term, trunc, _ = env.step(int(action.item()))
                total += r
                done = term or trunc
            rets.append(total)
        env.close()
        return float(np.mean(rets))

    def save(self, path: str) -> None:
        save_agent_checkpoint(path, actor=self.shared_model)

    def shutdown(self) -> None:
        self.stop_event.set()
        for p in self.workers:
            p.join(timeout=2.0)
            if p.is_alive():
                p.terminate()
        self.workers.clear()
