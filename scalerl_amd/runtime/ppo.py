"""DD-PPO: decentralized distributed PPO (benchmark config 5 — synthetic
128×128 RGB-D PointGoal, ResNet-18+LSTM, RCCL all-reduce).

Not present in the reference's code (README paper citation only) —
implemented fresh per SURVEY.md §7 step 9, following Wijmans et al. 2020:

- each learner rank owns its own vectorized envs and alternates
  {collect rollout → PPO epochs}, synchronizing ONLY through flat-grad
  all-reduces (no parameter server);
- straggler preemption: collection proceeds in chunks; after each chunk
  the ranks all-reduce a done-counter, and when ≥ ``preemption_threshold``
  of ranks have finished their T steps everyone truncates the rollout at
  the current step (every rank executes the same collective schedule, so
  the protocol is deadlock-free by construction);
- advantages via the GAE HIP scan, PPO clip loss over [T·B] minibatches,
  fused Adam on the flat parameter buffer.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..config import DDPPOArguments
from ..envs.synthetic import SyntheticPointGoalEnv
from ..envs.vec_env import SyncVectorEnv
from ..models.resnet import ResNetLSTMPolicy
from ..ops import FusedAdam, clip_grad_norm_, gae
from ..parallel import FlatParams, all_reduce_flat, get_rank, get_world_size
from ..parallel.dist import is_distributed
from ..utils import get_logger
from ..utils.checkpoint import save_checkpoint


class DDPPOTrainer:
    def __init__(self, args: DDPPOArguments, device: Optional[str] = None):
        self.args = args
        self.rank = get_rank()
        self.world = get_world_size()
        self.log = get_logger("ddppo")
        if device is None:
            device = ("cuda" if torch.cuda.is_available() else "cpu") \
                if args.device == "auto" else args.device
        self.device = torch.device(device)
        torch.manual_seed(args.seed + self.rank)

        B = args.num_envs
        self.env = SyncVectorEnv([
            (lambda i=i: SyntheticPointGoalEnv(seed=args.seed * 100 + i))
            for i in range(B)])
        self.num_actions = self.env.action_space.n
        self.obs_shape = self.env.observation_space.shape

        self.model = ResNetLSTMPolicy(self.num_actions,
                                      in_channels=self.obs_shape[0]).to(self.device)
        self.flat = FlatParams(self.model, device=self.device)
        if is_distributed():
            from ..parallel.dist import broadcast_flat
            broadcast_flat(self.flat.flat, src=0)  # identical init on all ranks
        self.optimizer = FusedAdam(self.flat.flat, lr=args.learning_rate,
                                   eps=1e-5)
        self.obs = torch.from_numpy(self.env.reset(seed=args.seed))
        self.done = torch.zeros(B, dtype=torch.bool)
        self.core_state = self.model.initial_state(B, device=self.device)
        self.global_step = 0
        self.preempted_steps = 0

    @torch.no_grad()
    def collect_rollout(self, chunk: int = 16):
        """Collect up to T steps; preemption checks once per chunk."""
        args = self.args
        T, B = args.rollout_length, args.num_envs
        dev = self.device
        obs_buf = torch.empty(T, B, *self.obs_shape, dtype=torch.uint8)
        act_buf = torch.empty(T, B, dtype=torch.int64)
        logp_buf = torch.empty(T, B)
        val_buf = torch.empty(T, B)
        rew_buf = torch.empty(T, B)
        done_buf = torch.empty(T, B)
        start_state = (self.core_state[0].clone(), self.core_state[1].clone())
        notdone_hist = torch.empty(T, B)

        t = 0
        while t < T:
            end = min(t + chunk, T)
            for i in range(t, end):
                obs_dev = self.obs.to(dev).unsqueeze(0)
                notdone = (~self.done).float().unsqueeze(0)
                logits, value, self.core_state = self.model(
                    obs_dev, notdone.to(dev), self.core_state)
                probs = F.softmax(logits[0].float(), dim=-1)
                action = torch.multinomial(probs, 1).squeeze(1)
                logp = torch.log(probs.gather(1, action.unsqueeze(1)).squeeze(1))
                obs_buf[i] = self.obs
                act_buf[i] = action.cpu()
                logp_buf[i] = logp.cpu()
                val_buf[i] = value[0].float().cpu()
                notdone_hist[i] = (~self.done).float()
                obs_np, rew_np, done_np = self.env.step(action.cpu().numpy())
                rew_buf[i] = torch.from_numpy(rew_np)
                done_buf[i] = torch.from_numpy(done_np).float()
                self.obs = torch.from_numpy(obs_np)
                self.done = torch.from_numpy(done_np)
            t = end
            # preemption vote (same collective schedule on every rank)
            if is_distributed():
                me_done = 1.0 if t >= T else 0.0
                v = torch.tensor([me_done], dtype=torch.float64,
                                 device=dev if dev.type == "cuda" else "cpu")
                dist.all_reduce(v)
                if v.item() / self.world >= self.args.preemption_threshold \
                        and t < T:
                    self.preempted_steps += T - t
                    break
        # bootstrap value
        with torch.no_grad():
            obs_dev = self.obs.to(dev).unsqueeze(0)
            notdone = (~self.done).float().unsqueeze(0)
            _, boot_val, _ = self.model(obs_dev, notdone.to(dev),
                                        self.core_state)
        self.global_step += t * B
        return {"obs": obs_buf[:t], "action": act_buf[:t],
                "logp": logp_buf[:t], "value": val_buf[:t],
                "reward": rew_buf[:t], "done": done_buf[:t],
                "notdone": notdone_hist[:t],
                "bootstrap": boot_val[0].float().cpu(),
                "start_state": start_state, "t": t}

    def update(self, rollout) -> Dict[str, float]:
        """PPO epochs over the rollout; one all-reduce per minibatch."""
        args = self.args
        dev = self.device
        T = rollout["t"]
        B = args.num_envs
        discounts = args.gamma * (1.0 - rollout["done"])
        adv, ret = gae(rollout["reward"].to(dev), rollout["value"].to(dev),
                       rollout["bootstrap"].to(dev), discounts.to(dev),
                       lam=args.gae_lambda)
        adv = (adv - adv.mean()) / (adv.std() + 1e-5)

        obs = rollout["obs"].to(dev)
        actions = rollout["action"].to(dev)
        old_logp = rollout["logp"].to(dev)
        notdone = rollout["notdone"].to(dev)
        start_state = tuple(s.to(dev) for s in rollout["start_state"])

        stats = {}
        for _ in range(args.ppo_epochs):
            # recurrent PPO: forward the whole sequence, minibatch over
            # env columns to keep LSTM state exact
            perm = torch.randperm(B, device=dev)
            mb_size = max(1, B // args.num_minibatches)
            for mb in range(args.num_minibatches):
                cols = perm[mb * mb_size:(mb + 1) * mb_size]
                if cols.numel() == 0:
                    continue
                self.flat.flat_grad.zero_()
                state = (start_state[0][:, cols], start_state[1][:, cols])
                logits, values, _ = self.model(obs[:, cols],
                                               notdone[:, cols], state)
                # fused HIP clip loss on GPU (one launch incl. analytic
                # grads), composed reference on CPU
                from ..ops import ppo_fused_loss
                k = cols.numel()
                loss, comps = ppo_fused_loss(
                    logits.float().reshape(T * k, -1),
                    values.float().reshape(T * k),
                    actions[:, cols].reshape(T * k),
                    old_logp[:, cols].reshape(T * k),
                    adv[:, cols].reshape(T * k),
                    ret[:, cols].reshape(T * k),
                    clip_eps=args.clip_eps, vcoef=args.value_loss_coef,
                    ecoef=args.entropy_coef)
                loss.backward()
                all_reduce_flat(self.flat.flat_grad, average=True)
                if args.max_grad_norm > 0:
                    clip_grad_norm_(self.flat.flat_grad, args.max_grad_norm)
                self.optimizer.step(self.flat.flat_grad)
                stats = {"loss": float(loss.detach()),
                         "pg_loss": float(comps[0]),
                         "v_loss": float(comps[1]),
                         "entropy": float(comps[2])}
        return stats

    def train_iteration(self) -> Dict[str, float]:
        rollout = self.collect_rollout()
        stats = self.update(rollout)
        stats["steps"] = rollout["t"] * self.args.num_envs
        return stats

    def save(self, path: str) -> None:
        save_checkpoint(path, model=self.model, optimizer=self.optimizer,
                        hparam=vars(self.args),
                        extra={"global_step": self.global_step})
