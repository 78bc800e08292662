"""Ape-X: distributed prioritized DQN on the shared actor-learner runtime
(benchmark config 4).

The reference's Ape-X is a non-functional sketch (SURVEY.md
"Broken-as-shipped": a plain deque posing as a shared prioritized buffer,
actors whose add() never reaches the learner).  This implements the
architecture the reference describes (apex/apex_train.py, apex/worker.py,
apex/memory.py semantics; Horgan et al. 2018):

- N actor processes with per-actor exploration ε_i = base^(1 + i/(N-1)·α)
  step vectorized envs with a shared-flat Q-network (CPU inference);
- transitions travel as rollout-store chunks (the same shared-memory slot
  machinery as IMPALA — chunk rows are transitions, not a recurrent
  unroll);
- the learner ingests chunks into an HBM-resident prioritized buffer:
  n-step folding (:func:`scalerl_amd.ops.nstep_fold`), initial priorities
  from the |TD| of the current nets (one batched forward per chunk —
  learner-side rather than the paper's actor-side evaluation, same
  formula, no CPU Q-targets needed), device sum-tree insert;
- SGD: stratified PER sampling + the fused TD-loss kernel (double-DQN
  target, IS weights from (prio, total, min) device scalars, |TD| priority
  write-back) — sample→learn→update touches the host only for queue ops;
- weight publication: one flat D2H copy, actors alias it.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch
import torch.multiprocessing as mp

from ..config import ApexArguments
from ..data import PrioritizedReplayBuffer
from ..models.atari import AtariQNet
from ..ops import (FusedAdam, clip_grad_norm_, fused_polyak_, fused_td_loss,
                   nstep_fold)
from ..parallel import FlatParams, all_reduce_flat, get_rank, get_world_size
from ..parallel.rollout import RolloutStore, build_actor_env
from ..utils import get_logger
from ..utils.checkpoint import save_agent_checkpoint
from ..envs.synthetic import SyntheticAtariVecEnv


class EpsGreedyPolicy:
    """Actor-side ε-greedy over a shared-flat Q-net (obs-only input)."""

    def __init__(self, model: AtariQNet, eps: float, seed: int = 0):
        self.model = model
        self.eps = eps
        self.rng = np.random.default_rng(seed)

    @torch.no_grad()
    def __call__(self, obs, reward, done, last_action, want_state=False):
        if getattr(self.model, "noisy", False):
            # NoisyNet exploration: resample parameter noise instead of ε
            self.model.train()
            self.model.reset_noise()
            q = self.model(obs)
            return q.argmax(dim=-1), q, None
        q = self.model(obs)
        action = q.argmax(dim=-1)
        explore = torch.from_numpy(
            self.rng.random(action.shape[0]) < self.eps)
        rand = torch.from_numpy(
            self.rng.integers(self.model.num_actions, size=action.shape[0]))
        action = torch.where(explore, rand, action)
        return action, q, None


def apex_actor_loop(actor_id: int, env_spec: dict, store: RolloutStore,
                    free_q, full_q, stop_event, step_counter, actor_model,
                    eps: float, seed: int = 0):
    from ..parallel.rollout import ActorState, run_rollout
    torch.manual_seed(seed + actor_id)
    torch.set_num_threads(1)
    actor_model.eval()
    env = build_actor_env(env_spec, actor_id)
    policy = EpsGreedyPolicy(actor_model, eps, seed=seed + actor_id)
    state = ActorState(env, policy, store.envs_per_slot)
    try:
        while not stop_event.is_set():
            slot = free_q.get()
            if slot is None:
                break
            steps = run_rollout(state, store, slot)
            full_q.put(slot)
            with step_counter.get_lock():
                step_counter.value += steps
    except KeyboardInterrupt:
        pass


class ApexTrainer:
    def __init__(self, args: ApexArguments, device: Optional[str] = None):
        self.args = args
        self.rank = get_rank()
        self.log = get_logger("apex")
        if device is None:
            device = ("cuda" if torch.cuda.is_available() else "cpu") \
                if args.device == "auto" else args.device
        self.device = torch.device(device)

        probe = SyntheticAtariVecEnv(1)
        self.obs_shape = probe.observation_space.shape
        self.num_actions = probe.action_space.n

        E = args.envs_per_actor
        self.chunk_len = 32  # transitions per slot row-block
        self.categorical = getattr(args, "categorical_dqn", False)
        torch.manual_seed(args.seed + self.rank)
        self.actor_model = self._build_qnet()
        self.actor_model.eval()
        self.shared_flat = FlatParams(self.actor_model, device="cpu",
                                      share=True)
        self.store = RolloutStore(
            2 * args.num_actors + 4, self.chunk_len, E, self.obs_shape,
            self.num_actions, lstm_hidden=0)
        import os as _os
        self._mp_ctx = ("fork" if (self.device.type == "cpu"
                                   and not torch.cuda.is_initialized()
                                   and not _os.environ.get(
                                       "SCALERL_FORCE_SPAWN"))
                        else "spawn")
        ctx = mp.get_context(self._mp_ctx)
        self.free_q = ctx.SimpleQueue()
        self.full_q = ctx.Queue()  # timeout-capable: the learner watchdogs it
        self.stop_event = ctx.Event()
        self.step_counter = ctx.Value("l", 0)
        self.actors: List[mp.Process] = []
        self.global_step = 0
        self.learn_iters = 0

    def _build_qnet(self):
        args = self.args
        if self.categorical:
            from ..models.atari import CategoricalAtariQNet
            return CategoricalAtariQNet(
                self.obs_shape, self.num_actions, num_atoms=args.num_atoms,
                v_min=args.v_min, v_max=args.v_max, noisy=args.noisy_dqn)
        return AtariQNet(self.obs_shape, self.num_actions,
                         dueling=args.dueling_dqn, noisy=args.noisy_dqn)

    def actor_eps(self, i: int) -> float:
        n = max(self.args.num_actors - 1, 1)
        return self.args.eps_base ** (1 + i / n * self.args.eps_alpha)

    def start_actors(self) -> None:
        args = self.args
        ctx = mp.get_context(self._mp_ctx)
        env_spec = {"env_id": args.env_id,
                    "envs_per_actor": args.envs_per_actor,
                    "seed": args.seed + 104729 * self.rank}
        for i in range(args.num_actors):
            p = ctx.Process(
                target=apex_actor_loop,
                args=(i, env_spec, self.store, self.free_q, self.full_q,
                      self.stop_event, self.step_counter, self.actor_model,
                      self.actor_eps(i), args.seed),
                daemon=True, name=f"apex-actor-{self.rank}-{i}")
            p.start()
            self.actors.append(p)
        for s in range(self.store.num_slots):
            self.free_q.put(s)

    def setup_learner(self) -> None:
        args = self.args
        dev = self.device
        if dev.type == "cuda":
            torch.backends.cudnn.benchmark = True
        self.model = self._build_qnet().to(dev)
        self.model.load_state_dict(self.actor_model.state_dict())
        self.target_model = self._build_qnet().to(dev)
        self.target_model.load_state_dict(self.model.state_dict())
        self.flat = FlatParams(self.model, device=dev)
        if get_world_size() > 1:
            from ..parallel import broadcast_flat
            broadcast_flat(self.flat.flat, src=0)  # identical init (DP)
            self.target_model.load_state_dict(self.model.state_dict())
        self.target_flat = FlatParams(self.target_model, device=dev)
        self.optimizer = FusedAdam(self.flat.flat, lr=args.learning_rate)
        if args.use_per:
            self.buffer = PrioritizedReplayBuffer(
                args.buffer_size, self.obs_shape, alpha=args.per_alpha,
                obs_dtype=torch.uint8, device=dev, gamma=args.gamma,
                seed=args.seed)
        else:  # ParallelDQN mode: same topology, uniform replay
            from ..data import ReplayBuffer
            self.buffer = ReplayBuffer(
                args.buffer_size, self.obs_shape, obs_dtype=torch.uint8,
                device=dev, gamma=args.gamma, seed=args.seed)
        self.beta_per = args.per_beta
        if dev.type == "cuda":
            from ..parallel.rollout import PinRegistry
            self._pins = PinRegistry()
            self._pins.pin_store(self.store)
            self._pins.pin(self.shared_flat.flat)
            self._ingest_done = torch.cuda.Event()
        self._publish()

    @torch.no_grad()
    def _publish(self) -> None:
        self.shared_flat.flat.copy_(self.flat.flat)

    @torch.no_grad()
    def ingest_slot(self, slot: int) -> int:
        """One chunk → n-step fold → initial |TD| priorities → PER insert."""
        args = self.args
        dev = self.device
        T = self.chunk_len
        obs = self.store.obs[slot].to(dev, non_blocking=True)       # [T+1,E,...]
        action = self.store.action[slot].to(dev, non_blocking=True)
        reward = self.store.reward[slot].to(dev, non_blocking=True)
        done = self.store.done[slot].to(dev, non_blocking=True)

        rew = reward[1:]                                  # r for action[t]
        dn = done[1:].float()
        fr, fd, steps = nstep_fold(rew, dn, args.gamma, args.n_steps)
        t_idx = (torch.arange(T, device=dev).unsqueeze(1) + steps.long())
        b_idx = torch.arange(obs.shape[1], device=dev).unsqueeze(0).expand_as(t_idx)
        next_obs = obs[t_idx.reshape(-1), b_idx.reshape(-1)]
        disc = (args.gamma ** steps.float()) * (1.0 - fd)

        flat_obs = obs[:-1].reshape(-1, *self.obs_shape)
        flat_act = action[:-1].reshape(-1)
        flat_r = fr.reshape(-1)
        flat_d = fd.reshape(-1)
        flat_disc = disc.reshape(-1)

        if args.use_per:
            # initial priorities: |TD| under the current nets (batched)
            q = self.model(flat_obs).gather(1, flat_act.unsqueeze(1)).squeeze(1)
            qn_t = self.target_model(next_obs)
            if args.double_dqn:
                astar = self.model(next_obs).argmax(dim=1, keepdim=True)
            else:
                astar = qn_t.argmax(dim=1, keepdim=True)
            target = flat_r + flat_disc * qn_t.gather(1, astar).squeeze(1)
            prio = (q - target).abs() + 1e-6
            self.buffer.add_batch(flat_obs, flat_act, flat_r, next_obs,
                                  flat_d, discount=flat_disc,
                                  priorities=prio)
            # keep max_priority tracking without a sync storm
            self.buffer.max_priority = max(self.buffer.max_priority,
                                           float(prio.max()))
        else:
            self.buffer.add_batch(flat_obs, flat_act, flat_r, next_obs,
                                  flat_d, discount=flat_disc)
        return flat_obs.shape[0]

    def train_iteration(self) -> Dict[str, float]:
        args = self.args
        # 1) ingest available slots — BOUNDED per iteration, or actors that
        # outproduce the ingest forward would live-lock the learner here
        ingested = 0
        max_slots = 4
        import queue as _queue
        for _ in range(max_slots):
            if self.full_q.empty():
                if len(self.buffer) >= args.warmup_learn_steps or ingested:
                    break
            try:
                slot = self.full_q.get(timeout=60.0)
            except _queue.Empty:
                dead = [p.name for p in self.actors if not p.is_alive()]
                if dead:
                    raise RuntimeError(f"actor process(es) died: {dead}")
                continue
            ingested += self.ingest_slot(slot)
            # the H2D copies out of the hipHostRegister'd shared slot are
            # async: fence them before recycling the slot, or an actor can
            # overwrite it mid-copy and corrupt replay data (the PER path
            # was only safe by accident via float(prio.max()))
            if self.device.type == "cuda":
                self._ingest_done.record(torch.cuda.current_stream())
                self._ingest_done.synchronize()
            self.free_q.put(slot)
            self.global_step += self.chunk_len * args.envs_per_actor
        stats: Dict[str, float] = {"ingested": ingested}
        if len(self.buffer) < args.warmup_learn_steps:
            return stats
        # 2) SGD steps
        for _ in range(args.learner_update_times):
            if args.use_per:
                batch, idx, prio, p_total, p_min = \
                    self.buffer.sample_with_priorities(args.batch_size)
                per_kw = dict(
                    prios=prio,
                    p_total=p_total.reshape(1) if p_total.dim() == 0 else p_total,
                    p_min=p_min.reshape(1) if p_min.dim() == 0 else p_min,
                    beta=self.beta_per, replay_size=len(self.buffer))
            else:
                batch = self.buffer.sample(args.batch_size)
                per_kw = {}
            self.flat.flat_grad.zero_()
            if self.categorical:
                # C51 projection loss (Bellemare et al. 2017); IS weights
                # from the sampled priorities, priorities ← per-sample KL
                from ..models.noisy import c51_loss
                weights = None
                if per_kw:
                    N = per_kw["replay_size"]
                    beta = per_kw["beta"]
                    w = (per_kw["prios"] / per_kw["p_total"] * N) ** -beta
                    w_max = (per_kw["p_min"] / per_kw["p_total"] * N) ** -beta
                    weights = w / w_max
                loss, td_abs = c51_loss(
                    self.model, self.target_model, batch["obs"],
                    batch["action"], batch["reward"], batch["discount"],
                    batch["next_obs"], double=args.double_dqn,
                    weights=weights)
            else:
                q = self.model(batch["obs"])
                with torch.no_grad():
                    qn_t = self.target_model(batch["next_obs"])
                    qn_o = self.model(batch["next_obs"]) \
                        if args.double_dqn else None
                loss, td_abs = fused_td_loss(
                    q, qn_o, qn_t, batch["action"], batch["reward"],
                    batch["discount"], **per_kw)
            loss.backward()
            all_reduce_flat(self.flat.flat_grad, average=True)
            if args.max_grad_norm > 0:
                clip_grad_norm_(self.flat.flat_grad, args.max_grad_norm)
            self.optimizer.step(self.flat.flat_grad)
            if args.use_per:
                self.buffer.update_priorities(idx, td_abs)
            self.learn_iters += 1
            stats["loss"] = loss
            # β anneal → 1
            self.beta_per = min(1.0, args.per_beta + (1 - args.per_beta) *
                                self.learn_iters / args.per_beta_anneal_steps)
            if self.learn_iters % args.target_update_frequency == 0:
                if args.soft_update_tau > 0:
                    fused_polyak_(self.target_flat.flat, self.flat.flat,
                                  args.soft_update_tau)
                else:
                    self.target_flat.flat.copy_(self.flat.flat)
            if self.learn_iters % args.publish_interval == 0:
                self._publish()
        return stats

    def save(self, path: str) -> None:
        save_agent_checkpoint(path, actor=self.model,
                              actor_target=self.target_model,
                              extra={"optimizer_state_dict":
                                     self.optimizer.state_dict()})

    def shutdown(self) -> None:
        self.stop_event.set()
        for _ in self.actors:
            self.free_q.put(None)
        for p in self.actors:
            p.join(timeout=2.0)
            if p.is_alive():
                p.terminate()
        self.actors.clear()
        if getattr(self, "_pins", None) is not None:
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            self._pins.unpin_all()
            self._pins = None


class ParallelDQNTrainer(ApexTrainer):
    """Self-contained actor-learner DQN (reference parallel_dqn.py:106-443
    semantics on the shared runtime): N actor processes with a common
    ε-greedy schedule feed transition chunks to a central learner with a
    UNIFORM replay buffer; periodic hard target sync and weight
    publication.  Identical topology to Ape-X minus prioritization — pass
    an ApexArguments with ``use_per=False`` (the constructor enforces it)."""

    def __init__(self, args: ApexArguments, device: Optional[str] = None):
        args.use_per = False
        super().__init__(args, device=device)

    def actor_eps(self, i: int) -> float:
        # single shared schedule endpoint (parallel_dqn.py eps handling),
        # not Ape-X's per-actor spread
        return self.args.eps_greedy_end
