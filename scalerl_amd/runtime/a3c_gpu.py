"""A3C at GPU scale — batched synchronous A2C on the actor-learner runtime
(BASELINE config 2: Pong 42×42, 16 CPU actors + 1 MI355X learner).

Reference counterpart: ``parallel_a3c.py:71-513`` (hogwild A3C with the
4-conv+LSTM 42×42 policy, ``a3c/utils/atari_model.py:57-144``).  Hogwild's
shared-memory gradient publication does not scale to a GPU learner, so the
MI355X design batches it: N CPU actor processes fill shared rollout slots
(the IMPALA transport, parallel/rollout.py) with 42×42 float observations;
the learner gathers K slots into a [T+1, K*E] batch, unrolls the
A3CAtariNet over T with done-masked MaskedLSTM, computes GAE advantages on
device (ops/scans.py HIP scan) and the A2C loss (policy gradient + value
MSE + entropy), steps a fused Adam on the flat parameter buffer, and
publishes weights back to the actors with one flat D2H copy.

The CPU-hogwild parity mode lives in :mod:`scalerl_amd.runtime.a3c`; this
module is the GPU-scale path for the same algorithm family.
"""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import torch
import torch.nn.functional as F
import torch.multiprocessing as mp

from ..config import A3CGpuArguments
from ..models.a3c_atari import A3CAtariNet
from ..ops import FusedAdam, clip_grad_norm_
from ..ops.scans import gae as gae_op
from ..parallel import FlatParams, all_reduce_flat, get_rank, get_world_size
from ..parallel.rollout import (ActorState, RolloutStore, build_actor_env,
                                BatchGatherer, run_rollout)
from ..utils import Timings, get_logger
from ..utils.checkpoint import save_checkpoint


class A3CLocalPolicy:
    """Actor-side CPU inference for A3CAtariNet (LocalPolicy call contract,
    parallel/rollout.py): one LSTM step with done masking, multinomial
    action sampling (the reference's train-mode behavior)."""

    def __init__(self, model: A3CAtariNet, num_envs: int):
        self.model = model
        self.state = model.initial_state(num_envs)

    @torch.no_grad()
    def __call__(self, obs, reward, done, last_action, want_state=False):
        snap = None
        if want_state:
            snap = torch.stack([self.state[0], self.state[1]])  # [2,L,E,H]
        notdone = (~done).float().unsqueeze(0)  # [1,E]
        # done-mask BEFORE the step (MaskedLSTM masks in-kernel per step)
        logits, _, self.state = self.model(obs.float(), self.state, notdone)
        probs = F.softmax(logits, dim=-1)
        action = torch.multinomial(probs, 1).squeeze(-1)
        return action, logits, snap


def a3c_actor_loop(actor_id: int, env_spec: dict, store: RolloutStore,
                   free_q, full_q, stop_event, step_counter,
                   actor_model=None, episode_queue=None, seed: int = 0):
    """Actor process main (mirrors parallel/rollout.actor_loop with the
    A3C policy; reference: parallel_a3c.py:327-389 rollout phase)."""
    import queue as _queue
    torch.manual_seed(seed + actor_id)
    torch.set_num_threads(1)
    env = build_actor_env(env_spec, actor_id)
    actor_model.eval()
    policy = A3CLocalPolicy(actor_model, store.envs_per_slot)
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
            if episode_queue is not None:
                mask = store.done[slot, 1:].numpy()
                if mask.any():
                    rets = store.episode_return[slot, 1:].numpy()[mask]
                    try:
                        episode_queue.put_nowait(rets.tolist())
                    except _queue.Full:
                        pass
    except KeyboardInterrupt:
        pass


class A3CGpuTrainer:
    def __init__(self, args: A3CGpuArguments, device: Optional[str] = None):
        self.args = args
        self.rank = get_rank()
        self.world_size = get_world_size()
        self.log = get_logger("a3c-gpu")
        if device is None:
            device = ("cuda" if torch.cuda.is_available() else "cpu") \
                if args.device == "auto" else args.device
        self.device = torch.device(device)

        # probe the 42×42 wrapped env for shapes
        from ..envs.a3c_env import create_atari_env
        probe = create_atari_env(args.env_id, seed=args.seed)
        self.obs_shape = probe.observation_space.shape     # (1, 42, 42)
        self.num_actions = probe.action_space.n
        probe.close()

        E = args.envs_per_actor
        K = args.slots_per_batch
        num_slots = args.num_actors + 2 * K + 2
        torch.manual_seed(args.seed + self.rank)
        self.actor_model = A3CAtariNet(self.obs_shape[0], self.num_actions)
        self.actor_model.eval()
        self.shared_flat = FlatParams(self.actor_model, device="cpu",
                                      share=True)
        self.store = RolloutStore(
            num_slots, args.rollout_steps, E, self.obs_shape,
            self.num_actions, lstm_layers=1, lstm_hidden=256,
            obs_dtype=torch.float32)

        self._mp_ctx = ("fork" if (self.device.type == "cpu"
                                   and not torch.cuda.is_initialized()
                                   and not os.environ.get(
                                       "SCALERL_FORCE_SPAWN"))
                        else "spawn")
        ctx = mp.get_context(self._mp_ctx)
        self.free_q = ctx.SimpleQueue()
        self.full_q = ctx.Queue()
        self.stop_event = ctx.Event()
        self.step_counter = ctx.Value("l", 0)
        self.episode_q = ctx.Queue(maxsize=256)
        self.actors: List[mp.Process] = []
        self.global_step = 0
        self.learn_iters = 0
        self.timings = Timings()
        self._pending = None
        self._num_slots = num_slots

    # -- lifecycle ---------------------------------------------------------
    def start_actors(self) -> None:
        args = self.args
        ctx = mp.get_context(self._mp_ctx)
        env_spec = {"env_id": args.env_id, "envs_per_actor": args.envs_per_actor,
                    "seed": args.seed + 7919 * self.rank, "a3c_wrap": True}
        for i in range(args.num_actors):
            p = ctx.Process(
                target=a3c_actor_loop,
                args=(i, env_spec, self.store, self.free_q, self.full_q,
                      self.stop_event, self.step_counter),
                kwargs=dict(actor_model=self.actor_model, seed=args.seed,
                            episode_queue=self.episode_q),
                daemon=True, name=f"a3c-actor-{self.rank}-{i}")
            p.start()
            self.actors.append(p)
        for s in range(self._num_slots):
            self.free_q.put(s)

    def setup_learner(self) -> None:
        args = self.args
        self.model = A3CAtariNet(self.obs_shape[0],
                                 self.num_actions).to(self.device)
        self.model.load_state_dict(self.actor_model.state_dict())
        self.flat = FlatParams(self.model, device=self.device)
        from ..parallel.dist import broadcast_flat, get_world_size as _ws
        if _ws() > 1:  # dynamic: dist may init after __init__
            broadcast_flat(self.flat.flat, src=0)  # identical init (DP)
        self.optimizer = FusedAdam(self.flat.flat, lr=args.learning_rate)
        self.gatherer = BatchGatherer(self.store, self.device,
                                      args.slots_per_batch)
        if self.device.type == "cuda":
            torch.backends.cudnn.benchmark = True
            from ..parallel.rollout import PinRegistry
            self._pins = PinRegistry()
            pinned = self._pins.pin_store(self.store)
            pinned += self._pins.pin(self.shared_flat.flat)
            self.log.info(f"pinned {pinned / 1e6:.1f} MB")
        else:
            self._pins = None
        self._publish()
        self.autocast_dtype = (torch.bfloat16 if args.dtype == "bf16" and
                               self.device.type == "cuda" else None)

    @torch.no_grad()
    def _publish(self) -> None:
        self.shared_flat.flat.copy_(self.flat.flat)

    # -- core step ---------------------------------------------------------
    def _dequeue_batch(self):
        import queue as _queue
        slot_ids = []
        while len(slot_ids) < self.args.slots_per_batch:
            try:
                slot_ids.append(self.full_q.get(timeout=30.0))
            except _queue.Empty:
                dead = [p.name for p in self.actors if not p.is_alive()]
                if dead:
                    raise RuntimeError(f"actor process(es) died: {dead}")
        self.timings.time("dequeue")
        return slot_ids

    def next_batch(self) -> Dict[str, torch.Tensor]:
        if self._pending is None:
            ids = self._dequeue_batch()
            self._pending = (self.gatherer.start(ids), ids)
        token, ids = self._pending
        batch = self.gatherer.finish(token)
        self.timings.time("gather_wait")
        for s in ids:
            self.free_q.put(s)
        nxt = self._dequeue_batch()
        self._pending = (self.gatherer.start(nxt), nxt)
        return batch

    def learn_step(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        args = self.args
        self.model.train()
        self.flat.flat_grad.zero_()
        obs = batch["obs"].float()                       # [T+1, B, 1,42,42]
        notdone = (~batch["done"]).float()               # [T+1, B]
        cs = batch["core_state"]                         # [2, L, B, H]
        state = (cs[0], cs[1])
        if self.autocast_dtype is not None:
            with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
                logits, values, _ = self.model.unroll(obs, notdone, state)
        else:
            logits, values, _ = self.model.unroll(obs, notdone, state)
        logits = logits.float()
        values = values.float()
        T = args.rollout_steps
        rewards = torch.clamp(batch["reward"][1:], -1, 1)  # r for action[t]
        discounts = notdone[1:] * args.gamma
        with torch.no_grad():
            adv, returns = gae_op(rewards, values[:-1].detach(),
                                  values[-1].detach(), discounts,
                                  args.gae_lambda)
        logp = F.log_softmax(logits[:-1], dim=-1)
        taken = logp.gather(-1, batch["action"][:-1].unsqueeze(-1)).squeeze(-1)
        pg_loss = -(taken * adv).mean()
        value_loss = 0.5 * F.mse_loss(values[:-1], returns)
        entropy = -(logp.exp() * logp).sum(-1).mean()
        total = (pg_loss + args.value_loss_coef * value_loss
                 - args.entropy_coef * entropy)
        self.timings.time("forward")
        total.backward()
        self.timings.time("backward")
        all_reduce_flat(self.flat.flat_grad, average=True)  # no-op
        # when torch.distributed is not initialized
        if args.max_grad_norm > 0:
            clip_grad_norm_(self.flat.flat_grad, args.max_grad_norm)
        self.optimizer.step(self.flat.flat_grad)
        self.timings.time("optimize")
        self._publish()
        self.timings.time("publish")
        self.gatherer.mark_consumed()
        self.learn_iters += 1
        self.global_step += T * batch["action"].shape[1]
        return {"total_loss": total.detach(), "pg_loss": pg_loss.detach(),
                "value_loss": value_loss.detach(),
                "entropy": entropy.detach()}

    def train_iteration(self) -> Dict[str, torch.Tensor]:
        batch = self.next_batch()
        return self.learn_step(batch)

    def train(self) -> None:
        args = self.args
        self.start_actors()
        self.setup_learner()
        last_log, last_step = time.time(), 0
        try:
            while self.global_step < args.total_steps:
                stats = self.train_iteration()
                now = time.time()
                if self.rank == 0 and now - last_log > 5.0:
                    sps = (self.global_step - last_step) / (now - last_log)
                    last_log, last_step = now, self.global_step
                    self.log.info(f"step {self.global_step} SPS {sps:,.0f} "
                                  f"loss {float(stats['total_loss']):.3f}")
        finally:
            if self.rank == 0 and not args.disable_checkpoint:
                save_checkpoint(os.path.join(args.output_dir, "model.tar"),
                                model=self.model, optimizer=None,
                                hparam=vars(args),
                                extra={"global_step": self.global_step})
            self.shutdown()

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
