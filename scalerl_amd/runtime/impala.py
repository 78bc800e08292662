"""IMPALA on the MI355X actor-learner runtime — the flagship trainer
(benchmark config 3: synthetic Atari, 1/2/4/8 learner GPUs).

Reference architecture being reimplemented (impala_atari.py:40-521):
actor procs fill shared rollout slots via free/full queues; learner batches
slots, runs V-trace + losses, RMSProp, publishes weights back.  MI355X
design:

- per-rank topology: each learner rank (1 process per GPU) owns its own
  actor processes + rollout store; ranks sync ONLY via one flat-grad
  all-reduce per learn step (weak scaling over RCCL/xGMI);
- learner hot path: model fwd (bf16 convs) → fused HIP V-trace+loss kernel
  → backward → flat-grad clip (HIP) → RCCL all-reduce → fused RMSProp (HIP);
- weight publication: one D2H flat copy into the shared CPU flat buffer the
  actor models alias (impala_atari.py:348 equivalent);
- process discipline: children are SPAWNED whenever the GPU is involved
  (the HIP runtime does not survive fork on ROCm); pure-CPU runs keep
  fork for startup speed.

Step accounting matches the reference (impala_atari.py:391): one learn
iteration consumes rollout_length × batch_size env steps.
"""

from __future__ import annotations

import os
import time
from typing import Dict, List, Optional

import torch
import torch.multiprocessing as mp

from ..config import ImpalaArguments
from ..envs.synthetic import SyntheticAtariVecEnv
from ..models.atari import AtariNet
from ..ops import FusedRMSprop, clip_grad_norm_, impala_loss
from ..parallel import FlatParams, all_reduce_flat, get_rank, get_world_size
from ..parallel.inference import InferenceSlots, inference_worker
from ..parallel.rollout import BatchGatherer, RolloutStore, actor_loop
from ..utils import Timings, get_logger
from ..utils.checkpoint import load_checkpoint, save_checkpoint


class ImpalaTrainer:
    def __init__(self, args: ImpalaArguments, device: Optional[str] = None):
        self.args = args
        self.rank = get_rank()
        self.world_size = get_world_size()
        self.log = get_logger("impala")
        if device is None:
            device = ("cuda" if torch.cuda.is_available() else "cpu") \
                if args.device == "auto" else args.device
        self.device = torch.device(device)

        # ---- shapes ----
        probe_env = SyntheticAtariVecEnv(1) if args.env_id == "synthetic-atari" \
            else None
        if probe_env is not None:
            self.obs_shape = probe_env.observation_space.shape
            self.num_actions = probe_env.action_space.n
        else:
            from ..envs.registry import make_env
            e = make_env(args.env_id, deepmind_wrap=True)
            self.obs_shape = e.observation_space.shape
            self.num_actions = e.action_space.n
            e.close()

        E = args.envs_per_actor
        assert args.batch_size % E == 0, \
            f"batch_size ({args.batch_size}) must be a multiple of " \
            f"envs_per_actor ({E})"
        self.slots_per_batch = args.batch_size // E
        # double-buffered actors run two env groups, so each holds TWO
        # slots in flight (gpu-inference only; see actor_loop_db)
        effective_inference = args.inference
        if effective_inference == "gpu" and self.device.type != "cuda":
            effective_inference = "cpu"
        self.double_buffer = bool(args.actor_double_buffer) and \
            effective_inference == "gpu"
        per_actor_slots = 2 if self.double_buffer else 1
        # enough slots that every actor can have its group(s) in flight
        # plus two batches queued; more would grow host shm for no
        # throughput (8 ranks x 24 actors x 290 MB slots must fit /dev/shm)
        num_buffers = args.num_buffers or (
            per_actor_slots * args.num_actors +
            2 * self.slots_per_batch + 2)
        self.local_slots = num_buffers
        num_buffers += args.remote_actor_slots  # reserved tail ids

        # ---- CPU phase: shared actor model + store + actor processes ----
        # (everything here must precede any CUDA/HIP initialization)
        torch.manual_seed(args.seed + self.rank)
        self.actor_model = AtariNet(self.obs_shape, self.num_actions,
                                    use_lstm=args.use_lstm)
        self.actor_model.eval()
        self.shared_flat = FlatParams(self.actor_model, device="cpu",
                                      share=True)
        lstm_hidden = (512 + self.num_actions + 1) if args.use_lstm else 0
        self.store = RolloutStore(
            num_buffers, args.rollout_length, E, self.obs_shape,
            self.num_actions, lstm_layers=2, lstm_hidden=lstm_hidden)

        # spawn, not fork, whenever the GPU is involved: the learner (and
        # inference worker) initialize the HIP runtime, which does not
        # survive fork on ROCm.  Shared-memory tensors travel to spawned
        # children as shm handles.  Pure-CPU runs keep fork (fast startup).
        self._mp_ctx = ("fork" if (self.device.type == "cpu"
                                   and not torch.cuda.is_initialized()
                                   and not os.environ.get(
                                       "SCALERL_FORCE_SPAWN"))
                        else "spawn")
        ctx = mp.get_context(self._mp_ctx)
        self.free_q = ctx.SimpleQueue()
        self.full_q = ctx.Queue()  # timeout-capable: the learner watchdogs it
        self.stop_event = ctx.Event()
        self.step_counter = ctx.Value("l", 0)
        self.episode_q = ctx.Queue(maxsize=256)
        self.actors: List[mp.Process] = []
        self._started = False
        self._num_buffers = num_buffers

        # ---- inference placement ----
        self.inference = args.inference
        if self.inference == "gpu" and self.device.type != "cuda":
            self.inference = "cpu"
        self.inference_proc: Optional[mp.Process] = None
        if self.inference == "gpu":
            rows = args.num_actors * (2 if self.double_buffer else 1)
            self.inf_slots = InferenceSlots(
                rows, E, self.obs_shape, self.num_actions,
                lstm_layers=2, lstm_hidden=lstm_hidden)
            self.inf_req_q = ctx.Queue()
            self.inf_sems = [ctx.Semaphore(0) for _ in range(rows)]
            self.weights_version = ctx.Value("l", 0)
            self.inf_pause = ctx.Value("i", 0)
            self.inf_paused_ack = ctx.Value("i", 0)

        # ---- learner state (device init deferred to setup()) ----
        self.learner_model: Optional[AtariNet] = None
        self.global_step = 0
        self.learn_iters = 0
        self.timings = Timings()
        self._pending = None  # in-flight prefetched batch token
        self._graphed = None
        self.use_graph = bool(args.use_graph) and self.device.type == "cuda"

    # -- lifecycle ---------------------------------------------------------
    def start_actors(self) -> None:
        if self._started:
            return
        args = self.args
        env_spec = {"env_id": args.env_id,
                    "envs_per_actor": args.envs_per_actor,
                    "seed": args.seed + 7919 * self.rank}
        ctx = mp.get_context(self._mp_ctx)

        if self.inference == "gpu":
            model_kwargs = dict(observation_shape=self.obs_shape,
                                num_actions=self.num_actions,
                                use_lstm=args.use_lstm)
            dev_index = self.device.index or 0
            wargs = (dev_index, model_kwargs, self.inf_slots,
                     self.shared_flat.flat, self.weights_version,
                     self.inf_req_q, self.inf_sems, self.stop_event)
            if getattr(args, "inference_worker", "thread") == "thread":
                import threading
                self.inference_proc = threading.Thread(
                    target=inference_worker, args=wargs,
                    kwargs=dict(seed=args.seed + 9999, as_thread=True),
                    daemon=True, name=f"impala-infer-{self.rank}")
                self.inference_proc.start()
            else:
                self.inference_proc = ctx.Process(
                    target=inference_worker, args=wargs,
                    kwargs=dict(seed=args.seed + 9999,
                                pause_flag=self.inf_pause,
                                paused_ack=self.inf_paused_ack),
                    daemon=True, name=f"impala-infer-{self.rank}")
                self.inference_proc.start()

        for i in range(args.num_actors):
            if self.double_buffer:
                from ..parallel.rollout import actor_loop_db
                target = actor_loop_db
                kw = dict(seed=args.seed, episode_queue=self.episode_q,
                          inf_slots=self.inf_slots,
                          inf_req_q=self.inf_req_q,
                          inf_sems=(self.inf_sems[2 * i],
                                    self.inf_sems[2 * i + 1]),
                          row_ids=(2 * i, 2 * i + 1))
            else:
                target = actor_loop
                kw = dict(seed=args.seed, episode_queue=self.episode_q)
                if self.inference == "gpu":
                    kw.update(inf_slots=self.inf_slots,
                              inf_req_q=self.inf_req_q,
                              inf_sem=self.inf_sems[i])
                else:
                    kw.update(actor_model=self.actor_model)
            p = ctx.Process(
                target=target,
                args=(i, env_spec, self.store, self.free_q, self.full_q,
                      self.stop_event, self.step_counter),
                kwargs=kw, daemon=True,
                name=f"impala-actor-{self.rank}-{i}")
            p.start()
            self.actors.append(p)
        for s in range(self.local_slots):
            self.free_q.put(s)
        self._started = True

    def setup_learner(self) -> None:
        """Device-side init (safe to call after start_actors)."""
        args = self.args
        self.learner_model = AtariNet(self.obs_shape, self.num_actions,
                                      use_lstm=args.use_lstm).to(self.device)
        # identical init across ranks and with the actor model
        self.learner_model.load_state_dict(self.actor_model.state_dict())
        self.flat = FlatParams(self.learner_model, device=self.device)
        # identical init on all ranks — per-rank seeds differ (they
        # must, for decorrelated actors), so DP grad-averaging is only
        # coherent after a rank-0 weight broadcast.  get_world_size() is
        # read HERE, not in __init__: bench.py constructs the trainer
        # before init_distributed().
        from ..parallel.dist import broadcast_flat, get_world_size as _ws
        if _ws() > 1:
            broadcast_flat(self.flat.flat, src=0)
        self.optimizer = FusedRMSprop(
            self.flat.flat, lr=args.learning_rate, alpha=args.rmsprop_alpha,
            eps=args.rmsprop_eps, momentum=args.rmsprop_momentum)
        self.gatherer = BatchGatherer(self.store, self.device,
                                      self.slots_per_batch)
        if self.device.type == "cuda":
            # MIOpen exhaustive-find once per conv shape, then cached
            torch.backends.cudnn.benchmark = True
            from ..parallel.rollout import PinRegistry
            self._pins = PinRegistry()
            pinned = self._pins.pin_store(self.store)
            pinned += self._pins.pin(self.shared_flat.flat)
            self.log.info(f"pinned {pinned/1e6:.1f} MB (rollout store + "
                          f"shared weights)")
            self.publish_stream = torch.cuda.Stream(device=self.device)
            self.publish_event = torch.cuda.Event()
        self._publish_weights()
        self.autocast_dtype = (torch.bfloat16 if args.dtype == "bf16" and
                               self.device.type == "cuda" else None)

        # hipGraph capture happens HERE — before the inference worker
        # process exists.  Round-1 finding: concurrent HIP submissions from
        # another process during stream capture abort the other process's
        # HSA queue on ROCm 7.2, so capture-from-a-synthetic-batch precedes
        # start_actors() (callers: bench.py, train()).  The pause handshake
        # remains as a fallback for the lazy capture path.
        if self.use_graph and self.device.type == "cuda" and not self._started:
            try:
                self._capture_graph()
            except Exception as e:  # capture is an optimization, not a
                # correctness requirement: fall back to the eager step
                self.log.warning(f"hipGraph capture failed ({e!r}); "
                                 f"falling back to the eager learner step")
                self._graphed = None
                self.use_graph = False
        self._setup_remote_server()

    def _capture_graph(self) -> None:
        from .graphed import GraphedImpalaStep
        args = self.args
        T, B = args.rollout_length, args.batch_size
        dev = self.device
        g = torch.Generator(device="cpu").manual_seed(0)
        synth = {
            "obs": torch.randint(0, 256, (T + 1, B, *self.obs_shape),
                                 dtype=torch.uint8, device=dev),
            "reward": torch.randn(T + 1, B, generator=g).to(dev),
            "done": (torch.rand(T + 1, B, generator=g) < 0.01).to(dev),
            "last_action": torch.randint(
                0, self.num_actions, (T + 1, B), generator=g).to(dev),
            "action": torch.randint(
                0, self.num_actions, (T + 1, B), generator=g).to(dev),
            "logits": torch.randn(T + 1, B, self.num_actions,
                                  generator=g).to(dev),
        }
        if args.use_lstm:
            synth["core_state"] = torch.zeros(
                2, 2, B, self.store.lstm_hidden, device=dev)
        self.log.info("capturing learner step into a hipGraph "
                      "(pre-actor-start) …")
        self._graphed = GraphedImpalaStep(
            self.learner_model, self.flat.flat_grad, self._loss_kwargs(),
            synth, args.use_lstm, self.autocast_dtype,
            args.reward_clipping == "abs_one", args.discounting)

    def _setup_remote_server(self) -> None:
        # optional TCP server for remote-node actor farms
        args = self.args
        self.remote_server = None
        if args.remote_actor_slots > 0:
            import queue as _q
            from ..parallel.remote_actors import RemoteSlotServer
            self._free_remote_q = _q.Queue()
            for s in range(self.local_slots, self._num_buffers):
                self._free_remote_q.put(s)
            lstm_hidden = self.store.lstm_hidden
            config = {"env_id": args.env_id,
                      "rollout_length": args.rollout_length,
                      "envs_per_actor": args.envs_per_actor,
                      "obs_shape": list(self.obs_shape),
                      "num_actions": self.num_actions,
                      "lstm_hidden": lstm_hidden, "seed": args.seed}
            self.remote_server = RemoteSlotServer(
                self.store, self.full_q, self._free_remote_q,
                self.shared_flat.flat, config, port=args.remote_port)
            self.remote_server.publish_weights()
            self.log.info(f"remote actor server on port "
                          f"{self.remote_server.port}")

    def _pause_inference(self, pause: bool, timeout_s: float = 30.0) -> None:
        """Quiesce the inference worker's HIP queue around graph capture
        (process mode only: thread-mode capture uses thread_local scope)."""
        if self.inference != "gpu":
            return
        if getattr(self.args, "inference_worker", "thread") == "thread":
            return
        if pause:
            self.inf_pause.value = 1
            deadline = time.time() + timeout_s
            while not self.inf_paused_ack.value and time.time() < deadline:
                time.sleep(0.005)
        else:
            self.inf_pause.value = 0

    @torch.no_grad()
    def _publish_weights(self) -> None:
        """learner flat → shared CPU flat (one D2H memcpy on the publish
        stream; CPU actors alias the buffer, the GPU inference worker
        reloads it when the version counter bumps)."""
        if self.device.type == "cuda":
            self.publish_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self.publish_stream):
                self.shared_flat.flat.copy_(self.flat.flat, non_blocking=True)
            self.publish_event.record(self.publish_stream)
        else:
            self.shared_flat.flat.copy_(self.flat.flat)
        if self.inference == "gpu":
            # version bumps when the D2H publish is ENQUEUED, not complete:
            # a reader racing the copy may see a torn weight mix for one
            # round.  That is deliberate — IMPALA only needs the BEHAVIOR
            # logits it records to match the ones it sampled from (they
            # always do; V-trace corrects any μ), and fencing here would
            # put a host sync on every learn step.
            with self.weights_version.get_lock():
                self.weights_version.value += 1

    # -- core step ---------------------------------------------------------
    def _get_full_slot(self, timeout_s: float = 30.0, max_wait_s: float = 600.0):
        """full_q.get with a liveness watchdog: a dead actor or inference
        worker turns a silent eternal hang into a fast, explicit error."""
        import queue as _queue
        waited = 0.0
        while True:
            try:
                return self.full_q.get(timeout=timeout_s)
            except _queue.Empty:
                waited += timeout_s
                dead = [p.name for p in self.actors if not p.is_alive()]
                if dead:
                    raise RuntimeError(f"actor process(es) died: {dead}")
                if (self.inference_proc is not None
                        and not self.inference_proc.is_alive()):
                    raise RuntimeError("GPU inference worker died")
                if waited >= max_wait_s:
                    raise RuntimeError(
                        f"no rollout slot arrived in {waited:.0f}s "
                        f"(actors alive but not producing)")

    def _start_prefetch(self):
        slot_ids = [self._get_full_slot()
                    for _ in range(self.slots_per_batch)]
        self.timings.time("dequeue")
        token = self.gatherer.start(slot_ids)
        self.timings.time("gather_start")
        return (token, slot_ids)

    def next_batch(self) -> Dict[str, torch.Tensor]:
        """Finish the pending prefetch (or do a cold gather) and kick off
        the next one — batch N+1's H2D copies overlap batch N's compute."""
        if self._pending is None:
            self._pending = self._start_prefetch()
        token, slot_ids = self._pending
        batch = self.gatherer.finish(token)
        self.timings.time("gather_wait")
        for s in slot_ids:
            if s >= self.local_slots:  # reserved remote-upload slot
                self._free_remote_q.put(s)
            else:
                self.free_q.put(s)
        self._pending = self._start_prefetch()
        return batch

    def _loss_kwargs(self) -> Dict[str, float]:
        args = self.args
        return dict(clip_rho_threshold=args.clip_rho_threshold,
                    clip_c_threshold=1.0,
                    clip_pg_rho_threshold=args.clip_pg_rho_threshold,
                    baseline_cost=args.baseline_cost,
                    entropy_cost=args.entropy_cost)

    def _eager_fwd_bwd(self, batch):
        args = self.args
        model = self.learner_model
        self.flat.flat_grad.zero_()
        inputs = {"obs": batch["obs"], "reward": batch["reward"],
                  "done": batch["done"], "last_action": batch["last_action"]}
        core_state = ()
        if args.use_lstm:
            cs = batch["core_state"]
            core_state = (cs[0], cs[1])
        if self.autocast_dtype is not None:
            with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
                out, _ = model(inputs, core_state, greedy=True)
        else:
            out, _ = model(inputs, core_state, greedy=True)
        logits = out["policy_logits"].float()
        baseline = out["baseline"].float()
        rewards = batch["reward"][1:]
        if args.reward_clipping == "abs_one":
            rewards = torch.clamp(rewards, -1, 1)
        discounts = (~batch["done"][1:]).float() * args.discounting
        total, comps, _ = impala_loss(
            batch["logits"][:-1], logits[:-1], batch["action"][:-1],
            rewards, discounts, baseline[:-1], baseline[-1].detach(),
            **self._loss_kwargs())
        self.timings.time("forward")
        total.backward()
        self.timings.time("backward")
        return total.detach(), comps

    def learn_step(self, batch: Dict[str, torch.Tensor]) -> Dict[str, float]:
        args = self.args
        self.learner_model.train()
        if self.use_graph:
            if self._graphed is None:
                from .graphed import GraphedImpalaStep
                self.log.info("capturing learner step into a hipGraph …")
                self._pause_inference(True)
                try:
                    self._graphed = GraphedImpalaStep(
                        self.learner_model, self.flat.flat_grad,
                        self._loss_kwargs(), batch, args.use_lstm,
                        self.autocast_dtype,
                        args.reward_clipping == "abs_one", args.discounting)
                finally:
                    self._pause_inference(False)
            total, comps = self._graphed.run(batch)
            self.timings.time("graph_replay")
        else:
            total, comps = self._eager_fwd_bwd(batch)

        all_reduce_flat(self.flat.flat_grad, average=True)
        clip_grad_norm_(self.flat.flat_grad, args.max_grad_norm)
        if self.device.type == "cuda":
            # don't overwrite weights while the previous publish D2H reads them
            torch.cuda.current_stream().wait_event(self.publish_event)
        # linear lr decay to min_learning_rate over total_steps (the
        # reference's monobeast-heritage schedule)
        frac = min(self.global_step / max(args.total_steps, 1), 1.0)
        lr = max(args.learning_rate * (1.0 - frac), args.min_learning_rate)
        self.optimizer.step(self.flat.flat_grad, lr=lr)
        self.timings.time("optimize")
        self._publish_weights()
        if (self.remote_server is not None and
                self.learn_iters % max(args.remote_publish_interval, 1) == 0):
            self.remote_server.publish_weights()
        self.timings.time("publish")

        self.gatherer.mark_consumed()
        self.learn_iters += 1
        self.global_step += args.rollout_length * args.batch_size
        # stats stay on-device: converting forces a host sync, so callers
        # float() them only when they actually log
        return {"total_loss": total.detach(), "pg_loss": comps[0],
                "baseline_loss": comps[1], "entropy_loss": comps[2]}

    def train_iteration(self) -> Dict[str, torch.Tensor]:
        self.timings.reset()
        batch = self.next_batch()
        return self.learn_step(batch)

    def reset_timings(self) -> None:
        self.timings = Timings()

    # -- driver ------------------------------------------------------------
    def train(self) -> None:
        args = self.args
        if self._mp_ctx == "spawn":
            # device init first so graph capture precedes the inference
            # worker; spawned children don't inherit the HIP runtime
            self.setup_learner()
            self.start_actors()
        else:
            self.start_actors()
            self.setup_learner()
        ckpt_path = args.checkpoint_path or os.path.join(
            args.output_dir, "model.tar")
        last_ckpt = time.time()
        last_log = time.time()
        last_step = 0
        stats: Dict[str, float] = {}
        try:
            while self.global_step < args.total_steps:
                stats = self.train_iteration()
                now = time.time()
                if self.rank == 0 and now - last_log > 5.0:
                    sps = (self.global_step - last_step) / (now - last_log)
                    last_log, last_step = now, self.global_step
                    rets = self._drain_episode_returns()
                    ret_str = (f" ret={sum(rets)/len(rets):.2f}" if rets else "")
                    self.log.info(
                        f"step {self.global_step} SPS {sps:,.0f} "
                        f"loss {float(stats['total_loss']):.3f}{ret_str}")
                if (self.rank == 0 and not args.disable_checkpoint and
                        now - last_ckpt > args.checkpoint_interval_s):
                    self.save(ckpt_path)
                    last_ckpt = now
        finally:
            if self.rank == 0 and not args.disable_checkpoint:
                self.save(ckpt_path)
            self.shutdown()

    @torch.no_grad()
    def evaluate(self, num_episodes: int = 3,
                 max_steps: int = 10_000) -> Dict[str, float]:
        """Greedy-policy evaluation with the current published weights on a
        fresh single env (reference test-mode parity; runs on the CPU actor
        model so it never perturbs the learner)."""
        from ..envs.registry import make_env
        deepmind = self.args.env_id not in ("synthetic-atari",)
        env = make_env(self.args.env_id, seed=self.args.seed + 4242,
                       deepmind_wrap=deepmind)
        model = self.actor_model
        model.eval()
        returns, lengths = [], []
        for ep in range(num_episodes):
            obs, _ = env.reset()
            state = model.initial_state(1)
            reward, done, last_action = 0.0, True, 0
            total, steps = 0.0, 0
            while steps < max_steps:
                inputs = {
                    "obs": torch.from_numpy(obs).view(1, 1, *obs.shape),
                    "reward": torch.tensor([[reward]], dtype=torch.float32),
                    "done": torch.tensor([[done]]),
                    "last_action": torch.tensor([[last_action]]),
                }
                out, state = model(inputs, state, greedy=True)
                a = int(out["action"].item())
                obs, reward, term, trunc, _ = env.step(a)
                total += reward
                steps += 1
                done = term or trunc
                last_action = a
                if done:
                    break
            returns.append(total)
            lengths.append(steps)
        env.close()
        import numpy as np
        return {"reward_mean": float(np.mean(returns)),
                "reward_std": float(np.std(returns)),
                "length_mean": float(np.mean(lengths))}

    def _drain_episode_returns(self) -> List[float]:
        rets: List[float] = []
        try:
            while True:
                rets.extend(self.episode_q.get_nowait())
        except Exception:
            pass
        return rets

    def save(self, path: str) -> None:
        """IMPALA-format checkpoint (model.tar keys, impala_atari.py:503)."""
        save_checkpoint(path, model=self.learner_model,
                        optimizer=self.optimizer,
                        hparam=vars(self.args),
                        extra={"global_step": self.global_step})

    def load(self, path: str) -> None:
        ckpt = load_checkpoint(path, map_location=self.device)
        # load_state_dict copies in-place into the existing params, which
        # are views of the flat buffer — the flat pair (and any captured
        # graph / optimizer state bound to it) stays valid
        self.learner_model.load_state_dict(ckpt["model_state_dict"])
        opt_sd = ckpt.get("optimizer_state_dict")
        if opt_sd:
            self.optimizer.load_state_dict(opt_sd)
        self.global_step = int(ckpt.get("global_step", 0))
        self._publish_weights()

    def shutdown(self) -> None:
        self.stop_event.set()
        pills = len(self.actors) * (2 if self.double_buffer else 1)
        for _ in range(pills):
            self.free_q.put(None)  # poison pills (impala_atari.py:480)
        if self.inference == "gpu":
            for s in self.inf_sems:
                s.release()  # unblock any actor waiting on a response
            try:
                self.inf_req_q.put_nowait(None)
            except Exception:
                pass
        for p in self.actors:
            p.join(timeout=2.0)
            if p.is_alive():
                p.terminate()
        self.actors.clear()
        if self.inference_proc is not None:
            self.inference_proc.join(timeout=2.0)
            if (self.inference_proc.is_alive()
                    and hasattr(self.inference_proc, "terminate")):
                self.inference_proc.terminate()  # Process only; a Thread
                # is daemon and dies with us
            self.inference_proc = None
        if getattr(self, "remote_server", None) is not None:
            self.remote_server.close()
            self.remote_server = None
        if getattr(self, "_pins", None) is not None:
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            self._pins.unpin_all()
            self._pins = None
