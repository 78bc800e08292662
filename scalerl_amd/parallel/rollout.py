"""IMPALA rollout transport: shared-memory trajectory slots, actor workers,
and the learner-side batch gatherer with side-stream H2D copies.

MI355X redesign of the reference's buffer machinery
(impala_atari.py:122-151 create_buffers, :153-220 get_action actor loop,
:222-268 get_batch):

- slots are Structure-of-Arrays shared-memory tensors [S, T+1, E, ...] with
  E envs per actor (the reference is E=1; vectorizing the actor batches its
  CPU inference and amortizes queue latency);
- the learner hipHostRegisters the shared region so H2D copies are true
  async DMA on a dedicated side stream (replaces `.to(device,
  non_blocking=True)` at impala_atari.py:259-266);
- free/full index queues carry slot ids (free: SimpleQueue; full: a
  timeout-capable Queue the learner watchdogs) — latency is amortized over
  T*E env steps per slot;
- weight publication: actors' model params alias ONE shared flat fp32
  buffer (parallel/flat.py); the learner publishes with a single flat copy
  (replaces load_state_dict at impala_atari.py:348).

Wire format per slot row t in [0, T]:
  obs[t], reward[t], done[t], last_action[t]  — env output entering step t
  action[t], logits[t]                        — policy decision taken at t
  core_state                                  — LSTM (h,c) BEFORE row 0
Learner consumes rows 0..T-1 of (action, logits), rows 1..T of (reward,
done), model values on rows 0..T (row T bootstraps).
"""

from __future__ import annotations

import ctypes
import ctypes.util
import queue
from typing import Dict, List, Tuple

import numpy as np
import torch


class RolloutStore:
    """Shared-memory SoA trajectory slots."""

    def __init__(self, num_slots: int, rollout_length: int, envs_per_slot: int,
                 obs_shape: Tuple[int, ...], num_actions: int,
                 lstm_layers: int = 2, lstm_hidden: int = 0,
                 obs_dtype: torch.dtype = torch.uint8):
        S, T1, E = num_slots, rollout_length + 1, envs_per_slot
        self.num_slots = S
        self.rollout_length = rollout_length
        self.envs_per_slot = E
        self.num_actions = num_actions
        self.lstm_layers = lstm_layers
        self.lstm_hidden = lstm_hidden

        def shared(shape, dtype):
            t = torch.zeros(shape, dtype=dtype)
            t.share_memory_()
            return t

        self.obs = shared((S, T1, E, *obs_shape), obs_dtype)
        self.reward = shared((S, T1, E), torch.float32)
        self.done = shared((S, T1, E), torch.bool)
        self.last_action = shared((S, T1, E), torch.int64)
        self.action = shared((S, T1, E), torch.int64)
        self.logits = shared((S, T1, E, num_actions), torch.float32)
        self.episode_return = shared((S, T1, E), torch.float32)
        if lstm_hidden > 0:
            self.core_state = shared((S, 2, lstm_layers, E, lstm_hidden),
                                     torch.float32)
        else:
            self.core_state = None

    def fields(self) -> Dict[str, torch.Tensor]:
        d = dict(obs=self.obs, reward=self.reward, done=self.done,
                 last_action=self.last_action, action=self.action,
                 logits=self.logits, episode_return=self.episode_return)
        if self.core_state is not None:
            d["core_state"] = self.core_state
        return d

    def nbytes(self) -> int:
        return sum(t.numel() * t.element_size() for t in self.fields().values())


def _libhip():
    try:
        return ctypes.CDLL("libamdhip64.so")
    except OSError:
        return None


class PinRegistry:
    """Tracks hipHostRegister'd ranges so they can be unregistered BEFORE
    the underlying shared-memory segments are freed — a stale registration
    aborts the HIP runtime on a later allocation/copy."""

    def __init__(self):
        self._ranges = []  # (ptr, nbytes)

    def pin(self, t: torch.Tensor) -> int:
        lib = _libhip()
        if lib is None or not torch.cuda.is_available():
            return 0
        nbytes = t.numel() * t.element_size()
        ptr = t.data_ptr()
        # hipHostRegisterDefault = 0
        if lib.hipHostRegister(ctypes.c_void_p(ptr), ctypes.c_size_t(nbytes),
                               ctypes.c_uint(0)) == 0:
            self._ranges.append((ptr, nbytes))
            return nbytes
        return 0

    def pin_store(self, store: "RolloutStore") -> int:
        return sum(self.pin(t) for t in store.fields().values())

    def unpin_all(self) -> None:
        lib = _libhip()
        if lib is None:
            return
        for ptr, _ in self._ranges:
            lib.hipHostUnregister(ctypes.c_void_p(ptr))
        self._ranges.clear()


def pin_tensor(t: torch.Tensor, registry: "PinRegistry" = None) -> int:
    """hipHostRegister one CPU tensor in this process (0 on failure)."""
    reg = registry or PinRegistry()
    return reg.pin(t)


def pin_store(store: RolloutStore, registry: "PinRegistry" = None) -> int:
    """hipHostRegister every shared slot tensor in THIS process so H2D
    copies from them are async DMA.  Registration is per-process; call it
    from the learner after CUDA init.  Returns bytes pinned (0 = no-op)."""
    reg = registry or PinRegistry()
    return reg.pin_store(store)


class LocalPolicy:
    """CPU inference against the shared-flat-aliased model (the reference's
    per-actor inference mode, impala_atari.py:196-198).  Same call contract
    as :class:`scalerl_amd.parallel.inference.RemotePolicy`."""

    def __init__(self, model, num_envs: int):
        self.model = model
        if model.use_lstm:
            self.core_state = model.initial_state(num_envs)
        else:
            self.core_state = ()

    @torch.no_grad()
    def __call__(self, obs, reward, done, last_action, want_state=False):
        snap = None
        if want_state and self.model.use_lstm:
            snap = torch.stack([self.core_state[0], self.core_state[1]])
        inputs = {"obs": obs.unsqueeze(0), "reward": reward.unsqueeze(0),
                  "done": done.unsqueeze(0),
                  "last_action": last_action.unsqueeze(0)}
        out, self.core_state = self.model(inputs, self.core_state)
        return (out["action"].squeeze(0), out["policy_logits"].squeeze(0),
                snap)


class ActorState:
    """Per-actor persistent env state between slots."""

    def __init__(self, env, policy, num_envs: int):
        self.env = env
        self.policy = policy
        obs = env.reset()
        E = num_envs
        self.obs = torch.from_numpy(np.ascontiguousarray(obs))
        self.reward = torch.zeros(E)
        self.done = torch.ones(E, dtype=torch.bool)  # episode starts
        self.last_action = torch.zeros(E, dtype=torch.int64)
        self.episode_return = torch.zeros(E)


@torch.no_grad()
def run_rollout(state: ActorState, store: RolloutStore, slot: int) -> int:
    """Fill one slot (T env steps across E envs).  Returns env steps done.

    Row layout (see module docstring): action[t]/logits[t] = decision taken
    at the state in row t; rows written 0..T for env fields, 0..T-1 for
    decisions (row T's decision belongs to the next slot's row 0).
    """
    T = store.rollout_length
    E = store.envs_per_slot

    # row 0 = carried-over env output (the continuation row)
    store.obs[slot, 0].copy_(state.obs)
    store.reward[slot, 0].copy_(state.reward)
    store.done[slot, 0].copy_(state.done)
    store.last_action[slot, 0].copy_(state.last_action)
    store.episode_return[slot, 0].copy_(state.episode_return)

    for t in range(T):
        action, logits, snap = state.policy(
            state.obs, state.reward, state.done, state.last_action,
            want_state=(t == 0 and store.core_state is not None))
        if snap is not None:
            store.core_state[slot].copy_(snap)
        store.action[slot, t].copy_(action)
        store.logits[slot, t].copy_(logits)

        obs_np, rew_np, done_np = state.env.step(action.numpy())
        # episode accounting: returns BEFORE reset (reward of this step incl.)
        state.episode_return += torch.from_numpy(rew_np)
        ep_ret = state.episode_return.clone()
        done_t = torch.from_numpy(done_np)
        state.episode_return[done_t] = 0.0

        state.obs = torch.from_numpy(np.ascontiguousarray(obs_np))
        state.reward = torch.from_numpy(rew_np)
        state.done = done_t
        state.last_action = action

        row = t + 1
        store.obs[slot, row].copy_(state.obs)
        store.reward[slot, row].copy_(state.reward)
        store.done[slot, row].copy_(state.done)
        store.last_action[slot, row].copy_(action)
        store.episode_return[slot, row].copy_(ep_ret)
    return T * E


def write_row0(state: ActorState, store: RolloutStore, slot: int) -> None:
    store.obs[slot, 0].copy_(state.obs)
    store.reward[slot, 0].copy_(state.reward)
    store.done[slot, 0].copy_(state.done)
    store.last_action[slot, 0].copy_(state.last_action)
    store.episode_return[slot, 0].copy_(state.episode_return)


@torch.no_grad()
def run_rollout_db(states, store: RolloutStore, slots) -> int:
    """Double-buffered rollout: two env groups per actor, each with its own
    slot, interleaved so one group's env stepping + slot writes overlap the
    other group's in-flight inference round (SEED-RL latency hiding; the
    single-group path pays the full round latency every step).
    Policies must expose request()/wait() (RemotePolicy)."""
    T = store.rollout_length
    E = store.envs_per_slot
    want0 = store.core_state is not None
    for g in (0, 1):
        write_row0(states[g], store, slots[g])
        st = states[g]
        st.policy.request(st.obs, st.reward, st.done, st.last_action,
                          want_state=want0)
    for t in range(T):
        for g in (0, 1):
            st = states[g]
            slot = slots[g]
            action, logits, snap = st.policy.wait()
            if snap is not None:
                store.core_state[slot].copy_(snap)
            store.action[slot, t].copy_(action)
            store.logits[slot, t].copy_(logits)

            obs_np, rew_np, done_np = st.env.step(action.numpy())
            st.episode_return += torch.from_numpy(rew_np)
            ep_ret = st.episode_return.clone()
            done_t = torch.from_numpy(done_np)
            st.episode_return[done_t] = 0.0
            st.obs = torch.from_numpy(np.ascontiguousarray(obs_np))
            st.reward = torch.from_numpy(rew_np)
            st.done = done_t
            st.last_action = action

            row = t + 1
            store.obs[slot, row].copy_(st.obs)
            store.reward[slot, row].copy_(st.reward)
            store.done[slot, row].copy_(st.done)
            store.last_action[slot, row].copy_(action)
            store.episode_return[slot, row].copy_(ep_ret)
            if t + 1 < T:
                st.policy.request(st.obs, st.reward, st.done,
                                  st.last_action)
    return 2 * T * E


def build_actor_env(env_spec: dict, actor_id: int):
    """Construct an actor's vectorized env from a picklable spec
    (module-level so spawned processes can build it)."""
    env_id = env_spec["env_id"]
    E = env_spec["envs_per_actor"]
    seed = env_spec.get("seed", 0)
    from ..envs.vec_env import SyncVectorEnv
    if env_spec.get("a3c_wrap"):
        # 42×42 normalized A3C preprocessing (a3c_env.create_atari_env)
        from ..envs.a3c_env import create_atari_env
        return SyncVectorEnv([
            (lambda i=i: create_atari_env(
                env_id, seed=seed * 1000 + actor_id * E + i))
            for i in range(E)])
    if env_id == "synthetic-atari":
        from ..envs.synthetic import SyntheticAtariVecEnv
        return SyntheticAtariVecEnv(E, seed=seed * 1000 + actor_id)
    from ..envs.registry import make_env
    return SyncVectorEnv([
        (lambda i=i: make_env(env_id, seed=seed * 1000 + actor_id * E + i,
                              deepmind_wrap=env_spec.get("deepmind_wrap", True)))
        for i in range(E)])


def actor_loop(actor_id: int, env_spec: dict, store: RolloutStore,
               free_q, full_q, stop_event, step_counter,
               episode_queue=None, seed: int = 0, actor_model=None,
               inf_slots=None, inf_req_q=None, inf_sem=None,
               torch_threads: int = 1):
    """Actor process main (reference: impala_atari.py:153-220).

    Runs under the *spawn* start method (the learner initializes HIP, so
    forked children would inherit a poisoned runtime): every argument is
    picklable; shared tensors travel as shm handles.  Policy: RemotePolicy
    when inference-slot plumbing is given, else LocalPolicy on
    ``actor_model`` (whose params alias the published shared flat buffer).
    Poison pill: a None on free_q exits.
    """
    torch.manual_seed(seed + actor_id)
    torch.set_num_threads(torch_threads)
    env = build_actor_env(env_spec, actor_id)
    if inf_slots is not None:
        from .inference import RemotePolicy
        policy = RemotePolicy(actor_id, inf_slots, inf_req_q, inf_sem)
    else:
        actor_model.eval()
        policy = LocalPolicy(actor_model, store.envs_per_slot)
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
                # completed-episode returns for logging (best effort)
                mask = store.done[slot, 1:].numpy()
                if mask.any():
                    rets = store.episode_return[slot, 1:].numpy()[mask]
                    try:
                        episode_queue.put_nowait(rets.tolist())
                    except queue.Full:
                        pass
    except KeyboardInterrupt:
        pass


def actor_loop_db(actor_id: int, env_spec: dict, store: RolloutStore,
                  free_q, full_q, stop_event, step_counter,
                  inf_slots=None, inf_req_q=None, inf_sems=(),
                  row_ids=(), episode_queue=None, seed: int = 0,
                  torch_threads: int = 1):
    """Double-buffered actor process main: two env groups (rows
    ``row_ids`` in the inference slots), each filling its own rollout slot,
    interleaved via :func:`run_rollout_db`.  Poison pill: None on free_q."""
    import queue as _q
    from .inference import RemotePolicy
    torch.manual_seed(seed + actor_id)
    torch.set_num_threads(torch_threads)
    states = []
    for g in (0, 1):
        spec = dict(env_spec)
        spec["seed"] = env_spec.get("seed", 0) + 500_000 * g
        env = build_actor_env(spec, actor_id)
        policy = RemotePolicy(row_ids[g], inf_slots, inf_req_q, inf_sems[g])
        states.append(ActorState(env, policy, store.envs_per_slot))
    try:
        while not stop_event.is_set():
            slot_a = free_q.get()
            if slot_a is None:
                break
            slot_b = free_q.get()
            if slot_b is None:
                break
            steps = run_rollout_db(states, store, (slot_a, slot_b))
            full_q.put(slot_a)
            full_q.put(slot_b)
            with step_counter.get_lock():
                step_counter.value += steps
            if episode_queue is not None:
                rets = []
                for slot in (slot_a, slot_b):
                    mask = store.done[slot, 1:].numpy()
                    if mask.any():
                        rets.extend(
                            store.episode_return[slot, 1:].numpy()[mask]
                            .tolist())
                if rets:
                    try:
                        episode_queue.put_nowait(rets)
                    except _q.Full:
                        pass
    except KeyboardInterrupt:
        pass


class BatchGatherer:
    """Learner-side: gather K slots into device batch tensors [T+1, K*E, ...]
    via side-stream async H2D copies (the device boundary of the pipeline,
    replacing impala_atari.py:248-266).

    Double-buffered: ``start(slot_ids)`` launches the copies into the next
    staging set and returns a token; ``finish(token)`` host-syncs the copy
    event (slots are then recyclable) and returns the batch views.  The
    trainer starts batch N+1's copies before computing batch N, so H2D
    rides the side stream under the learner's kernels.
    """

    NBUF = 3  # 3 staging sets: copy into buf i+1 never fences on the
    # learner step that is currently executing (2 would serialize them)

    def __init__(self, store: RolloutStore, device: torch.device,
                 slots_per_batch: int):
        self.store = store
        self.device = device
        self.K = slots_per_batch
        self.is_cuda = device.type == "cuda"
        K = self.K
        self._next_buf = 0
        if self.is_cuda:
            self.stream = torch.cuda.Stream(device=device)
            self.staging = [
                {name: torch.empty((K, *t.shape[1:]), dtype=t.dtype,
                                   device=device)
                 for name, t in store.fields().items()}
                for _ in range(self.NBUF)
            ]
            self.events = [torch.cuda.Event() for _ in range(self.NBUF)]
            # recorded on the MAIN stream when the learner is done reading a
            # staging set; the copy stream waits on it before overwriting
            self.consumed = [torch.cuda.Event() for _ in range(self.NBUF)]
            self._last_buf = None

    def start(self, slot_ids: List[int]):
        """Launch async H2D of the given slots; returns a token."""
        if not self.is_cuda:
            return ("cpu", list(slot_ids))
        buf = self._next_buf
        self._next_buf = (buf + 1) % self.NBUF
        with torch.cuda.stream(self.stream):
            # don't overwrite a staging set the learner might still read
            self.stream.wait_event(self.consumed[buf])
            staging = self.staging[buf]
            for name, t in self.store.fields().items():
                dst = staging[name]
                for j, s in enumerate(slot_ids):
                    dst[j].copy_(t[s], non_blocking=True)
            self.events[buf].record(self.stream)
        return ("cuda", buf)

    def finish(self, token) -> Dict[str, torch.Tensor]:
        """Host-sync the token's copies (slots reusable afterwards) and
        return the batch in [T+1, B, ...] layout."""
        kind, payload = token
        if kind == "cpu":
            batch = {}
            for name, t in self.store.fields().items():
                stacked = torch.stack([t[s] for s in payload])
                batch[name] = self._to_batch_layout(name, stacked)
            return batch
        buf = payload
        self.events[buf].synchronize()
        torch.cuda.current_stream(self.device).wait_event(self.events[buf])
        self._last_buf = buf
        return {name: self._to_batch_layout(name, self.staging[buf][name])
                for name in self.staging[buf]}

    def mark_consumed(self) -> None:
        """Record (on the main stream) that all work reading the last
        finished staging set has been enqueued; called by the learner after
        its step so the next prefetch into that set can proceed."""
        if self.is_cuda and self._last_buf is not None:
            self.consumed[self._last_buf].record(
                torch.cuda.current_stream(self.device))

    def gather(self, slot_ids: List[int]) -> Dict[str, torch.Tensor]:
        """Synchronous convenience path (tests, non-pipelined callers)."""
        return self.finish(self.start(slot_ids))

    def copies_done(self) -> None:
        """Back-compat no-op: finish() already syncs the copy event."""

    def _to_batch_layout(self, name: str, stacked: torch.Tensor) -> torch.Tensor:
        """[K, T+1, E, ...] → [T+1, K*E, ...]; core_state → [L, K*E, H] x2."""
        K = stacked.shape[0]
        if name == "core_state":
            # [K, 2, L, E, H] → [2, L, K*E, H]
            _, _, L, E, H = stacked.shape
            return stacked.permute(1, 2, 0, 3, 4).reshape(2, L, K * E, H).contiguous()
        T1, E = stacked.shape[1], stacked.shape[2]
        rest = stacked.shape[3:]
        return stacked.permute(1, 0, 2, *range(3, stacked.dim())).reshape(
            T1, K * E, *rest).contiguous()
