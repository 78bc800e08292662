"""Centralized GPU inference for actors (SEED-RL-style).

The reference runs per-actor CPU inference against a shared-memory model
(impala_atari.py:196-198).  That burns one CPU core per ~500 env-steps/s;
on an 8-GPU node the cores run out long before the learners do.  The
MI355X-native alternative: actor processes are pure env-steppers, and each
learner rank runs ONE inference worker process that batches all its actors'
observation batches through the policy on the rank's GPU:

- per-actor request/response slots in shared memory (hipHostRegistered by
  the worker → async H2D/D2H DMA);
- request fan-in over an mp.Queue; responses signalled per-actor via
  semaphores;
- the policy's LSTM states for every actor env live on the GPU inside the
  worker; actors never see them.  At rollout-slot start the actor sets
  `want_state`, and the worker writes the pre-step (h,c) snapshot back so
  the learner can initialize its unroll (replaces the reference's
  create_rnn_state_buffers, impala_atari.py:108-120);
- weights refresh from the learner-published shared CPU flat buffer
  (parallel/flat.py), gated by a version counter — one H2D flat copy.

Both processes share the GPU: inference forwards interleave with learner
kernels (separate HIP contexts timeslice the CU array).
"""

from __future__ import annotations

import queue
import time
from typing import Dict, List, Tuple

import torch


class InferenceSlots:
    """Shared-memory request/response slots, one row per actor."""

    def __init__(self, num_actors: int, envs_per_actor: int,
                 obs_shape: Tuple[int, ...], num_actions: int,
                 lstm_layers: int, lstm_hidden: int):
        A, E = num_actors, envs_per_actor
        self.num_actors, self.envs_per_actor = A, E
        self.num_actions = num_actions
        self.lstm_layers, self.lstm_hidden = lstm_layers, lstm_hidden

        def shared(shape, dtype):
            t = torch.zeros(shape, dtype=dtype)
            t.share_memory_()
            return t

        self.obs = shared((A, E, *obs_shape), torch.uint8)
        self.reward = shared((A, E), torch.float32)
        self.done = shared((A, E), torch.bool)
        self.last_action = shared((A, E), torch.int64)
        self.want_state = shared((A,), torch.int32)
        self.action = shared((A, E), torch.int64)
        self.logits = shared((A, E, num_actions), torch.float32)
        if lstm_hidden > 0:
            self.core_state = shared((A, 2, lstm_layers, E, lstm_hidden),
                                     torch.float32)
        else:
            self.core_state = None

    def tensors(self):
        ts = [self.obs, self.reward, self.done, self.last_action,
              self.want_state, self.action, self.logits]
        if self.core_state is not None:
            ts.append(self.core_state)
        return ts


class RemotePolicy:
    """Actor-side client: request an inference round from the worker.

    Split request/wait API so a double-buffered actor can overlap one env
    group's inference round with the other group's env stepping (SEED-RL's
    latency-hiding pattern); ``__call__`` is the blocking convenience."""

    def __init__(self, actor_id: int, slots: InferenceSlots, req_q, sem):
        self.aid = actor_id
        self.slots = slots
        self.req_q = req_q
        self.sem = sem

    def request(self, obs, reward, done, last_action,
                want_state: bool = False) -> None:
        a = self.aid
        s = self.slots
        s.obs[a].copy_(obs)
        s.reward[a].copy_(reward)
        s.done[a].copy_(done)
        s.last_action[a].copy_(last_action)
        s.want_state[a] = 1 if want_state else 0
        self._want_state = want_state
        self.req_q.put(a)

    def wait(self):
        a = self.aid
        s = self.slots
        self.sem.acquire()
        state = s.core_state[a].clone() if (self._want_state and
                                            s.core_state is not None) else None
        return s.action[a].clone(), s.logits[a].clone(), state

    def __call__(self, obs, reward, done, last_action,
                 want_state: bool = False):
        self.request(obs, reward, done, last_action, want_state)
        return self.wait()


def inference_worker(device_index: int, model_kwargs: Dict, slots: InferenceSlots,
                     shared_flat: torch.Tensor, version: "mp.Value",
                     req_q, sems: List, stop_event, max_batch_actors: int = 0,
                     seed: int = 0, pause_flag=None, paused_ack=None,
                     as_thread: bool = False):
    """Inference worker main — either a dedicated process (own HIP context)
    or a THREAD of the learner process on its own stream (``as_thread``).

    The thread mode exists because hipGraph execution in one process with
    concurrent HIP submissions from another is fragile on ROCm 7.2 (r1/r2
    measured HSA_STATUS_ERROR_EXCEPTION aborts both during capture and
    during replay); a same-context second stream plus
    ``capture_error_mode="thread_local"`` is the supported shape for this.
    """
    from ..models.atari import AtariNet
    from .flat import FlatParams
    from .rollout import pin_tensor

    import os
    device = torch.device(f"cuda:{device_index}")
    if not as_thread:
        torch.manual_seed(seed)
        torch.cuda.set_device(device)
    stream = torch.cuda.Stream(device=device) if as_thread else None
    done_ev = torch.cuda.Event()
    model = AtariNet(**model_kwargs).to(device)
    model.train()  # multinomial action sampling
    flat = FlatParams(model, device=device)
    for t in slots.tensors():
        pin_tensor(t)
    if not as_thread:  # the learner already pinned it in thread mode
        pin_tensor(shared_flat)
    # bf16 convs/FC for the behavior forward: the recorded behavior logits
    # are the ones actions are sampled from, so the V-trace correction
    # stays exact; heads/LSTM run fp32 inside AtariNet regardless.
    inf_bf16 = os.environ.get("SCALERL_INF_BF16", "1") != "0"

    A, E = slots.num_actors, slots.envs_per_actor
    nact = slots.num_actions
    use_lstm = slots.core_state is not None
    if use_lstm:
        L, H = slots.lstm_layers, slots.lstm_hidden
        h_all = torch.zeros(L, A * E, H, device=device)
        c_all = torch.zeros(L, A * E, H, device=device)
    max_batch_actors = max_batch_actors or A
    seen_version = -1

    # device staging for a batched round
    obs_d = torch.empty((A, E, *slots.obs.shape[2:]), dtype=torch.uint8,
                        device=device)
    rew_d = torch.empty((A, E), device=device)
    done_d = torch.empty((A, E), dtype=torch.bool, device=device)
    lastact_d = torch.empty((A, E), dtype=torch.int64, device=device)

    nonlocal_state = {"seen_version": seen_version}

    def run_round(ids):
        if version.value != nonlocal_state["seen_version"]:
            nonlocal_state["seen_version"] = version.value
            flat.load_from(shared_flat, non_blocking=False)

        # FIXED-SHAPE round: forward ALL A actor slots every time (only
        # requesting actors' inputs are refreshed and only their states/
        # outputs are written back).  A varying batch size would retrigger
        # MIOpen find per new shape and preclude graph capture.
        for a in ids:
            obs_d[a].copy_(slots.obs[a], non_blocking=True)
            rew_d[a].copy_(slots.reward[a], non_blocking=True)
            done_d[a].copy_(slots.done[a], non_blocking=True)
            lastact_d[a].copy_(slots.last_action[a], non_blocking=True)
        inputs = {
            "obs": obs_d.reshape(1, A * E, *slots.obs.shape[2:]),
            "reward": rew_d.reshape(1, A * E),
            "done": done_d.reshape(1, A * E),
            "last_action": lastact_d.reshape(1, A * E),
        }
        state = ()
        snap = {}
        if use_lstm:
            for a in ids:
                if slots.want_state[a]:
                    snap[a] = (h_all[:, a * E:(a + 1) * E].clone(),
                               c_all[:, a * E:(a + 1) * E].clone())
            # no clone: the LSTM builds fresh state tensors, and only
            # requesting actors' slices are copied back below — cloning
            # h/c here cost ~50 MB of D2D per round at A=24,E=128
            state = (h_all, c_all)
        if inf_bf16:
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                out, new_state = model(inputs, state)
        else:
            out, new_state = model(inputs, state)
        action = out["action"].view(A, E)
        logits = out["policy_logits"].view(A, E, nact)
        for a in ids:
            if use_lstm:
                sl = slice(a * E, (a + 1) * E)
                h_all[:, sl] = new_state[0][:, sl]
                c_all[:, sl] = new_state[1][:, sl]
            slots.action[a].copy_(action[a], non_blocking=True)
            slots.logits[a].copy_(logits[a], non_blocking=True)
            if a in snap:
                hj, cj = snap[a]
                slots.core_state[a][0].copy_(hj, non_blocking=True)
                slots.core_state[a][1].copy_(cj, non_blocking=True)
        # fence only OUR stream before releasing the actors (a device-wide
        # synchronize in thread mode would stall the learner stream too)
        done_ev.record(stream if stream is not None
                       else torch.cuda.current_stream(device))
        done_ev.synchronize()

    import contextlib
    with torch.no_grad():
        while not stop_event.is_set():
            if pause_flag is not None and pause_flag.value:
                # learner is hipGraph-capturing on this device: quiesce our
                # HIP queue entirely until it clears the flag (concurrent
                # submissions from another process during stream capture
                # fault the HSA queue on ROCm 7.2; thread mode makes this
                # path unnecessary but it is kept for process mode)
                torch.cuda.synchronize()
                if paused_ack is not None:
                    paused_ack.value = 1
                while pause_flag.value and not stop_event.is_set():
                    time.sleep(0.002)
                if paused_ack is not None:
                    paused_ack.value = 0
                continue
            try:
                first = req_q.get(timeout=0.2)
            except queue.Empty:
                continue
            if first is None:
                break
            ids = [first]
            while len(ids) < max_batch_actors:
                try:
                    nxt = req_q.get_nowait()
                except queue.Empty:
                    break
                if nxt is None:
                    stop_event.set()
                    break
                ids.append(nxt)
            sctx = (torch.cuda.stream(stream) if stream is not None
                    else contextlib.nullcontext())
            with sctx:
                run_round(ids)
            for a in ids:
                sems[a].release()
