"""Inference slot/semaphore protocol, CPU-level: a fake worker thread
implements the server side; RemotePolicy is exercised exactly as actors
use it (the real GPU worker is covered by the gpu-marked IMPALA
integration test)."""

import multiprocessing as mp
import threading

import torch

from scalerl_amd.parallel.inference import InferenceSlots, RemotePolicy


def _fake_worker(slots: InferenceSlots, req_q, sems, stop):
    """Deterministic 'policy': action = obs[...,0,0,0] % A; logits = arange
    offset; state snapshot = constant 7s."""
    A = slots.num_actors
    while not stop.is_set():
        try:
            aid = req_q.get(timeout=0.1)
        except Exception:
            continue
        if aid is None:
            break
        obs = slots.obs[aid]
        act = (obs[:, 0, 0, 0].long() + slots.last_action[aid]) % slots.num_actions
        slots.action[aid].copy_(act)
        slots.logits[aid].copy_(
            torch.arange(slots.num_actions).float().repeat(
                slots.envs_per_actor, 1) + float(aid))
        if slots.want_state[aid] and slots.core_state is not None:
            slots.core_state[aid].fill_(7.0)
        sems[aid].release()


def test_remote_policy_roundtrip():
    ctx = mp.get_context("spawn")
    slots = InferenceSlots(num_actors=2, envs_per_actor=3,
                           obs_shape=(4, 8, 8), num_actions=5,
                           lstm_layers=2, lstm_hidden=6)
    req_q = ctx.Queue()
    sems = [ctx.Semaphore(0) for _ in range(2)]
    stop = threading.Event()
    worker = threading.Thread(target=_fake_worker,
                              args=(slots, req_q, sems, stop), daemon=True)
    worker.start()
    try:
        pol0 = RemotePolicy(0, slots, req_q, sems[0])
        pol1 = RemotePolicy(1, slots, req_q, sems[1])
        obs = torch.randint(0, 256, (3, 4, 8, 8), dtype=torch.uint8)
        rew = torch.zeros(3)
        done = torch.zeros(3, dtype=torch.bool)
        last = torch.tensor([1, 2, 3])

        a0, logits0, snap0 = pol0(obs, rew, done, last, want_state=True)
        assert snap0 is not None and (snap0 == 7.0).all()
        expect = (obs[:, 0, 0, 0].long() + last) % 5
        assert (a0 == expect).all()
        assert logits0.shape == (3, 5)
        assert torch.allclose(logits0[0], torch.arange(5).float())

        a1, logits1, snap1 = pol1(obs, rew, done, last, want_state=False)
        assert snap1 is None
        assert torch.allclose(logits1[0], torch.arange(5).float() + 1.0)

        # second round without snapshot: want_state must reset to 0
        _, _, snap0b = pol0(obs, rew, done, last, want_state=False)
        assert snap0b is None
        assert int(slots.want_state[0]) == 0
    finally:
        stop.set()
        req_q.put(None)
        worker.join(timeout=5)
