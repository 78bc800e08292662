"""Asynchronous multiprocess vectorized env with a shared-memory obs block.

Capability parity with the reference's ``AsyncPettingZooVecEnv``
(scalerl/envs/vector/pz_async_vec_env.py:36-541): per-env worker processes
over a Pipe command protocol (reset/step/close), observations written into
ONE shared-memory tensor viewed [num_envs, *obs_shape], an async
step_async/step_wait API with a state machine, per-worker error funneling
(a crashed env worker surfaces its exception in the parent instead of
hanging it).

In the MI355X stack the IMPALA/Ape-X actors use in-process vectorized envs
(SyncVectorEnv / SyntheticAtariVecEnv) because the actor process itself is
the parallelism unit; this class is the general-purpose escape hatch for
envs whose step is expensive enough to want its own process.
"""

from __future__ import annotations

import enum
import multiprocessing as mp
from typing import Callable, Optional, Sequence

import numpy as np
import torch

from .base import Env


class AsyncState(enum.Enum):
    DEFAULT = 0
    WAITING_STEP = 1
    WAITING_RESET = 2


def _async_worker(index: int, env_fn, pipe, obs_block, obs_shape, error_q):
    obs_view = obs_block[index]
    try:
        env = env_fn()
        while True:
            cmd, data = pipe.recv()
            if cmd == "reset":
                obs, info = env.reset(seed=data)
                obs_view.copy_(torch.from_numpy(np.ascontiguousarray(obs)))
                pipe.send(("ok", info))
            elif cmd == "step":
                obs, r, term, trunc, info = env.step(data)
                done = term or trunc
                if done:
                    obs, _ = env.reset()
                obs_view.copy_(torch.from_numpy(np.ascontiguousarray(obs)))
                pipe.send(("ok", (float(r), bool(done), info)))
            elif cmd == "close":
                env.close()
                pipe.send(("ok", None))
                break
    except Exception as e:  # funnel to parent (pz_async_vec_env.py:890-894)
        error_q.put((index, type(e).__name__, str(e)))
        try:
            pipe.send(("error", str(e)))
        except Exception:
            pass


class AsyncVectorEnv:
    def __init__(self, env_fns: Sequence[Callable[[], Env]],
                 context: str = "fork"):
        self.num_envs = len(env_fns)
        probe = env_fns[0]()
        self.observation_space = probe.observation_space
        self.action_space = probe.action_space
        probe.close()
        shape = self.observation_space.shape
        dtype = {np.dtype("uint8"): torch.uint8,
                 np.dtype("float32"): torch.float32}.get(
                     np.dtype(self.observation_space.dtype), torch.float32)
        self.obs_block = torch.zeros((self.num_envs, *shape), dtype=dtype)
        self.obs_block.share_memory_()
        ctx = mp.get_context(context)
        self.error_q = ctx.Queue()
        self.pipes = []
        self.procs = []
        for i, fn in enumerate(env_fns):
            parent, child = ctx.Pipe()
            p = ctx.Process(target=_async_worker,
                            args=(i, fn, child, self.obs_block, shape,
                                  self.error_q),
                            daemon=True)
            p.start()
            self.pipes.append(parent)
            self.procs.append(p)
        self._state = AsyncState.DEFAULT

    def _raise_if_errors(self):
        if not self.error_q.empty():
            idx, etype, msg = self.error_q.get()
            self.close(terminate=True)
            raise RuntimeError(f"env worker {idx} crashed: {etype}: {msg}")

    def _gather(self, timeout: float):
        out = []
        for i, pipe in enumerate(self.pipes):
            if not pipe.poll(timeout):
                self._raise_if_errors()
                raise TimeoutError(f"env worker {i} timed out")
            status, payload = pipe.recv()
            if status == "error":
                self._raise_if_errors()
                raise RuntimeError(f"env worker {i}: {payload}")
            out.append(payload)
        return out

    def reset(self, seed: Optional[int] = None, timeout: float = 60.0):
        assert self._state == AsyncState.DEFAULT
        for i, pipe in enumerate(self.pipes):
            pipe.send(("reset", None if seed is None else seed + i))
        self._state = AsyncState.WAITING_RESET
        self._gather(timeout)
        self._state = AsyncState.DEFAULT
        return self.obs_block.numpy().copy()

    def step_async(self, actions) -> None:
        assert self._state == AsyncState.DEFAULT
        for pipe, a in zip(self.pipes, actions):
            pipe.send(("step", a))
        self._state = AsyncState.WAITING_STEP

    def step_wait(self, timeout: float = 60.0):
        assert self._state == AsyncState.WAITING_STEP
        payloads = self._gather(timeout)
        self._state = AsyncState.DEFAULT
        rewards = np.array([p[0] for p in payloads], dtype=np.float32)
        dones = np.array([p[1] for p in payloads], dtype=bool)
        return self.obs_block.numpy().copy(), rewards, dones

    def step(self, actions):
        self.step_async(actions)
        return self.step_wait()

    def close(self, terminate: bool = False):
        for pipe in self.pipes:
            try:
                if not terminate:
                    pipe.send(("close", None))
            except (BrokenPipeError, OSError):
                pass
        for p in self.procs:
            p.join(timeout=1.0)
            if p.is_alive():
                p.terminate()
