#!/usr/bin/env python3
"""Env-layer throughput harness (parity with the reference's
examples/test_env_throughput.py:15-606, on this framework's env stack):
steps/s for the natively-vectorized synthetic env, SyncVectorEnv and
AsyncVectorEnv across a (num_envs,) grid, printed as a table."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from scalerl_amd.envs import AsyncVectorEnv, CartPoleEnv, make_vect_envs
from scalerl_amd.envs.synthetic import SyntheticAtariEnv, SyntheticAtariVecEnv


def _measure(step_fn, reset_fn, num_envs, steps):
    reset_fn()
    t0 = time.perf_counter()
    for _ in range(steps):
        step_fn()
    dt = time.perf_counter() - t0
    return num_envs * steps / dt


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=300)
    p.add_argument("--grid", type=int, nargs="+", default=[4, 16, 64])
    args = p.parse_args()

    rows = []
    for n in args.grid:
        v = SyntheticAtariVecEnv(n, seed=0)
        acts = np.zeros(n, dtype=np.int64)
        fps = _measure(lambda: v.step(acts), v.reset, n, args.steps)
        rows.append((f"SyntheticAtariVecEnv[{n}]", fps))

        sv = make_vect_envs("synthetic-atari", n, seed=0)
        fps = _measure(lambda: sv.step(acts), sv.reset, n, args.steps)
        rows.append((f"SyncVectorEnv[synthetic x{n}]", fps))
        sv.close()

        cv = make_vect_envs("CartPole-v1", n, seed=0)
        fps = _measure(lambda: cv.step(acts % 2), cv.reset, n, args.steps)
        rows.append((f"SyncVectorEnv[cartpole x{n}]", fps))
        cv.close()

        if n <= 16:  # process-per-env: keep the grid sane
            av = AsyncVectorEnv([lambda i=i: CartPoleEnv(seed=i)
                                 for i in range(n)])
            fps = _measure(lambda: av.step((acts % 2).tolist()), av.reset, n,
                           args.steps)
            rows.append((f"AsyncVectorEnv[cartpole x{n}]", fps))
            av.close()

    width = max(len(r[0]) for r in rows)
    for name, fps in rows:
        print(f"{name:<{width}}  {fps:>12,.0f} env-steps/s")


if __name__ == "__main__":
    main()
