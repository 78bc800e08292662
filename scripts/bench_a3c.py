#!/usr/bin/env python3
"""Throughput probe for BASELINE config 2: A3C Pong-42×42 family,
16 CPU actors + 1 GPU learner (A3CGpuTrainer).  Synthetic frames stand in
for ALE (no ROMs in the image); the 42×42 preprocessing stack runs for
real in the actors.  Prints one JSON line (NOT the headline bench)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    os.environ.setdefault("MIOPEN_FIND_MODE", "1")
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--num-actors", type=int, default=16)
    p.add_argument("--envs-per-actor", type=int, default=16)
    p.add_argument("--rollout-steps", type=int, default=20)
    p.add_argument("--slots-per-batch", type=int, default=8)
    p.add_argument("--dtype", type=str, default="bf16")
    args = p.parse_args()

    import torch
    from scalerl_amd.config import A3CGpuArguments
    from scalerl_amd.runtime.a3c_gpu import A3CGpuTrainer

    cfg = A3CGpuArguments(
        num_actors=args.num_actors, envs_per_actor=args.envs_per_actor,
        rollout_steps=args.rollout_steps, slots_per_batch=args.slots_per_batch,
        dtype=args.dtype, seed=11, disable_checkpoint=True)
    t = A3CGpuTrainer(cfg)
    t.start_actors()
    t.setup_learner()
    try:
        for _ in range(args.warmup):
            t.train_iteration()
        if t.device.type == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            t.train_iteration()
        if t.device.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        spi = args.rollout_steps * args.envs_per_actor * args.slots_per_batch
        print(json.dumps({
            "bench": "a3c-gpu (BASELINE config 2)",
            "env_steps_per_sec": round(args.steps * spi / dt, 1),
            "ms_per_step": round(1000 * dt / args.steps, 2),
            "config": {"actors": args.num_actors, "E": args.envs_per_actor,
                       "T": args.rollout_steps, "K": args.slots_per_batch,
                       "batch": args.envs_per_actor * args.slots_per_batch,
                       "dtype": args.dtype,
                       "device": str(t.device)}}), flush=True)
    finally:
        t.shutdown()


if __name__ == "__main__":
    main()
