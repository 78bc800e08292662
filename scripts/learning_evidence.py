#!/usr/bin/env python3
"""GPU-scale learning evidence: run the flagship
IMPALA config (bench.py defaults, bf16) on the synthetic-Atari env —
reward = 1 iff action == state % A, so random policy averages 1/A ≈ 0.167
per step and a converged policy approaches 1.0 — and log the batch mean
reward over learn iterations.  Rising reward under the bf16 hot path is
the learning check at bench scale (the CPU fp32 learning tests cover
numerics at small scale)."""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    os.environ.setdefault("MIOPEN_FIND_MODE", "1")
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=120)
    p.add_argument("--rollout-length", type=int, default=80)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--envs-per-actor", type=int, default=128)
    p.add_argument("--num-actors", type=int, default=24)
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--learning-rate", type=float, default=3e-4)
    p.add_argument("--entropy-cost", type=float, default=0.01)
    p.add_argument("--discounting", type=float, default=0.5,
                   help="gamma; 0.5 by default — the synthetic env's "
                        "reward is contextual-bandit-shaped, and a CPU "
                        "fp32 A/B (profiles/README.md) showed gamma=0.99 "
                        "drowns the advantage signal for ANY dtype while "
                        "0.5 learns within 400 updates")
    args = p.parse_args()

    import torch
    from scalerl_amd.config import ImpalaArguments
    from scalerl_amd.runtime.impala import ImpalaTrainer

    cfg = ImpalaArguments(
        rollout_length=args.rollout_length, batch_size=args.batch_size,
        envs_per_actor=args.envs_per_actor, num_actors=args.num_actors,
        use_lstm=True, dtype=args.dtype, inference="gpu",
        learning_rate=args.learning_rate, entropy_cost=args.entropy_cost,
        discounting=args.discounting,
        seed=314, total_steps=1 << 60, disable_checkpoint=True,
        output_dir="/tmp/scalerl_learn")
    t = ImpalaTrainer(cfg)
    t.start_actors()
    t.setup_learner()
    rew_trace = []
    try:
        t0 = time.time()
        for i in range(args.iters):
            batch = t.next_batch()
            r = float(batch["reward"][1:].float().mean())
            t.learn_step(batch)
            rew_trace.append(round(r, 4))
            if i % 10 == 0:
                print(f"iter {i:4d} batch-mean reward {r:.4f} "
                      f"({time.time() - t0:.0f}s)", flush=True)
        first = sum(rew_trace[:10]) / 10
        last = sum(rew_trace[-10:]) / 10
        print(json.dumps({
            "evidence": "impala-bf16-learning",
            "reward_first10": round(first, 4),
            "reward_last10": round(last, 4),
            "random_policy": round(1.0 / t.num_actions, 4),
            "improved": last > first + 0.05,
            "iters": args.iters, "dtype": args.dtype, "gamma": args.discounting,
            "trace_every10": rew_trace[::10]}), flush=True)
    finally:
        t.shutdown()


if __name__ == "__main__":
    main()
