#!/usr/bin/env python3
"""Ape-X throughput probe (benchmark config 4 shape, single rank).

Reports learner SGD steps/s and actor-ingest env-steps/s for the
HBM-resident PER pipeline.  Not the driver headline bench (that is
bench.py / IMPALA); used for round-to-round comparisons."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ.setdefault("MIOPEN_FIND_MODE", "1")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=30)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--num-actors", type=int, default=8)
    p.add_argument("--envs-per-actor", type=int, default=32)
    p.add_argument("--batch-size", type=int, default=512)
    p.add_argument("--buffer-size", type=int, default=200_000)
    p.add_argument("--updates-per-iter", type=int, default=4)
    args = p.parse_args()

    import torch
    from scalerl_amd.config import ApexArguments
    from scalerl_amd.runtime.apex import ApexTrainer

    cfg = ApexArguments(
        num_actors=args.num_actors, envs_per_actor=args.envs_per_actor,
        batch_size=args.batch_size, buffer_size=args.buffer_size,
        warmup_learn_steps=args.batch_size * 2,
        learner_update_times=args.updates_per_iter,
        device="auto", seed=7)
    t = ApexTrainer(cfg)
    t.start_actors()
    t.setup_learner()
    for _ in range(args.warmup):
        t.train_iteration()
    if t.device.type == "cuda":
        torch.cuda.synchronize()
    step0, iters0 = t.global_step, t.learn_iters
    t0 = time.perf_counter()
    for _ in range(args.iters):
        t.train_iteration()
    if t.device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"apex: {(t.learn_iters - iters0) / dt:.1f} sgd-steps/s  "
          f"{(t.global_step - step0) / dt:,.0f} ingested env-steps/s  "
          f"buffer {len(t.buffer)}  ({dt / args.iters * 1000:.1f} ms/iter)")
    t.shutdown()


if __name__ == "__main__":
    main()
