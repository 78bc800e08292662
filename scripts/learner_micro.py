#!/usr/bin/env python3
"""Learner-step microbenchmark: no actor processes, synthetic device-resident
batch.  Isolates the learner pipeline (model fwd → fused V-trace/loss →
backward → clip → fused RMSProp → publish) for rocprof kernel profiling and
launch-overhead analysis.  THIS IS NOT THE HEADLINE BENCH (bench.py runs the
full actor-learner pipeline); it measures the learner's ceiling.
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.models import AtariNet
from scalerl_amd.ops import FusedRMSprop, clip_grad_norm_, impala_loss
from scalerl_amd.parallel import FlatParams


def main():
    os.environ.setdefault("MIOPEN_FIND_MODE", "1")
    p = argparse.ArgumentParser()
    p.add_argument("--rollout-length", type=int, default=80)
    p.add_argument("--batch-size", type=int, default=32)
    p.add_argument("--num-actions", type=int, default=6)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--use-lstm", type=int, default=1)
    p.add_argument("--dtype", type=str, default="bf16")
    args = p.parse_args()

    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(0)
    torch.backends.cudnn.benchmark = True
    T, B, A = args.rollout_length, args.batch_size, args.num_actions
    model = AtariNet((4, 84, 84), A, use_lstm=bool(args.use_lstm)).to(dev)
    flat = FlatParams(model, device=dev)
    opt = FusedRMSprop(flat.flat, lr=6e-4, alpha=0.99, eps=0.01)
    pub = torch.zeros_like(flat.flat, device="cpu").pin_memory() \
        if dev.type == "cuda" else torch.zeros_like(flat.flat)

    batch = {
        "obs": torch.randint(0, 256, (T + 1, B, 4, 84, 84), dtype=torch.uint8,
                             device=dev),
        "reward": torch.randn(T + 1, B, device=dev),
        "done": torch.rand(T + 1, B, device=dev) < 0.01,
        "last_action": torch.randint(0, A, (T + 1, B), device=dev),
        "action": torch.randint(0, A, (T + 1, B), device=dev),
        "logits": torch.randn(T + 1, B, A, device=dev),
    }
    state = model.initial_state(B, device=dev) if args.use_lstm else ()
    autocast = (args.dtype == "bf16" and dev.type == "cuda")

    def step():
        flat.flat_grad.zero_()
        inputs = {k: batch[k] for k in ("obs", "reward", "done", "last_action")}
        if autocast:
            with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
                out, _ = model(inputs, state, greedy=True)
        else:
            out, _ = model(inputs, state, greedy=True)
        logits = out["policy_logits"].float()
        baseline = out["baseline"].float()
        total, comps, _ = impala_loss(
            batch["logits"][:-1], logits[:-1], batch["action"][:-1],
            torch.clamp(batch["reward"][1:], -1, 1),
            (~batch["done"][1:]).float() * 0.99,
            baseline[:-1], baseline[-1].detach())
        total.backward()
        clip_grad_norm_(flat.flat_grad, 40.0)
        opt.step(flat.flat_grad)
        pub.copy_(flat.flat, non_blocking=True)
        return total

    for _ in range(args.warmup):
        step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    ms = 1000 * dt / args.steps
    sps = T * B * args.steps / dt
    print(f"learner-micro: {ms:.2f} ms/iter  {sps:,.0f} env-steps/s "
          f"(T={T} B={B} lstm={bool(args.use_lstm)} dtype={args.dtype})")


if __name__ == "__main__":
    main()
