#!/usr/bin/env python3
"""A3C at GPU scale (benchmark config 2: Pong-42x42 family, 16 CPU actors
+ 1 MI355X learner) — batched synchronous A2C on the actor-learner runtime.

    python examples/train_a3c_gpu.py --env-id synthetic-atari \
        --num-actors 16 --total-steps 10000000
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.config import A3CGpuArguments, parse_cli
from scalerl_amd.runtime.a3c_gpu import A3CGpuTrainer


def main():
    args = parse_cli(A3CGpuArguments)
    trainer = A3CGpuTrainer(args)
    trainer.train()


if __name__ == "__main__":
    main()
