#!/usr/bin/env python3
"""IMPALA training entry point (benchmark config 3 shape).

Single GPU:  python examples/train_impala.py --total-steps 10000000
Multi-GPU:   python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
                 --master-addr 127.0.0.1 examples/train_impala.py ...
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.config import ImpalaArguments, parse_cli
from scalerl_amd.runtime.impala import ImpalaTrainer


def main():
    args = parse_cli(ImpalaArguments)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    trainer = ImpalaTrainer(args)
    trainer.start_actors()
    if world > 1:
        from scalerl_amd.parallel.dist import init_distributed
        init_distributed()
    trainer.train()


if __name__ == "__main__":
    main()
