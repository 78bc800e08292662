#!/usr/bin/env python3
"""DD-PPO training entry point (benchmark config 5 shape).

Multi-GPU: python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
               --master-addr 127.0.0.1 examples/train_ddppo.py ...
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.config import DDPPOArguments, parse_cli
from scalerl_amd.runtime.ppo import DDPPOTrainer


def main():
    args = parse_cli(DDPPOArguments)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        from scalerl_amd.parallel.dist import init_distributed
        init_distributed()
    trainer = DDPPOTrainer(args)
    t0 = time.time()
    while trainer.global_step < args.max_train_steps:
        stats = trainer.train_iteration()
        if trainer.rank == 0:
            sps = trainer.global_step / (time.time() - t0)
            print(f"steps {trainer.global_step} SPS {sps:,.0f} "
                  f"loss {stats['loss']:.4f}", flush=True)
    if args.save_model and trainer.rank == 0:
        trainer.save(os.path.join(args.work_dir, "ddppo_checkpoint.pth"))


if __name__ == "__main__":
    main()
