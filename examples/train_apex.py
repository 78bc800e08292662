#!/usr/bin/env python3
"""Ape-X training entry point (benchmark config 4 shape)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.config import ApexArguments, parse_cli
from scalerl_amd.runtime.apex import ApexTrainer


def main():
    args = parse_cli(ApexArguments)
    world = int(os.environ.get("WORLD_SIZE", "1"))
    trainer = ApexTrainer(args)
    trainer.start_actors()
    if world > 1:
        from scalerl_amd.parallel.dist import init_distributed
        init_distributed()
    trainer.setup_learner()
    t0 = time.time()
    try:
        while trainer.global_step < args.max_train_steps:
            stats = trainer.train_iteration()
            if trainer.learn_iters % 100 == 0 and "loss" in stats:
                sps = trainer.global_step / (time.time() - t0)
                print(f"steps {trainer.global_step} SPS {sps:,.0f} "
                      f"loss {float(stats['loss']):.4f} "
                      f"buffer {len(trainer.buffer)}", flush=True)
    finally:
        if args.save_model:
            trainer.save(os.path.join(args.work_dir, "apex_checkpoint.pth"))
        trainer.shutdown()


if __name__ == "__main__":
    main()
