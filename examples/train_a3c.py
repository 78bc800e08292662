#!/usr/bin/env python3
"""A3C (hogwild) training entry point (benchmark config 2 small scale)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from scalerl_amd.config import A3CArguments, parse_cli
from scalerl_amd.runtime.a3c import A3CTrainer


def main():
    args = parse_cli(A3CArguments)
    trainer = A3CTrainer(args)
    trainer.start()
    try:
        while trainer.global_step.value < args.max_train_steps:
            time.sleep(5)
            ret = trainer.evaluate(args.eval_episodes)
            print(f"step {trainer.global_step.value} eval return {ret:.1f}",
                  flush=True)
    finally:
        if args.save_model:
            trainer.save(os.path.join(args.work_dir, "a3c_checkpoint.pth"))
        trainer.shutdown()


if __name__ == "__main__":
    main()
