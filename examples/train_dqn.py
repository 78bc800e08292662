#!/usr/bin/env python3
"""DQN training entry point (benchmark config 1 when run with defaults:
CartPole-v1, CPU).  Usage: python examples/train_dqn.py --env-id CartPole-v1
--max-train-steps 50000"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from scalerl_amd.config import DQNArguments, parse_cli
from scalerl_amd.envs.registry import make_env
from scalerl_amd.runtime.dqn import DQNAgent
from scalerl_amd.trainer import OffPolicyTrainer


def main():
    args = parse_cli(DQNArguments)
    device = ("cuda" if torch.cuda.is_available() else "cpu") \
        if args.device == "auto" else args.device
    probe = make_env(args.env_id)
    obs_dim = int(probe.observation_space.shape[0])
    action_dim = probe.action_space.n
    probe.close()
    agent = DQNAgent(args, obs_dim, action_dim, device=device)
    trainer = OffPolicyTrainer(args, agent, device=device)
    trainer.run()
    ev = trainer.run_evaluate_episodes(args.eval_episodes)
    trainer.text_logger.info(f"final eval: {ev}")
    trainer.close()


if __name__ == "__main__":
    main()
