from .base import Box, Discrete, Env, Space
from .cartpole import CartPoleEnv
from .synthetic import SyntheticAtariEnv, SyntheticPointGoalEnv
from .registry import make_env, make_gym_env
from .torch_env import TorchEnvWrapper
from .vec_env import SyncVectorEnv, make_vect_envs
from .async_vec_env import AsyncVectorEnv
from .a3c_env import AtariRescale42x42, NormalizedEnv, create_atari_env
from .multi_agent import (MultiAgentEnv, MultiAgentVecEnv,
                          SyntheticMultiAgentEnv)

__all__ = [
    "Env", "Space", "Box", "Discrete", "CartPoleEnv", "SyntheticAtariEnv",
    "SyntheticPointGoalEnv", "make_env", "make_gym_env", "TorchEnvWrapper", "SyncVectorEnv",
    "make_vect_envs", "AsyncVectorEnv", "AtariRescale42x42", "NormalizedEnv",
    "create_atari_env", "MultiAgentEnv", "MultiAgentVecEnv",
    "SyntheticMultiAgentEnv",
]
