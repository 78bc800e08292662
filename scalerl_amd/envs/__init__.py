from .base import Box, Discrete, Env, Space
from .cartpole import CartPoleEnv
from .synthetic import SyntheticAtariEnv, SyntheticPointGoalEnv
from .registry import make_env
from .torch_env import TorchEnvWrapper
from .vec_env import SyncVectorEnv, make_vect_envs

__all__ = [
    "Env", "Space", "Box", "Discrete", "CartPoleEnv", "SyntheticAtariEnv",
    "SyntheticPointGoalEnv", "make_env", "TorchEnvWrapper", "SyncVectorEnv",
    "make_vect_envs",
]
