from .a3c import A3CTrainer, SharedAdam
from .a3c_gpu import A3CGpuTrainer
from .apex import ApexTrainer, ParallelDQNTrainer
from .base import BaseAgent
from .dqn import DQNAgent
from .impala import ImpalaTrainer
from .ppo import DDPPOTrainer

__all__ = ["ImpalaTrainer", "DQNAgent", "A3CTrainer", "SharedAdam",
           "A3CGpuTrainer", "ApexTrainer", "ParallelDQNTrainer", "DDPPOTrainer", "BaseAgent"]
