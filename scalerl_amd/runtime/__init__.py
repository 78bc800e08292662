from .a3c import A3CTrainer, SharedAdam
from .apex import ApexTrainer, ParallelDQNTrainer
from .base import BaseAgent
from .dqn import DQNAgent
from .impala import ImpalaTrainer
from .ppo import DDPPOTrainer

__all__ = ["ImpalaTrainer", "DQNAgent", "A3CTrainer", "SharedAdam",
           "ApexTrainer", "ParallelDQNTrainer", "DDPPOTrainer", "BaseAgent"]
