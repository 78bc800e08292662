from .base import BaseTrainer
from .off_policy import OffPolicyTrainer

__all__ = ["BaseTrainer", "OffPolicyTrainer"]
