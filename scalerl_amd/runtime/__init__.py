from .impala import ImpalaTrainer

__all__ = ["ImpalaTrainer"]
