from .base import BaseLogger, make_logger
from .jsonl import JsonlLogger

__all__ = ["BaseLogger", "JsonlLogger", "make_logger"]
