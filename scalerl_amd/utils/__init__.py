from .logging import get_logger
from .schedulers import LinearDecayScheduler, MultiStepScheduler, PiecewiseScheduler
from .timings import Timings, Timer, check_time, timer_report
from .model_utils import hard_target_update, soft_target_update
from .checkpoint import load_checkpoint, save_checkpoint
from .progress import ProgressBar, track_parallel_progress, track_progress
from .algo_utils import chkpt_attribute_to_device, compile_model, remove_compile_prefix

__all__ = [
    "get_logger", "LinearDecayScheduler", "MultiStepScheduler",
    "PiecewiseScheduler", "Timings", "Timer", "check_time", "timer_report", "hard_target_update",
    "soft_target_update", "save_checkpoint", "load_checkpoint",
    "ProgressBar", "track_progress", "track_parallel_progress",
    "chkpt_attribute_to_device", "compile_model", "remove_compile_prefix",
]
