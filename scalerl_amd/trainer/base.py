"""Generic trainer scaffolding (parity with ``scalerl/trainer/base.py:26-179``:
run-dir layout ``{work_dir}/{project}/{env}/{algo}/{tb_log, text_log,
video_dir, model_dir}``, main-process gating, metrics-logger selection)."""

from __future__ import annotations

import os
import time
from typing import Dict

from ..parallel.dist import get_rank
from ..utils import get_logger
from ..utils.loggers import make_logger


class BaseTrainer:
    def __init__(self, args, train_env=None, test_env=None, agent=None):
        self.args = args
        self.train_env = train_env
        self.test_env = test_env
        self.agent = agent
        self.rank = get_rank()
        self.is_main_process = self.rank == 0

        stamp = time.strftime("%Y%m%d_%H%M%S")
        self.work_dir = os.path.join(
            args.work_dir, args.project, args.env_id, args.algo_name, stamp)
        self.model_dir = os.path.join(self.work_dir, "model_dir")
        self.log_dir = os.path.join(self.work_dir, "text_log")
        self.metrics_dir = os.path.join(self.work_dir, "tb_log")
        self.video_dir = os.path.join(self.work_dir, "video_dir")
        if self.is_main_process:
            for d in (self.model_dir, self.log_dir, self.metrics_dir,
                      self.video_dir):
                os.makedirs(d, exist_ok=True)
            self.text_logger = get_logger(
                args.algo_name,
                log_file=os.path.join(self.log_dir, "train.log"))
            self.vis_logger = make_logger(args.logger_backend,
                                          self.metrics_dir)
        else:
            self.text_logger = get_logger(args.algo_name)
            self.vis_logger = None

    def log_train(self, data: Dict, step: int) -> None:
        if self.vis_logger is not None:
            self.vis_logger.log_train_data(data, step)

    def log_test(self, data: Dict, step: int) -> None:
        if self.vis_logger is not None:
            self.vis_logger.log_test_data(data, step)

    def run(self) -> None:
        raise NotImplementedError

    def close(self) -> None:
        if self.vis_logger is not None:
            self.vis_logger.close()
