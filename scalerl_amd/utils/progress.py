"""Terminal progress bar + parallel progress tracking (parity with
``scalerl/utils/progress_bar.py:16-247``)."""

from __future__ import annotations

import sys
import time
from multiprocessing import Pool
from typing import Callable, Iterable, List


class ProgressBar:
    def __init__(self, total: int, width: int = 40, stream=None):
        self.total = total
        self.width = width
        self.count = 0
        self.start = time.time()
        self.stream = stream or sys.stderr

    def update(self, n: int = 1) -> None:
        self.count += n
        frac = min(self.count / max(self.total, 1), 1.0)
        filled = int(self.width * frac)
        elapsed = time.time() - self.start
        rate = self.count / elapsed if elapsed > 0 else 0.0
        eta = (self.total - self.count) / rate if rate > 0 else 0.0
        bar = "#" * filled + "-" * (self.width - filled)
        self.stream.write(f"\r[{bar}] {self.count}/{self.total} "
                          f"{rate:.1f}it/s eta {eta:.0f}s")
        if self.count >= self.total:
            self.stream.write("\n")
        self.stream.flush()


def track_progress(fn: Callable, items: Iterable, **bar_kw) -> List:
    items = list(items)
    bar = ProgressBar(len(items), **bar_kw)
    out = []
    for it in items:
        out.append(fn(it))
        bar.update()
    return out


def track_parallel_progress(fn: Callable, items: Iterable, nproc: int = 4,
                            **bar_kw) -> List:
    items = list(items)
    bar = ProgressBar(len(items), **bar_kw)
    out: List = [None] * len(items)
    with Pool(nproc) as pool:
        for i, res in enumerate(pool.imap(fn, items)):
            out[i] = res
            bar.update()
    return out
