"""File+stdout logger and a JSONL metrics sink.

The reference uses a closure-based file logger (``utils/log.py:4-17``) plus
wandb (initialized disabled, main.py:53). Here: the same ``create_logger``
surface, plus ``MetricsLogger`` — an offline JSONL metrics stream (wandb is
not installed in this environment) that rank-gates itself in distributed
runs.
"""

import json
import os
import time
from typing import Optional


def create_logger(log_filename: str, display: bool = True):
    f = open(log_filename, 'a')
    counter = [0]

    def logger(text: str):
        if display:
            print(text)
        f.write(text + '\n')
        counter[0] += 1
        if counter[0] % 10 == 0:
            f.flush()
            os.fsync(f.fileno())

    return logger, f.close


class MetricsLogger:
    """Append-only JSONL metrics (one object per log call, with step/time)."""

    def __init__(self, path: Optional[str] = None, rank: int = 0):
        self.rank = rank
        self.path = path
        self._f = None
        if path is not None and rank == 0:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            self._f = open(path, 'a')
        self.step = 0

    def log(self, metrics: dict, step: Optional[int] = None):
        if self._f is None:
            return
        rec = {'_step': self.step if step is None else step,
               '_time': time.time()}
        for k, v in metrics.items():
            try:
                rec[k] = float(v)
            except (TypeError, ValueError):
                rec[k] = str(v)
        self._f.write(json.dumps(rec) + '\n')
        self._f.flush()
        if step is None:
            self.step += 1

    def close(self):
        if self._f is not None:
            self._f.close()
            self._f = None
