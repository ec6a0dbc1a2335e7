"""Metric logging: TensorBoard events + datestamped text logs.

Replaces the reference's Redis→Logger-thread→SummaryWriter pump (logger.py
per variant, e.g. MT10_Distributed_CARE/src/logger.py:9-132) with direct
in-process writes: players/learners hand scalars to a MetricLogger which
writes tfevents (via the dependency-free writer) and an append-only text
log (reference Learner.my_print, learner.py:173-176).
"""

from __future__ import annotations

import datetime
import os
import time
from typing import Dict, Optional

from .tfevents import TFEventWriter


class MetricLogger:
    def __init__(self, logdir: Optional[str], text_log: bool = True,
                 stdout: bool = False):
        self.logdir = logdir
        self.writer = TFEventWriter(logdir) if logdir else None
        self.stdout = stdout
        self._text = None
        if logdir and text_log:
            os.makedirs(logdir, exist_ok=True)
            stamp = datetime.datetime.now().strftime("%Y-%m-%d")
            self._text = os.path.join(logdir, f"dsac_log_{stamp}.txt")

    def add_scalar(self, tag: str, value: float, step: int) -> None:
        if self.writer:
            self.writer.add_scalar(tag, value, step)

    def add_scalars(self, prefix: str, values: Dict[str, float], step: int) -> None:
        for k, v in values.items():
            self.add_scalar(f"{prefix}/{k}", v, step)

    def write_hyperparameters(self, cfg: Dict) -> None:
        """Reference MT loggers dump the cfg as TB text at startup
        (MT10_Distributed_CARE/src/logger.py:34-43); we log flattened
        numerics as scalars + the full cfg to the text log."""
        self.print(f"cfg: {cfg}")

    def print(self, content: str) -> None:
        if self.stdout:
            print(content, flush=True)
        if self._text:
            with open(self._text, "a") as f:
                f.write(content + "\n")

    def flush(self) -> None:
        if self.writer:
            self.writer.flush()

    def close(self) -> None:
        if self.writer:
            self.writer.close()
