"""Experiment logging.

Reference observability surface (engine.py:72-98,124-125,149-158): a file
logger under experiments/<exp>/logs and TensorBoard scalars.  Here:

* python logging to experiments/<exp>/logs/<tag>.log (always)
* scalars to a JSONL file experiments/<exp>/scalars.jsonl (always;
  greppable, no extra deps)
* TensorBoard SummaryWriter when the tensorboard package is importable
  (it is optional in this environment)

Only rank 0 writes.
"""

from __future__ import annotations

import json
import logging
import os
import sys
import time
from typing import Optional


def setup_logger(root: str, exp_path: str, tag: str, rank: int = 0) -> logging.Logger:
    logger = logging.getLogger(f"pvraft.{tag}.r{rank}")
    logger.setLevel(logging.INFO)
    logger.propagate = False
    if logger.handlers:
        return logger
    fmt = logging.Formatter("%(asctime)s %(levelname)s %(message)s")
    if rank == 0:
        log_dir = os.path.join(root, "experiments", exp_path, "logs")
        os.makedirs(log_dir, exist_ok=True)
        fh = logging.FileHandler(os.path.join(log_dir, f"{tag}.log"))
        fh.setFormatter(fmt)
        logger.addHandler(fh)
        sh = logging.StreamHandler(sys.stdout)
        sh.setFormatter(fmt)
        logger.addHandler(sh)
    else:
        logger.addHandler(logging.NullHandler())
    return logger


class ScalarLogger:
    """JSONL scalar sink + optional TensorBoard."""

    def __init__(self, root: str, exp_path: str, rank: int = 0):
        self.rank = rank
        self._jsonl = None
        self._tb = None
        if rank != 0:
            return
        exp_dir = os.path.join(root, "experiments", exp_path)
        os.makedirs(exp_dir, exist_ok=True)
        self._jsonl = open(os.path.join(exp_dir, "scalars.jsonl"), "a")
        try:
            from torch.utils.tensorboard import SummaryWriter

            self._tb = SummaryWriter(log_dir=os.path.join(exp_dir, "tb"), flush_secs=10)
        except Exception:
            self._tb = None

    def add_scalar(self, key: str, value: float, step: int) -> None:
        if self.rank != 0:
            return
        self._jsonl.write(
            json.dumps({"t": time.time(), "key": key, "value": float(value), "step": int(step)}) + "\n"
        )
        self._jsonl.flush()
        if self._tb is not None:
            self._tb.add_scalar(key, value, step)

    def close(self) -> None:
        if self._jsonl is not None:
            self._jsonl.close()
        if self._tb is not None:
            self._tb.close()
