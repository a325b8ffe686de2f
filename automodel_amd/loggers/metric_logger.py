"""JSONL metric logging + rank-0 console logging.

Reference behavior: nemo_automodel/components/loggers/metric_logger.py:92
(per train/val JSONL streams next to checkpoints) and loggers/log_utils.py
(rank-0 log filter).
"""

from __future__ import annotations

import json
import logging
import os
import sys
from typing import Any

import torch


def get_rank() -> int:
    if torch.distributed.is_available() and torch.distributed.is_initialized():
        return torch.distributed.get_rank()
    return int(os.environ.get("RANK", 0))


class RankFilter(logging.Filter):
    def filter(self, record: logging.LogRecord) -> bool:
        return get_rank() == 0


def setup_logging(level: int = logging.INFO) -> logging.Logger:
    logger = logging.getLogger("automodel_amd")
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stdout)
        handler.setFormatter(logging.Formatter("[%(asctime)s %(levelname)s] %(message)s"))
        handler.addFilter(RankFilter())
        logger.addHandler(handler)
    logger.setLevel(level)
    return logger


class MetricLogger:
    """Appends one JSON object per step to a .jsonl file (rank 0 only)."""

    def __init__(self, path: str | os.PathLike | None):
        self.path = str(path) if path else None
        self._fh = None
        if self.path and get_rank() == 0:
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            self._fh = open(self.path, "a")

    def log(self, metrics: dict[str, Any]) -> None:
        if self._fh is None:
            return
        clean = {}
        for k, v in metrics.items():
            if isinstance(v, torch.Tensor):
                v = v.item()
            clean[k] = v
        self._fh.write(json.dumps(clean) + "\n")
        self._fh.flush()

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None
