"""Logging + local metrics sink.

Replaces the reference's Python-logging setup (utils/utils.py:299-332) and
its AzureML ``run.log`` coupling (reference: e2e_trainer.py:21,72-74,
server.py:43-44) with a local JSONL metrics sink behind the same
``log_metric(key, value)`` callback shape.
"""

from __future__ import annotations

import json
import logging
import os
import time
from typing import Any, Dict, Optional

_LOGGER = logging.getLogger("msrflute_amd")
_METRICS_SINK: Optional["MetricsSink"] = None


def _rank() -> int:
    return int(os.environ.get("RANK", 0))


def init_logging(log_dir: Optional[str] = None, loglevel=logging.INFO):
    """Configure root logging to stdout (+ <log_dir>/log.out when given)."""
    handlers = [logging.StreamHandler()]
    if log_dir:
        os.makedirs(log_dir, exist_ok=True)
        handlers.append(logging.FileHandler(os.path.join(log_dir, "log.out")))
    logging.basicConfig(
        level=loglevel,
        format="%(asctime)s : %(levelname)s : %(message)s",
        handlers=handlers,
        force=True,
    )


def print_rank(msg: str, loglevel=logging.INFO):
    """Rank-prefixed log line (reference: utils/utils.py:318-322)."""
    _LOGGER.log(loglevel, "rank %d: %s", _rank(), msg)


def print_cuda_stats():
    import torch
    if torch.cuda.is_available():
        dev = torch.cuda.current_device()
        print_rank(
            "cuda mem: allocated=%.1fMB max_allocated=%.1fMB reserved=%.1fMB" % (
                torch.cuda.memory_allocated(dev) / 2**20,
                torch.cuda.max_memory_allocated(dev) / 2**20,
                torch.cuda.memory_reserved(dev) / 2**20,
            ))


class MetricsSink:
    """Append-only JSONL metrics log; one record per ``log_metric`` call."""

    def __init__(self, path: Optional[str]):
        self.path = path
        self._fh = None
        if path:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._fh = open(path, "a", buffering=1)

    def log(self, key: str, value: Any, step: Optional[int] = None):
        rec: Dict[str, Any] = {"t": time.time(), "key": key, "value": value}
        if step is not None:
            rec["step"] = step
        if self._fh:
            self._fh.write(json.dumps(rec) + "\n")

    def close(self):
        if self._fh:
            self._fh.close()
            self._fh = None


def init_metrics_sink(path: Optional[str]):
    global _METRICS_SINK
    _METRICS_SINK = MetricsSink(path)
    return _METRICS_SINK


def log_metric(key: str, value: Any, step: Optional[int] = None):
    """Process-local metric logging (AzureML ``run.log`` replacement)."""
    if _METRICS_SINK is not None:
        _METRICS_SINK.log(key, value, step)
    print_rank(f"metric {key} = {value}", loglevel=logging.DEBUG)
