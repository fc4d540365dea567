"""Metrics/logging: rank-0 structured JSONL + human-readable stdout lines
(the reference's ``common/timer.py`` images/sec prints + Horovod
``MetricAverageCallback`` — SURVEY.md §5.5)."""
from __future__ import annotations

import json
import logging
import sys
import time
from typing import Optional

import torch

from ..parallel import api

log = logging.getLogger("ddl")


def setup_logging(rank: int) -> None:
    if log.handlers:
        return
    h = logging.StreamHandler(sys.stdout)
    h.setFormatter(logging.Formatter(
        f"%(asctime)s rank{rank} %(levelname)s %(message)s", "%H:%M:%S"))
    log.addHandler(h)
    log.setLevel(logging.INFO if rank == 0 else logging.WARNING)


class MetricLogger:
    def __init__(self, jsonl_path: Optional[str] = None, rank: int = 0):
        self.rank = rank
        self.f = open(jsonl_path, "a") if (jsonl_path and rank == 0) else None

    def write(self, **fields) -> None:
        if self.rank != 0:
            return
        if self.f:
            self.f.write(json.dumps(fields) + "\n")
            self.f.flush()

    def close(self):
        if self.f:
            self.f.close()


def average_metric(value: float, device=None) -> float:
    """hvd.allreduce of a validation metric (C4 call site)."""
    if api.size() <= 1:
        return value
    t = torch.tensor([value], dtype=torch.float64,
                     device=device if device is not None else "cpu")
    api.allreduce_(t, average=True)
    return float(t.item())


class StepTimer:
    """images/sec over a logging window; device-synchronising on GPU."""

    def __init__(self, device: torch.device):
        self.device = device
        self.reset()

    def reset(self):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        self.t0 = time.perf_counter()
        self.images = 0

    def add(self, n_images: int):
        self.images += n_images

    def rate(self) -> float:
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        dt = time.perf_counter() - self.t0
        return self.images / dt if dt > 0 else 0.0
