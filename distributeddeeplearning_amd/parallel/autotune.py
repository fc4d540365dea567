"""Online gradient-bucket autotuning (the Horovod
``common/parameter_manager.cc`` role — HOROVOD_AUTOTUNE — adapted to the
persistent-flat-bucket design).

Horovod retunes its fusion threshold by timing training cycles online;
here buckets are persistent flat buffers fixed at construction, so the
tuner REBUCKETS between optimizer steps instead: during early training it
walks a candidate-size ladder, times a window of whole steps per
candidate on every rank, allreduces the totals so every rank picks the
SAME winner, and rebuilds the buckets once at that size.  Deterministic
schedule — every rank rebuckets at identical step indices, so the static
collective order is preserved throughout.

Enable with ``DDL_AUTOTUNE=1`` (or ``TrainConfig.autotune_buckets``);
candidates via ``DDL_AUTOTUNE_MB`` (comma list, default "10,25,50,100").
"""
from __future__ import annotations

import os
import time
from typing import Optional, Sequence

import torch

import logging

log = logging.getLogger("ddl")


class OnlineBucketTuner:
    """Call :meth:`step` once per completed optimizer step."""

    def __init__(self, opt, sizes_mb: Optional[Sequence[float]] = None,
                 steps_per_candidate: int = 8, warmup_steps: int = 3):
        if sizes_mb is None:
            sizes_mb = [float(x) for x in os.environ.get(
                "DDL_AUTOTUNE_MB", "10,25,50,100").split(",")]
        self.opt = opt
        self.sizes = list(sizes_mb)
        self.steps_per = int(steps_per_candidate)
        self.warmup = int(warmup_steps)
        self.times: dict[float, float] = {}
        self._idx = -1          # current candidate index (-1 = not started)
        self._step_in_phase = 0
        self._t0 = 0.0
        self.done = False
        self.chosen_mb: Optional[float] = None

    def _enter(self, idx: int) -> None:
        self._idx = idx
        self._step_in_phase = 0
        self.opt.rebucket(int(self.sizes[idx] * (1 << 20)))

    def step(self) -> None:
        """Advance the tuning state machine (after an optimizer step)."""
        if self.done:
            return
        if self._idx < 0:
            self._enter(0)
            return
        self._step_in_phase += 1
        if self._step_in_phase == self.warmup:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            self._t0 = time.monotonic()
        elif self._step_in_phase == self.steps_per:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            self.times[self.sizes[self._idx]] = time.monotonic() - self._t0
            if self._idx + 1 < len(self.sizes):
                self._enter(self._idx + 1)
            else:
                self._finish()

    def _finish(self) -> None:
        # every rank must pick the SAME winner: reduce the per-candidate
        # totals across ranks (host-side, control-plane collective)
        from . import api
        vals = torch.tensor([self.times[s] for s in self.sizes],
                            dtype=torch.float64)
        if api.size() > 1:
            api.allreduce_(vals, average=False)
        best = int(torch.argmin(vals))
        self.chosen_mb = self.sizes[best]
        self.opt.rebucket(int(self.chosen_mb * (1 << 20)))
        self.done = True
        log.info("bucket autotune: %s -> chose %.0f MB (totals %s)",
                 self.sizes, self.chosen_mb,
                 [round(float(v), 3) for v in vals])
