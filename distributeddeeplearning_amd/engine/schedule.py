"""Large-batch LR schedule (Goyal et al. 2017), kept call-compatible with the
reference recipe (Horovod ``LearningRateWarmupCallback`` + the scripts'
piecewise decay — SURVEY.md §3.4 / Appendix B):

* linear scaling: peak_lr = base_lr × (global_batch / 256)
* linear warmup from base_lr to peak_lr over ``warmup_epochs``
* then ×``decay_factor`` at each epoch in ``decay_epochs``.
"""
from __future__ import annotations

from typing import Sequence


class WarmupPiecewiseLR:
    def __init__(
        self,
        optimizer,
        base_lr: float,
        global_batch: int,
        steps_per_epoch: int,
        warmup_epochs: int = 5,
        decay_epochs: Sequence[int] = (30, 60, 80),
        decay_factor: float = 0.1,
        reference_batch: int = 256,
    ):
        self.optimizer = optimizer
        self.base_lr = base_lr
        self.peak_lr = base_lr * max(1.0, global_batch / reference_batch)
        self.steps_per_epoch = max(1, steps_per_epoch)
        self.warmup_steps = warmup_epochs * self.steps_per_epoch
        self.decay_epochs = sorted(decay_epochs)
        self.decay_factor = decay_factor
        self.last_lr = self.get(0)

    def get(self, step: int) -> float:
        if self.warmup_steps > 0 and step < self.warmup_steps:
            frac = step / self.warmup_steps
            return self.base_lr + (self.peak_lr - self.base_lr) * frac
        epoch = step // self.steps_per_epoch
        lr = self.peak_lr
        for e in self.decay_epochs:
            if epoch >= e:
                lr *= self.decay_factor
        return lr

    def step(self, step: int) -> float:
        lr = self.get(step)
        for group in self.optimizer.param_groups:
            group["lr"] = lr
        self.last_lr = lr
        return lr

    def state_dict(self):
        return {"base_lr": self.base_lr, "peak_lr": self.peak_lr}

    def load_state_dict(self, sd):
        self.base_lr = sd["base_lr"]
        self.peak_lr = sd["peak_lr"]
