"""Dynamic loss scaling for the fp16 mixed-precision config (K13,
BASELINE.json:9).  Scale the loss up before backward; the fused optimizer
folds the 1/scale unscale into its update pass; on overflow the step is
skipped and the scale halved (grow back 2× after ``growth_interval`` good
steps) — the standard AMP recipe, implemented against the bucketed flat
gradients so the finite-check is a handful of kernel launches."""
from __future__ import annotations

import torch


class DynamicLossScaler:
    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 enabled: bool = True):
        self.scale = init_scale if enabled else 1.0
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.enabled = enabled
        self._good_steps = 0
        self.skipped_steps = 0

    def scale_loss(self, loss: torch.Tensor) -> torch.Tensor:
        return loss * self.scale if self.enabled else loss

    def check_and_update(self, flat_grads) -> bool:
        """After allreduce, verify finiteness of the (still-scaled) grads.

        Returns True if the step should proceed.  Does NOT unscale — the
        fused optimizer applies 1/scale during its update (one fewer pass
        over the gradients).
        """
        if not self.enabled:
            return True
        finite = _all_finite(flat_grads)
        if finite:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= self.growth_factor
                self._good_steps = 0
            return True
        self.scale = max(1.0, self.scale * self.backoff_factor)
        self._good_steps = 0
        self.skipped_steps += 1
        return False

    def state_dict(self):
        return {"scale": self.scale, "good_steps": self._good_steps,
                "skipped": self.skipped_steps}

    def load_state_dict(self, sd):
        self.scale = sd["scale"]
        self._good_steps = sd["good_steps"]
        self.skipped_steps = sd["skipped"]


def _all_finite(flats) -> bool:
    if not flats:
        return True
    if flats[0].is_cuda:
        from ..ops._ext import load_kernels_ext
        ext = load_kernels_ext()
        found_inf = torch.zeros(1, dtype=torch.int32, device=flats[0].device)
        for f in flats:
            ext.check_finite(f, found_inf)
        return int(found_inf.item()) == 0
    return all(bool(torch.isfinite(f).all()) for f in flats)
