"""Bit-exact training determinism.

Every reduction in the framework is fixed-order by design (two-stage BN
stats, split-K wgrad slabs with a deterministic combine, one fused
optimizer launch, no atomics on the training path), so two identically
seeded runs must produce BIT-IDENTICAL parameters — a property the
reference stack (cuDNN autotuned algorithms + atomics) cannot offer.
"""
import dataclasses

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run(steps=8, dtype="bf16", model="resnet50", lars=False):
    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    cfg = TrainConfig(model=model, image_size=64, num_classes=100,
                      batch_size=32, dtype=dtype, warmup_epochs=0,
                      base_lr=0.05, comm="single", seed=123, lars=lars)
    tr = Trainer(cfg, steps_per_epoch=10 ** 9)
    it = tr.data_iter()
    for _ in range(steps):
        x, y = next(it)
        loss = tr.train_step(x, y)
    torch.cuda.synchronize()
    params = torch.cat([p.detach().reshape(-1).float()
                        for p in tr.model.parameters()]).cpu()
    stats = torch.cat([b.float().reshape(-1).cpu()
                       for m in tr.model.modules()
                       for n, b in getattr(m, "_buffers", {}).items()
                       if torch.is_tensor(b) and b.is_floating_point()])
    return params, stats, float(loss.detach().float())


@pytest.mark.parametrize("dtype,lars", [("bf16", False), ("fp32", False),
                                        ("bf16", True)])
def test_training_is_bit_deterministic(dtype, lars):
    p1, s1, l1 = _run(dtype=dtype, lars=lars)
    p2, s2, l2 = _run(dtype=dtype, lars=lars)
    assert l1 == l2
    assert torch.equal(p1, p2), "parameters diverged between identical runs"
    assert torch.equal(s1, s2), "BN running stats diverged"
