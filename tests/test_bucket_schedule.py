"""Bucket manager invariants + LR schedule values."""
import torch

from distributeddeeplearning_amd.engine.schedule import WarmupPiecewiseLR
from distributeddeeplearning_amd.parallel.bucket import BucketManager


def _model():
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(), torch.nn.Linear(128, 32))


def test_grad_views_alias_flat_buffers():
    m = _model()
    bm = BucketManager(m.named_parameters(), bucket_bytes=1 << 16)
    assert bm.grad_views_valid()
    x = torch.randn(4, 64)
    m(x).sum().backward()
    total = sum(b.flat.abs().sum() for b in bm.buckets)
    assert total > 0  # backward accumulated into the flat buffers
    bm.zero_grads()
    for p in m.parameters():
        assert p.grad.abs().sum() == 0  # views see the zeroing


def test_bucket_order_is_reverse_registration():
    m = _model()
    bm = BucketManager(m.named_parameters(), bucket_bytes=1 << 30)
    first_bucket = bm.buckets[0]
    # last registered param (bias of final linear) must be in bucket 0
    last_param = list(m.parameters())[-1]
    assert any(p is last_param for p in first_bucket.params)


def test_bucket_split_respects_size():
    m = torch.nn.Sequential(*[torch.nn.Linear(256, 256) for _ in range(8)])
    bm = BucketManager(m.named_parameters(), bucket_bytes=300 * 1024)
    assert len(bm.buckets) > 1
    for b in bm.buckets[:-1]:
        assert b.flat.numel() * 4 <= 2 * 300 * 1024


def test_mark_ready_fills_bucket():
    m = _model()
    bm = BucketManager(m.named_parameters(), bucket_bytes=1 << 30)
    b = bm.buckets[0]
    full = None
    for p in b.params:
        full = bm.mark_ready(p)
    assert full is b
    bm.reset()
    assert b.ready == 0


def test_warmup_piecewise_lr():
    class Opt:
        param_groups = [{"lr": 0.0}]

    spe = 100
    s = WarmupPiecewiseLR(Opt(), base_lr=0.1, global_batch=2048,
                          steps_per_epoch=spe, warmup_epochs=5,
                          decay_epochs=(30, 60, 80), decay_factor=0.1)
    assert abs(s.peak_lr - 0.8) < 1e-9  # 0.1 * 2048/256
    assert abs(s.get(0) - 0.1) < 1e-9
    mid = s.get(250)  # halfway through warmup
    assert 0.1 < mid < 0.8
    assert abs(s.get(5 * spe) - 0.8) < 1e-9
    assert abs(s.get(31 * spe) - 0.08) < 1e-9
    assert abs(s.get(61 * spe) - 0.008) < 1e-9
    assert abs(s.get(81 * spe) - 0.0008) < 1e-9


def test_set_to_none_guard():
    import pytest
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("single")
    m = _model()
    opt = DistributedOptimizer(FusedSGD(m.parameters(), lr=0.1),
                               named_parameters=m.named_parameters())
    m(torch.randn(2, 64)).sum().backward()
    opt.step()
    m.zero_grad(set_to_none=True)  # detaches the flat views
    with pytest.raises(RuntimeError, match="flat bucket view"):
        m(torch.randn(2, 64)).sum().backward()
    # recovery path: DistributedOptimizer.zero_grad rebinds
    opt.zero_grad()
    m(torch.randn(2, 64)).sum().backward()
    opt.step()


def test_grad_accumulation_sums_both_passes():
    """2-pass accumulation must apply g1+g2, and zero_grad between passes
    must not wipe the accumulated gradients."""
    import copy
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("single")
    torch.manual_seed(11)
    m1 = torch.nn.Linear(8, 8)
    m2 = copy.deepcopy(m1)

    o1 = DistributedOptimizer(FusedSGD(m1.parameters(), lr=0.1),
                              named_parameters=m1.named_parameters(),
                              backward_passes_per_step=2)
    x1, x2 = torch.randn(4, 8), torch.randn(4, 8)
    m1(x1).sum().backward()
    o1.step()        # accumulation pass: no update
    o1.zero_grad()   # must be a no-op mid-accumulation
    m1(x2).sum().backward()
    o1.step()        # real step on g(x1)+g(x2)
    o1.zero_grad()

    o2 = DistributedOptimizer(FusedSGD(m2.parameters(), lr=0.1),
                              named_parameters=m2.named_parameters())
    m2(x1).sum().backward()
    m2(x2).sum().backward()  # same accumulated grads in one pass
    o2.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-7)
