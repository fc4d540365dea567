"""FusedSGD CPU math vs torch.optim.SGD; LARS sanity; master-weight flow."""
import torch
import pytest

from distributeddeeplearning_amd.ops.optim import FusedSGD


def _models():
    torch.manual_seed(0)
    m1 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                             torch.nn.Linear(16, 4))
    m2 = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                             torch.nn.Linear(16, 4))
    m2.load_state_dict(m1.state_dict())
    return m1, m2


@pytest.mark.parametrize("momentum,wd,nesterov", [
    (0.9, 1e-4, False), (0.9, 0.0, True), (0.0, 1e-4, False),
])
def test_matches_torch_sgd(momentum, wd, nesterov):
    m1, m2 = _models()
    o1 = torch.optim.SGD(m1.parameters(), lr=0.1, momentum=momentum,
                         weight_decay=wd, nesterov=nesterov)
    o2 = FusedSGD(m2.parameters(), lr=0.1, momentum=momentum,
                  weight_decay=wd, nesterov=nesterov)
    for step in range(5):
        x = torch.randn(4, 8)
        for m, o in ((m1, o1), (m2, o2)):
            o.zero_grad()
            m(x).square().mean().backward()
            o.step()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_lars_changes_update():
    m1, m2 = _models()
    o_plain = FusedSGD(m1.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    o_lars = FusedSGD(m2.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4,
                      lars=True)
    x = torch.randn(4, 8)
    for m, o in ((m1, o_plain), (m2, o_lars)):
        o.zero_grad()
        m(x).square().mean().backward()
        o.step()
    w1 = m1[0].weight
    w2 = m2[0].weight
    assert not torch.allclose(w1, w2)  # trust ratio altered the step
    # biases (dim==1) are excluded from LARS → identical updates
    assert torch.allclose(m1[0].bias, m2[0].bias, atol=1e-7)


def test_bf16_master_weights():
    torch.manual_seed(1)
    lin = torch.nn.Linear(16, 16)
    lin.to(torch.bfloat16)
    opt = FusedSGD(lin.parameters(), lr=1e-3, momentum=0.9)
    for _ in range(3):
        opt.zero_grad()
        x = torch.randn(4, 16, dtype=torch.bfloat16)
        lin(x).float().square().mean().backward()
        opt.step()
    st = opt.state[lin.weight]
    assert st["master"].dtype == torch.float32
    # master tracks higher precision than the bf16 param
    assert torch.allclose(st["master"].to(torch.bfloat16), lin.weight.detach())
