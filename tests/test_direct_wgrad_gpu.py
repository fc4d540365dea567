"""Direct-wgrad path: conv weight grads land in the
flat-bucket views straight from the wgrad kernel — no AccumulateGrad
`grad += dw` elementwise kernels — with readiness marked through the
registry exactly like the autograd hook would."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _grads(direct: bool):
    import distributeddeeplearning_amd.ops.functional as DF
    from distributeddeeplearning_amd.models import build_model
    from distributeddeeplearning_amd.ops import cast_model_
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("single")
    torch.manual_seed(4)
    m = build_model("resnet18", num_classes=10).cuda()
    cast_model_(m, torch.bfloat16)
    opt = DistributedOptimizer(FusedSGD(m.parameters(), lr=0.1),
                               named_parameters=m.named_parameters())
    if not direct:
        DF.set_direct_wgrad(None)
    torch.manual_seed(9)
    x = torch.randn(8, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (8,), device="cuda")
    loss = DF.softmax_cross_entropy(m(x), y)
    loss.backward()
    torch.cuda.synchronize()
    out = {n: p.grad.detach().float().cpu().clone()
           for n, p in m.named_parameters()}
    out["_loss"] = float(loss.detach().float())
    DF.set_direct_wgrad(None)
    return out


def test_direct_wgrad_matches_autograd_accumulate():
    a = _grads(direct=True)
    b = _grads(direct=False)
    assert a["_loss"] == b["_loss"]
    for k in a:
        if k == "_loss":
            continue
        assert torch.equal(a[k], b[k]), k


def test_direct_wgrad_accumulation_window():
    """Pass 1 writes the view directly; pass 2 must go through autograd
    accumulation so the window sums both passes."""
    import distributeddeeplearning_amd.ops.functional as DF
    from distributeddeeplearning_amd.ops.modules import Conv2d
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    from distributeddeeplearning_amd.parallel import DistributedOptimizer, api
    api.init("single")
    torch.manual_seed(5)
    conv = Conv2d(64, 64, 3, padding=1).cuda()
    conv.weight.data = conv.weight.data.to(torch.bfloat16).contiguous(
        memory_format=torch.channels_last)
    opt = DistributedOptimizer(
        FusedSGD(conv.parameters(), lr=0.1),
        named_parameters=conv.named_parameters(),
        backward_passes_per_step=2)
    cl = lambda t: t.contiguous(memory_format=torch.channels_last)
    x1 = cl(torch.randn(4, 64, 14, 14, device="cuda", dtype=torch.bfloat16))
    x2 = cl(torch.randn(4, 64, 14, 14, device="cuda", dtype=torch.bfloat16))
    conv(x1).float().pow(2).mean().backward()
    g1 = conv.weight.grad.detach().float().clone()
    conv(x2).float().pow(2).mean().backward()
    g12 = conv.weight.grad.detach().float().clone()
    torch.cuda.synchronize()
    # reference: independent single-pass grads
    ref = Conv2d(64, 64, 3, padding=1).cuda()
    ref.weight.data = conv.weight.data.clone()
    DF.set_direct_wgrad(None)
    ref.weight.requires_grad_(True)
    y = ref(x2).float().pow(2).mean()
    y.backward()
    g2 = ref.weight.grad.detach().float()
    assert torch.allclose(g12 - g1, g2, atol=3e-3, rtol=3e-2), \
        float((g12 - g1 - g2).abs().max())
    DF.set_direct_wgrad(None)
