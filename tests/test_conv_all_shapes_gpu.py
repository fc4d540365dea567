"""Exhaustive tier-1 conv coverage: every distinct conv shape in
ResNet-18/50/152 and VGG-16, fwd + dgrad + wgrad vs
fp32 PyTorch references.  Spatial dims are scaled down 4x (14->4 floor 7)
where possible to keep runtime bounded — the kernels' tiling/masking paths
depend on (C, K, R, stride, pad) and M%128, all preserved."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu
DEV = "cuda"

# (C, H, K, R, stride, pad) — H scaled down from the real net where large
RESNET_SHAPES = [
    # conv1 stem (flat/im2col path)
    (3, 112, 64, 7, 2, 3),
    # layer1 (56^2 -> 28 here)
    (64, 28, 64, 1, 1, 0), (64, 28, 64, 3, 1, 1), (64, 28, 256, 1, 1, 0),
    (256, 28, 64, 1, 1, 0),
    # layer2
    (256, 28, 128, 1, 1, 0), (128, 28, 128, 3, 2, 1), (128, 14, 128, 3, 1, 1),
    (128, 14, 512, 1, 1, 0), (512, 14, 128, 1, 1, 0), (256, 28, 512, 1, 2, 0),
    # layer3
    (512, 14, 256, 1, 1, 0), (256, 14, 256, 3, 2, 1), (256, 7, 256, 3, 1, 1),
    (256, 7, 1024, 1, 1, 0), (1024, 7, 256, 1, 1, 0), (512, 14, 1024, 1, 2, 0),
    # layer4
    (1024, 7, 512, 1, 1, 0), (512, 7, 512, 3, 2, 1), (512, 4, 512, 3, 1, 1),
    (512, 4, 2048, 1, 1, 0), (2048, 4, 512, 1, 1, 0), (1024, 7, 2048, 1, 2, 0),
    # resnet18 basicblock extras
    (64, 28, 128, 3, 2, 1), (128, 14, 256, 3, 2, 1), (256, 7, 512, 3, 2, 1),
    # VGG-16 3x3 chain (+bias+relu covered elsewhere)
    (3, 56, 64, 3, 1, 1), (64, 56, 64, 3, 1, 1), (64, 28, 128, 3, 1, 1),
    (128, 28, 128, 3, 1, 1), (256, 14, 256, 3, 1, 1), (512, 7, 512, 3, 1, 1),
]


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def relerr(a, b):
    a, b = a.detach().float().cpu(), b.detach().float().cpu()
    return float((a - b).abs().mean() / b.abs().mean().clamp_min(1e-6))


@pytest.mark.parametrize("c,h,k,r,s,p", RESNET_SHAPES)
def test_conv_shape_fwd_bwd(c, h, k, r, s, p):
    torch.manual_seed(c * 1000 + k + r)
    n = 2
    x = (torch.randn(n, c, h, h).bfloat16()).float().requires_grad_(True)
    wt = (torch.randn(k, c, r, r).bfloat16().float() * 0.05).requires_grad_(True)
    yr = F.conv2d(x, wt, None, s, p)
    dy = torch.randn_like(yr).bfloat16().float()
    yr.backward(dy)

    from distributeddeeplearning_amd.ops._ext import load_kernels_ext
    ext = load_kernels_ext()
    xg = cl(x.detach().to(DEV).bfloat16())
    wg = cl(wt.detach().to(DEV).bfloat16())
    dyg = cl(dy.to(DEV).bfloat16())

    y = ext.conv_fwd(xg, wg, None, s, s, p, p, False)
    assert y.shape == yr.shape
    assert relerr(y, yr) < 0.02, f"fwd {c},{h},{k},{r},{s},{p}"

    if c % 8 == 0:
        dx = ext.conv_dgrad(dyg, wg, list(x.shape), s, s, p, p)
        assert relerr(dx, x.grad) < 0.03, f"dgrad {c},{h},{k},{r},{s},{p}"
        dw = ext.conv_wgrad(dyg, xg, list(wt.shape), s, s, p, p)
    else:
        _, col, _p = ext.conv_fwd_smallc(xg, wg, None, s, s, p, p, False)
        dw = ext.conv_wgrad_smallc(dyg, col, list(wt.shape), list(x.shape))
    assert relerr(dw, wt.grad) < 0.03, f"wgrad {c},{h},{k},{r},{s},{p}"
