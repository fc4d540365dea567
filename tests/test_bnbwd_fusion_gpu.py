"""GPU parity tests for the dgrad->BN-backward-stats fusion.

The conv dgrad epilogue emits the preceding BN's backward partials
(csrc/kernels/conv_igemm.hip BSTATS); these tests compare every gradient
of a bn(relu)->conv chain against (a) the unfused kernel path and (b) the
plain-PyTorch fp32 reference (SURVEY.md §4.3 Tier 1).
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

cl = lambda t: t.contiguous(memory_format=torch.channels_last)


def _chain(dtype, C=64, K=128, H=28, N=8, stride=1, ksize=1, seed=3):
    """bn(relu) -> conv via the module path; returns grads + the exact
    inputs/params so a reference can replay them."""
    from distributeddeeplearning_amd.ops.modules import BatchNorm2d, Conv2d
    torch.manual_seed(seed)
    bn = BatchNorm2d(C, fuse_bwd_stats=True).cuda().train()
    pad = ksize // 2
    conv = Conv2d(C, K, ksize, stride=stride, padding=pad).cuda().train()
    bn.weight.data = (torch.randn(C).abs().cuda() + 0.5).to(dtype)
    bn.bias.data = (torch.randn(C).cuda() * 0.1).to(dtype)
    conv.weight.data = cl(conv.weight.data.to(dtype))
    x = cl(torch.randn(N, C, H, H, device="cuda", dtype=dtype) * 0.5)
    x.requires_grad_(True)
    y = conv(bn(x, relu=True))
    loss = (y.float() ** 2).mean()
    loss.backward()
    return {
        "dx": x.grad.float().cpu(),
        "bn_dw": bn.weight.grad.float().cpu(),
        "bn_db": bn.bias.grad.float().cpu(),
        "conv_dw": conv.weight.grad.float().cpu(),
        "_x": x.detach().float().cpu(),
        "_bnw": bn.weight.detach().float().cpu(),
        "_bnb": bn.bias.detach().float().cpu(),
        "_cw": conv.weight.detach().float().cpu(),
        "_stride": stride, "_pad": pad,
    }


@pytest.mark.parametrize("stride,ksize", [(1, 1), (1, 3), (2, 3)])
def test_fused_matches_unfused(monkeypatch, stride, ksize):
    import distributeddeeplearning_amd.ops.modules as M
    monkeypatch.setattr(M, "_FUSE_BNBWD", True)
    fused = _chain(torch.bfloat16, stride=stride, ksize=ksize)
    monkeypatch.setattr(M, "_FUSE_BNBWD", False)
    ref = _chain(torch.bfloat16, stride=stride, ksize=ksize)
    for k in ("dx", "bn_dw", "bn_db", "conv_dw"):
        # same masked g values summed in fp32, different reduction order
        scale = ref[k].abs().max().clamp_min(1e-6)
        diff = (fused[k] - ref[k]).abs().max() / scale
        assert diff < 3e-2, (k, float(diff))


def test_fused_matches_fp32_reference(monkeypatch):
    import torch.nn.functional as F
    import distributeddeeplearning_amd.ops.modules as M
    monkeypatch.setattr(M, "_FUSE_BNBWD", True)
    got = _chain(torch.bfloat16, C=64, K=64, H=14, N=4, seed=9)

    x = got["_x"].requires_grad_(True)
    bw = got["_bnw"].requires_grad_(True)
    bb = got["_bnb"].requires_grad_(True)
    w = got["_cw"].requires_grad_(True)
    C = x.shape[1]
    y = F.conv2d(F.relu(F.batch_norm(
        x, torch.zeros(C), torch.ones(C), bw, bb, True, 0.1, 1e-5)),
        w, stride=got["_stride"], padding=got["_pad"])
    (y ** 2).mean().backward()
    for name, mine, refv in [("dx", got["dx"], x.grad),
                             ("bn_dw", got["bn_dw"], bw.grad),
                             ("bn_db", got["bn_db"], bb.grad),
                             ("conv_dw", got["conv_dw"], w.grad)]:
        scale = refv.abs().max().clamp_min(1e-6)
        diff = (mine - refv).abs().max() / scale
        assert diff < 5e-2, (name, float(diff))


def _bottleneck_grads(enabled: bool):
    """Full Bottleneck chain: exercises BOTH fusion forms — bn1/bn2
    (recomputed mask via the chain conv) and bn3 (saved bit-mask via the
    next block's fork-conv accum epilogue)."""
    import distributeddeeplearning_amd.ops.modules as M
    import distributeddeeplearning_amd.ops.functional as DF
    from distributeddeeplearning_amd.models.resnet import Bottleneck
    old = M._FUSE_BNBWD
    M._FUSE_BNBWD = enabled
    DF.set_direct_wgrad(None)
    try:
        torch.manual_seed(11)
        b1 = Bottleneck(64, 16).cuda().train()
        b2 = Bottleneck(64, 16).cuda().train()
        for blk in (b1, b2):
            for p in blk.parameters():
                p.data = (cl(p.data.to(torch.bfloat16)) if p.dim() == 4
                          else p.data.to(torch.bfloat16))
        x = cl(torch.randn(4, 64, 14, 14, device="cuda",
                           dtype=torch.bfloat16) * 0.5)
        x.requires_grad_(True)
        # two chained blocks: b1's bn3 output is consumed by b2's fork
        y = b2(b1(x))
        (y.float() ** 2).mean().backward()
        torch.cuda.synchronize()
        g = {n: p.grad.detach().float().cpu().clone()
             for blk, tag in ((b1, "b1"), (b2, "b2"))
             for n, p in blk.named_parameters()
             for n in (f"{tag}.{n}",)}
        g["dx"] = x.grad.detach().float().cpu().clone()
        return g
    finally:
        M._FUSE_BNBWD = old


def test_bottleneck_chain_fused_matches_unfused():
    a = _bottleneck_grads(True)
    b = _bottleneck_grads(False)
    assert a.keys() == b.keys()
    for k in a:
        scale = b[k].abs().max().clamp_min(1e-6)
        diff = (a[k] - b[k]).abs().max() / scale
        assert diff < 3e-2, (k, float(diff))
