"""Tier-1 GPU kernel numerics: every HIP kernel vs a plain fp32 PyTorch
reference of the same op (inputs pre-rounded to the kernel dtype so only
accumulation error is measured).  SURVEY.md §4.3 Tier 1."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda"


def _ext():
    from distributeddeeplearning_amd.ops._ext import load_kernels_ext
    return load_kernels_ext()


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def mk(shape, dtype=torch.bfloat16, seed=0):
    g = torch.Generator().manual_seed(seed)
    t = torch.randn(*shape, generator=g)
    return t.to(dtype)  # round first so CPU ref sees identical inputs


def relerr(a, b):
    a = a.float().cpu()
    b = b.float().cpu()
    denom = b.abs().mean().clamp_min(1e-6)
    return float((a - b).abs().mean() / denom)


# ---- conv -----------------------------------------------------------------

CONV_SHAPES = [
    # (N, C, H, W, K, R, stride, pad)  — ResNet-50/VGG-representative
    (4, 64, 56, 56, 64, 1, 1, 0),      # 1x1
    (4, 64, 56, 56, 64, 3, 1, 1),      # 3x3
    (4, 256, 56, 56, 512, 1, 2, 0),    # 1x1 stride-2 downsample
    (4, 128, 28, 28, 128, 3, 2, 1),    # 3x3 stride-2
    (2, 3, 64, 64, 64, 7, 2, 3),       # conv1 (small-C flat path)
    (2, 512, 14, 14, 512, 3, 1, 1),    # deep layer
]


@pytest.mark.parametrize("n,c,h,w,k,r,s,p", CONV_SHAPES)
def test_conv_fwd(n, c, h, w, k, r, s, p):
    x = mk((n, c, h, w), seed=1)
    wt = mk((k, c, r, r), seed=2) * 0.1
    ref = F.conv2d(x.float(), wt.float(), None, s, p)
    from distributeddeeplearning_amd.ops import functional as DF
    y = DF.conv2d(cl(x.to(DEV)), cl(wt.to(DEV)), None, s, p)
    assert y.shape == ref.shape
    e = relerr(y, ref)
    assert e < 0.01, f"conv fwd mean rel err {e}"


def test_conv_fwd_bias_relu():
    x = mk((2, 64, 28, 28), seed=3)
    wt = mk((128, 64, 3, 3), seed=4) * 0.1
    b = mk((128,), seed=5)
    ref = F.relu(F.conv2d(x.float(), wt.float(), b.float(), 1, 1))
    from distributeddeeplearning_amd.ops import functional as DF
    y = DF.conv2d(cl(x.to(DEV)), cl(wt.to(DEV)), b.to(DEV), 1, 1, relu=True)
    assert relerr(y, ref) < 0.01


@pytest.mark.parametrize("n,c,h,w,k,r,s,p", [
    (4, 64, 56, 56, 64, 1, 1, 0),
    (4, 64, 56, 56, 64, 3, 1, 1),
    (4, 128, 28, 28, 256, 3, 2, 1),
    (4, 256, 56, 56, 512, 1, 2, 0),
])
def test_conv_dgrad_wgrad(n, c, h, w, k, r, s, p):
    x = mk((n, c, h, w), seed=6).float().requires_grad_(True)
    wt = (mk((k, c, r, r), seed=7) * 0.1).float().requires_grad_(True)
    y = F.conv2d(x, wt, None, s, p)
    dy = mk(y.shape, seed=8).float()
    y.backward(dy)

    ext = _ext()
    dyg = cl(dy.to(DEV).bfloat16())
    xg = cl(x.detach().to(DEV).bfloat16())
    wg = cl(wt.detach().to(DEV).bfloat16())
    dx = ext.conv_dgrad(dyg, wg, list(x.shape), s, s, p, p)
    dw = ext.conv_wgrad(dyg, xg, list(wt.shape), s, s, p, p)
    assert relerr(dx, x.grad) < 0.015, "dgrad"
    assert relerr(dw, wt.grad) < 0.015, "wgrad"


def test_conv1_wgrad_flat():
    # first-layer wgrad uses the FLAT path (C=3)
    x = mk((2, 3, 32, 32), seed=9).float().requires_grad_(True)
    wt = (mk((64, 3, 7, 7), seed=10) * 0.1).float().requires_grad_(True)
    y = F.conv2d(x, wt, None, 2, 3)
    dy = mk(y.shape, seed=11).float()
    y.backward(dy)
    ext = _ext()
    dw = ext.conv_wgrad(cl(dy.to(DEV).bfloat16()), cl(x.detach().to(DEV).bfloat16()),
                        list(wt.shape), 2, 2, 3, 3)
    assert relerr(dw, wt.grad) < 0.015


# ---- batchnorm ------------------------------------------------------------

def test_bn_fwd_train():
    torch.manual_seed(0)
    x = mk((8, 64, 14, 14), seed=12)
    wgt = torch.rand(64) + 0.5
    b = torch.randn(64) * 0.1
    rm = torch.zeros(64)
    rv = torch.ones(64)
    ref = F.batch_norm(x.float(), rm.clone(), rv.clone(), wgt, b, True, 0.1, 1e-5)
    rm_ref, rv_ref = rm.clone(), rv.clone()
    F.batch_norm(x.float(), rm_ref, rv_ref, wgt, b, True, 0.1, 1e-5)

    ext = _ext()
    rm_g, rv_g = rm.clone().to(DEV), rv.clone().to(DEV)
    y, mean, rstd, _mask = ext.bn_relu_fwd(cl(x.to(DEV)), wgt.bfloat16().to(DEV),
                                    b.bfloat16().to(DEV), rm_g, rv_g,
                                    0.1, 1e-5, False, None)
    assert relerr(y, ref) < 0.01
    assert relerr(rm_g, rm_ref) < 0.01
    assert relerr(rv_g, rv_ref) < 0.01


def test_bn_relu_residual_fwd_bwd():
    x = mk((4, 128, 14, 14), seed=13).float().requires_grad_(True)
    res = mk((4, 128, 14, 14), seed=14).float().requires_grad_(True)
    wgt = (torch.rand(128) + 0.5).requires_grad_(True)
    b = (torch.randn(128) * 0.1).requires_grad_(True)
    rm, rv = torch.zeros(128), torch.ones(128)
    ref = F.relu(F.batch_norm(x, rm.clone(), rv.clone(), wgt, b, True, 0.1, 1e-5) + res)
    dy = mk(ref.shape, seed=15).float()
    ref.backward(dy)

    ext = _ext()
    xg = cl(x.detach().to(DEV).bfloat16())
    resg = cl(res.detach().to(DEV).bfloat16())
    wgt_g = wgt.detach().bfloat16().to(DEV)
    bg = b.detach().bfloat16().to(DEV)
    y, mean, rstd, _mask = ext.bn_relu_fwd(xg, wgt_g, bg, rm.to(DEV), rv.to(DEV),
                                    0.1, 1e-5, True, resg)
    assert relerr(y, ref) < 0.02
    dyg = cl(dy.to(DEV).bfloat16())
    dx, dw, db, dres = ext.bn_relu_bwd(dyg, xg, wgt_g, bg, mean, rstd, y,
                                       True, True)
    assert relerr(dres, res.grad) < 0.02
    assert relerr(dw, wgt.grad) < 0.02
    assert relerr(db, b.grad) < 0.02
    assert relerr(dx, x.grad) < 0.05  # bf16 stats amplify; mean-rel still small
    # mode 3: bit-packed forward mask instead of the y re-read.  The masks
    # are verified identical; the values differ only by -ffast-math FMA
    # contraction between the two template instances.
    assert _mask.numel() == xg.numel() // 8
    dx3, dw3, db3, dres3 = ext.bn_relu_bwd(dyg, xg, wgt_g, bg, mean, rstd, y,
                                           True, True, None, _mask)
    assert torch.equal(dres3, dres)  # dres = masked dy: no arithmetic
    assert relerr(dx3, dx) < 1e-2
    assert relerr(dw3, dw) < 1e-2 and relerr(db3, db) < 1e-2
    assert relerr(dx3, x.grad) < 0.05


# ---- pooling --------------------------------------------------------------

def test_maxpool_fwd_bwd():
    x = mk((4, 64, 32, 32), seed=16).float().requires_grad_(True)
    ref = F.max_pool2d(x, 3, 2, 1)
    dy = mk(ref.shape, seed=17).float()
    ref.backward(dy)
    from distributeddeeplearning_amd.ops import functional as DF
    xg = cl(x.detach().to(DEV).bfloat16()).requires_grad_(True)
    y = DF.max_pool2d(xg, 3, 2, 1)
    assert relerr(y, ref) < 0.01
    y.backward(cl(dy.to(DEV).bfloat16()))
    assert relerr(xg.grad, x.grad) < 0.01


def test_gap_fwd_bwd():
    x = mk((4, 2048, 7, 7), seed=18).float().requires_grad_(True)
    ref = x.mean(dim=(2, 3))
    dy = mk(ref.shape, seed=19).float()
    ref.backward(dy)
    from distributeddeeplearning_amd.ops import functional as DF
    xg = cl(x.detach().to(DEV).bfloat16()).requires_grad_(True)
    y = DF.global_avg_pool(xg)
    assert relerr(y, ref) < 0.01
    y.backward(dy.to(DEV).bfloat16())
    assert relerr(xg.grad, x.grad) < 0.01


# ---- softmax-xent ---------------------------------------------------------

def test_softmax_xent():
    logits = mk((64, 1000), seed=20).float().requires_grad_(True)
    tgt = torch.randint(0, 1000, (64,))
    ref = F.cross_entropy(logits, tgt)
    ref.backward()
    from distributeddeeplearning_amd.ops import functional as DF
    lg = logits.detach().to(DEV).bfloat16().requires_grad_(True)
    loss = DF.softmax_cross_entropy(lg, tgt.to(DEV))
    assert abs(float(loss) - float(ref)) < 0.02
    loss.backward()
    assert relerr(lg.grad, logits.grad) < 0.02


def test_softmax_xent_label_smoothing():
    logits = mk((32, 100), seed=21).float()
    tgt = torch.randint(0, 100, (32,))
    ref = F.cross_entropy(logits, tgt, label_smoothing=0.1)
    from distributeddeeplearning_amd.ops import functional as DF
    loss = DF.softmax_cross_entropy(logits.to(DEV).bfloat16(), tgt.to(DEV), 0.1)
    assert abs(float(loss) - float(ref)) < 0.03


# ---- optimizer ------------------------------------------------------------

def test_fused_sgd_matches_cpu():
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    torch.manual_seed(2)
    w_cpu = torch.nn.Parameter(torch.randn(1000, 33))
    w_gpu = torch.nn.Parameter(w_cpu.detach().bfloat16().to(DEV).clone())
    g = torch.randn(1000, 33)
    o_cpu = FusedSGD([w_cpu], lr=0.1, momentum=0.9, weight_decay=1e-4)
    o_gpu = FusedSGD([w_gpu], lr=0.1, momentum=0.9, weight_decay=1e-4)
    for step in range(3):
        w_cpu.grad = g.clone() * (step + 1)
        w_gpu.grad = g.bfloat16().to(DEV) * (step + 1)
        o_cpu.step()
        o_gpu.step()
    m_cpu = o_cpu.state[w_cpu]["master"]
    m_gpu = o_gpu.state[w_gpu]["master"].cpu()
    assert relerr(m_gpu, m_cpu) < 5e-3


def test_fused_sgd_lars_matches_cpu():
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    torch.manual_seed(3)
    w_cpu = torch.nn.Parameter(torch.randn(512, 64))
    w_gpu = torch.nn.Parameter(w_cpu.detach().bfloat16().to(DEV).clone())
    g = torch.randn(512, 64) * 0.1
    o_cpu = FusedSGD([w_cpu], lr=0.1, momentum=0.9, weight_decay=1e-4, lars=True)
    o_gpu = FusedSGD([w_gpu], lr=0.1, momentum=0.9, weight_decay=1e-4, lars=True)
    for _ in range(2):
        w_cpu.grad = g.clone()
        w_gpu.grad = g.bfloat16().to(DEV)
        o_cpu.step()
        o_gpu.step()
    assert relerr(o_gpu.state[w_gpu]["master"].cpu(), o_cpu.state[w_cpu]["master"]) < 5e-3


# ---- misc -----------------------------------------------------------------

def test_relu_bwd():
    ext = _ext()
    y = mk((1000,), seed=22).to(DEV)
    dy = mk((1000,), seed=23).to(DEV)
    out = ext.relu_bwd(dy, y)
    ref = torch.where(y.float().cpu() > 0, dy.float().cpu(), torch.zeros(1))
    assert relerr(out, ref) < 1e-4


def test_check_finite_and_unscale():
    ext = _ext()
    t = torch.randn(4096, device=DEV, dtype=torch.bfloat16)
    flag = torch.zeros(1, dtype=torch.int32, device=DEV)
    ext.check_finite(t, flag)
    assert int(flag.item()) == 0
    t[100] = float("inf")
    ext.check_finite(t, flag)
    assert int(flag.item()) == 1
    t2 = torch.full((512,), 4.0, device=DEV, dtype=torch.bfloat16)
    flag.zero_()
    ext.unscale_check(t2, 0.25, flag)
    assert int(flag.item()) == 0
    assert torch.allclose(t2.float(), torch.ones(512, device=DEV))


def test_dropout():
    from distributeddeeplearning_amd.ops import functional as DF
    x = torch.ones(100000, device=DEV, dtype=torch.bfloat16)
    y = DF.dropout(x, p=0.3, training=True, seed=42)
    kept = (y.float() > 0).float().mean().item()
    assert abs(kept - 0.7) < 0.02
    nz = y[y.float() > 0].float()
    assert torch.allclose(nz, torch.full_like(nz, 1.0 / 0.7), rtol=1e-2)


def test_bn_relu_no_residual_bwd_recomputed_mask():
    """bn1/bn2 style: relu without residual — backward recomputes the mask
    from w*xhat+b (never reads y)."""
    x = mk((4, 128, 14, 14), seed=31).float().requires_grad_(True)
    wgt = (torch.rand(128) - 0.2).requires_grad_(True)  # mixed-sign gammas
    b = (torch.randn(128) * 0.2).requires_grad_(True)
    rm, rv = torch.zeros(128), torch.ones(128)
    ref = F.relu(F.batch_norm(x, rm.clone(), rv.clone(), wgt, b, True, 0.1, 1e-5))
    dy = mk(ref.shape, seed=32).float()
    ref.backward(dy)
    ext = _ext()
    xg = cl(x.detach().to(DEV).bfloat16())
    wgt_g = wgt.detach().bfloat16().to(DEV)
    bg = b.detach().bfloat16().to(DEV)
    y, mean, rstd, _mask = ext.bn_relu_fwd(xg, wgt_g, bg, rm.to(DEV), rv.to(DEV),
                                    0.1, 1e-5, True, None)
    dx, dw, db, _ = ext.bn_relu_bwd(cl(dy.to(DEV).bfloat16()), xg, wgt_g, bg,
                                    mean, rstd, y, True, False)
    assert relerr(dw, wgt.grad) < 0.03
    assert relerr(db, b.grad) < 0.03
    assert relerr(dx, x.grad) < 0.06


def test_conv_bn_stats_fusion():
    """K1+K4 fusion: BN mean/rstd folded from the conv epilogue's partials
    match the standalone bn_stats pass over the conv output."""
    ext = _ext()
    x = cl(mk((8, 64, 28, 28), seed=41).to(DEV))
    w = cl(mk((128, 64, 3, 3), seed=42).to(DEV) * 0.1)
    y1, partials = ext.conv_fwd_stats(x, w, None, 1, 1, 1, 1, False)
    y0 = ext.conv_fwd(x, w, None, 1, 1, 1, 1, False)
    assert torch.equal(y1, y0), "fused-stats conv changed y"
    assert partials.numel() > 0
    C = 128
    M = y1.numel() // C
    assert partials.numel() % (2 * C) == 0
    wgt = torch.ones(C, dtype=torch.bfloat16, device=DEV)
    b = torch.zeros(C, dtype=torch.bfloat16, device=DEV)
    rm0, rv0 = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    rm1, rv1 = torch.zeros(C, device=DEV), torch.ones(C, device=DEV)
    yb0, mean0, rstd0, _m0 = ext.bn_relu_fwd(y0, wgt, b, rm0, rv0, 0.1, 1e-5,
                                        False, None)
    yb1, mean1, rstd1, _m1 = ext.bn_relu_fwd(y1, wgt, b, rm1, rv1, 0.1, 1e-5,
                                        False, None, partials)
    # partials are pre-bf16-rounding sums: equal up to rounding noise
    assert relerr(mean1, mean0) < 5e-3
    assert relerr(rstd1, rstd0) < 5e-3
    assert relerr(yb1, yb0) < 5e-3
    assert relerr(rv1, rv0) < 5e-3


def test_conv_bn_fusion_module_training_step():
    """Module-level wiring: resnet conv(emit_bn_stats) -> bn trains and
    matches the unfused configuration closely."""
    from distributeddeeplearning_amd.ops.modules import BatchNorm2d, Conv2d
    torch.manual_seed(7)
    conv_f = Conv2d(64, 128, 3, padding=1, emit_bn_stats=True).to(DEV)
    bn_f = BatchNorm2d(128).to(DEV)
    conv_u = Conv2d(64, 128, 3, padding=1, emit_bn_stats=False).to(DEV)
    bn_u = BatchNorm2d(128).to(DEV)
    conv_u.load_state_dict(conv_f.state_dict())
    bn_u.load_state_dict(bn_f.state_dict())
    from distributeddeeplearning_amd.ops.modules import cast_model_
    for m in (conv_f, bn_f, conv_u, bn_u):
        cast_model_(m, torch.bfloat16)
        m.train()
    x = cl(mk((8, 64, 14, 14), seed=43).to(DEV)).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    yf = bn_f(conv_f(x), relu=True)
    yu = bn_u(conv_u(x2), relu=True)
    assert relerr(yf, yu) < 5e-3
    gf = torch.autograd.grad(yf.float().sum(), [x, conv_f.weight])
    gu = torch.autograd.grad(yu.float().sum(), [x2, conv_u.weight])
    assert relerr(gf[0], gu[0]) < 2e-2
    assert relerr(gf[1], gu[1]) < 2e-2


def test_conv_dgrad_accum_fusion():
    """dgrad epilogue accumulate: conv_dgrad(dy, w, accum=a) equals
    conv_dgrad(dy, w) + a (the fused residual-join gradient add)."""
    ext = _ext()
    dy = cl(mk((4, 128, 28, 28), seed=51).to(DEV))
    w = cl(mk((128, 64, 3, 3), seed=52).to(DEV) * 0.1)
    a = cl(mk((4, 64, 28, 28), seed=53).to(DEV))
    xs = [4, 64, 28, 28]
    base = ext.conv_dgrad(dy, w, xs, 1, 1, 1, 1)
    fused = ext.conv_dgrad(dy, w, xs, 1, 1, 1, 1, a)
    ref = (base.float() + a.float()).bfloat16()
    assert relerr(fused, ref) < 2e-3
    # strided phase-decomposed path too
    dy2 = cl(mk((4, 128, 14, 14), seed=54).to(DEV))
    a2 = cl(mk((4, 64, 28, 28), seed=55).to(DEV))
    base2 = ext.conv_dgrad(dy2, w, xs, 2, 2, 1, 1)
    fused2 = ext.conv_dgrad(dy2, w, xs, 2, 2, 1, 1, a2)
    ref2 = (base2.float() + a2.float()).bfloat16()
    assert relerr(fused2, ref2) < 2e-3


def test_fork_conv_residual_fusion_grads():
    """fork_conv2d (residual-join add fused into dgrad) produces the same
    gradients as the plain fork + autograd add."""
    from distributeddeeplearning_amd.models.resnet import Bottleneck
    from distributeddeeplearning_amd.ops import functional as DF
    from distributeddeeplearning_amd.ops.modules import cast_model_
    torch.manual_seed(11)
    blk_f = Bottleneck(256, 64).to(DEV)
    blk_u = Bottleneck(256, 64).to(DEV)
    blk_u.load_state_dict(blk_f.state_dict())
    for b in (blk_f, blk_u):
        cast_model_(b, torch.bfloat16)
        b.train()
    x = cl(mk((4, 256, 14, 14), seed=61).to(DEV)).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    yf = blk_f(x)
    old = DF._FUSE_RESJOIN
    DF._FUSE_RESJOIN = False
    try:
        yu = blk_u(x2)
    finally:
        DF._FUSE_RESJOIN = old
    assert torch.equal(yf.float(), yu.float()), "forward changed by fusion"
    dy = mk(yf.shape, seed=62).to(DEV)
    yf.backward(cl(dy))
    yu.backward(cl(dy))
    # dx: fused add in f32 inside the epilogue vs bf16 autograd add —
    # equal up to one rounding step
    assert relerr(x.grad, x2.grad) < 2e-3
    for (n1, p1), (n2, p2) in zip(blk_f.named_parameters(),
                                  blk_u.named_parameters()):
        assert n1 == n2
        assert relerr(p1.grad, p2.grad) < 2e-3, n1
