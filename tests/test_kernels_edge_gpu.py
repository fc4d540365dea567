"""Edge-shape GPU kernel tests: small batch, 7x7 spatial, deep channels,
non-multiple-of-tile sizes, fp16 dtype, VGG-style pooling."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu
DEV = "cuda"


def cl(t):
    return t.contiguous(memory_format=torch.channels_last)


def mk(shape, dtype=torch.bfloat16, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g).to(dtype)


def relerr(a, b):
    a, b = a.detach().float().cpu(), b.detach().float().cpu()
    return float((a - b).abs().mean() / b.abs().mean().clamp_min(1e-6))


@pytest.mark.parametrize("n,c,h,w,k,r,s,p,dt", [
    (1, 2048, 7, 7, 512, 1, 1, 0, torch.bfloat16),   # tiny M, deep C
    (3, 64, 13, 17, 72, 3, 1, 1, torch.bfloat16),    # odd spatial + odd K mult
    (2, 512, 7, 7, 512, 3, 1, 1, torch.float16),     # fp16, 7x7
    (5, 96, 9, 9, 128, 3, 2, 1, torch.bfloat16),     # odd batch, stride 2
])
def test_conv_fwd_edge(n, c, h, w, k, r, s, p, dt):
    x = mk((n, c, h, w), dt, 1)
    wt = mk((k, c, r, r), dt, 2) * 0.05
    ref = F.conv2d(x.float(), wt.float(), None, s, p)
    from distributeddeeplearning_amd.ops import functional as DF
    y = DF.conv2d(cl(x.to(DEV)), cl(wt.to(DEV)), None, s, p)
    assert y.shape == ref.shape
    assert relerr(y, ref) < 0.02


@pytest.mark.parametrize("n,c,h,w,k,r,s,p", [
    (1, 2048, 7, 7, 512, 1, 1, 0),
    (3, 64, 13, 17, 72, 3, 1, 1),
])
def test_conv_bwd_edge(n, c, h, w, k, r, s, p):
    x = mk((n, c, h, w), seed=3).float().requires_grad_(True)
    wt = (mk((k, c, r, r), seed=4) * 0.05).float().requires_grad_(True)
    yr = F.conv2d(x, wt, None, s, p)
    dy = mk(yr.shape, seed=5).float()
    yr.backward(dy)
    from distributeddeeplearning_amd.ops._ext import load_kernels_ext
    ext = load_kernels_ext()
    dyg = cl(dy.to(DEV).bfloat16())
    dx = ext.conv_dgrad(dyg, cl(wt.detach().to(DEV).bfloat16()),
                        list(x.shape), s, s, p, p)
    dw = ext.conv_wgrad(dyg, cl(x.detach().to(DEV).bfloat16()),
                        list(wt.shape), s, s, p, p)
    assert relerr(dx, x.grad) < 0.02
    assert relerr(dw, wt.grad) < 0.02


def test_vgg_pool_2x2():
    x = mk((2, 128, 56, 56), seed=6).float().requires_grad_(True)
    ref = F.max_pool2d(x, 2, 2)
    dy = mk(ref.shape, seed=7).float()
    ref.backward(dy)
    from distributeddeeplearning_amd.ops import functional as DF
    xg = cl(x.detach().to(DEV).bfloat16()).requires_grad_(True)
    y = DF.max_pool2d(xg, 2, 2)
    assert relerr(y, ref) < 0.01
    y.backward(cl(dy.to(DEV).bfloat16()))
    assert relerr(xg.grad, x.grad) < 0.01


def test_bn_small_spatial_deep_channels():
    x = mk((2, 2048, 7, 7), seed=8)
    wgt = (torch.rand(2048) + 0.5)
    b = torch.randn(2048) * 0.1
    rm, rv = torch.zeros(2048), torch.ones(2048)
    ref = F.relu(F.batch_norm(x.float(), rm.clone(), rv.clone(), wgt, b,
                              True, 0.1, 1e-5))
    from distributeddeeplearning_amd.ops._ext import load_kernels_ext
    ext = load_kernels_ext()
    y, mean, rstd, _mask = ext.bn_relu_fwd(cl(x.to(DEV)), wgt.bfloat16().to(DEV),
                                    b.bfloat16().to(DEV), rm.to(DEV),
                                    rv.to(DEV), 0.1, 1e-5, True, None)
    assert relerr(y, ref) < 0.02


def test_resnet101_smoke():
    from distributeddeeplearning_amd.config import TrainConfig
    from distributeddeeplearning_amd.engine.trainer import Trainer
    cfg = TrainConfig(model="resnet101", image_size=64, num_classes=100,
                      batch_size=8, dtype="bf16", warmup_epochs=0,
                      comm="single")
    tr = Trainer(cfg, steps_per_epoch=10)
    it = tr.data_iter()
    x, y = next(it)
    loss = tr.train_step(x, y)
    torch.cuda.synchronize()
    assert torch.isfinite(loss.float())


def test_inference_matches_train_path_stats():
    """eval-mode BN uses running stats via the single-pass infer kernel."""
    from distributeddeeplearning_amd.ops.modules import BatchNorm2d
    torch.manual_seed(0)
    bn = BatchNorm2d(64).to(DEV)
    from distributeddeeplearning_amd.ops.modules import cast_model_
    cast_model_(bn, torch.bfloat16)
    x = mk((8, 64, 14, 14), seed=9).to(DEV)
    bn.train()
    for _ in range(10):
        bn(cl(x + torch.randn_like(x.float()).bfloat16()))
    bn.eval()
    with torch.no_grad():
        y = bn(cl(x), relu=True)
    ref = F.relu(F.batch_norm(
        x.float().cpu(), bn.running_mean.cpu(), bn.running_var.cpu(),
        bn.weight.float().cpu(), bn.bias.float().cpu(), False, 0.1, 1e-5))
    assert relerr(y, ref) < 0.02


@pytest.mark.parametrize("n,c,h,w,k,r,s,p", [
    (2, 64, 28, 28, 64, 1, 1, 0),
    (2, 64, 28, 28, 128, 3, 2, 1),
    (2, 3, 64, 64, 64, 7, 2, 3),   # fp32 stem via im2col
])
def test_conv_fp32_exact_mfma(n, c, h, w, k, r, s, p):
    """fp32 conv on v_mfma_f32_16x16x4_f32: exact f32 fmaf-chain numerics
    (tight tolerance vs torch fp32 — only summation order differs)."""
    torch.manual_seed(11)
    x = torch.randn(n, c, h, w).requires_grad_(True)
    wt = (torch.randn(k, c, r, r) * 0.05).requires_grad_(True)
    yr = F.conv2d(x, wt, None, s, p)
    dy = torch.randn_like(yr)
    yr.backward(dy)
    from distributeddeeplearning_amd.ops import functional as DF
    xg = cl(x.detach().to(DEV))
    if c % 4 == 0:
        xg.requires_grad_(True)  # stem (C=3) has no input grad by design
    wg = cl(wt.detach().to(DEV)).requires_grad_(True)
    y = DF.conv2d(xg, wg, None, s, p)
    assert relerr(y, yr) < 2e-5, "fp32 fwd"
    y.backward(cl(dy.to(DEV)))
    if c % 4 == 0:
        assert relerr(xg.grad, x.grad) < 2e-5, "fp32 dgrad"
    assert relerr(wg.grad, wt.grad) < 5e-5, "fp32 wgrad"


def test_resnet18_fp32_whole_model_grad_parity():
    """Every parameter gradient of a full fp32 ResNet-18 on the GPU HIP path
    matches the CPU fp32 reference to ~1e-5 relative (exact-f32 MFMA +
    deterministic reductions)."""
    from distributeddeeplearning_amd import models
    from distributeddeeplearning_amd.ops import cast_model_
    from distributeddeeplearning_amd.ops import functional as DF
    torch.manual_seed(3)
    m_cpu = models.resnet18(num_classes=100)
    m_gpu = models.resnet18(num_classes=100)
    m_gpu.load_state_dict(m_cpu.state_dict())
    m_gpu = m_gpu.to(DEV)
    cast_model_(m_gpu, torch.float32)
    x = torch.randn(8, 3, 64, 64)
    yt = torch.randint(0, 100, (8,))
    l_cpu = DF.softmax_cross_entropy(m_cpu(x), yt)
    l_cpu.backward()
    l_gpu = DF.softmax_cross_entropy(m_gpu(cl(x.to(DEV))), yt.to(DEV))
    l_gpu.backward()
    assert abs(float(l_cpu) - float(l_gpu.detach())) < 1e-5
    for (n1, p1), (_, p2) in zip(m_cpu.named_parameters(),
                                 m_gpu.named_parameters()):
        assert relerr(p2.grad, p1.grad) < 1e-4, n1


def test_training_step_is_deterministic():
    """Every kernel in the step is deterministic by construction (two-stage
    reductions, no atomics in reduce paths): identical state + batch must
    give bitwise-identical gradients."""
    from distributeddeeplearning_amd import models
    from distributeddeeplearning_amd.ops import cast_model_
    from distributeddeeplearning_amd.ops import functional as DF
    torch.manual_seed(5)
    m = models.resnet18(num_classes=50).to(DEV)
    cast_model_(m, torch.bfloat16)
    x = cl(torch.randn(8, 3, 64, 64).bfloat16().to(DEV))
    yt = torch.randint(0, 50, (8,), device=DEV)

    def grads():
        for p in m.parameters():
            p.grad = None
        loss = DF.softmax_cross_entropy(m(x), yt)
        loss.backward()
        return [p.grad.detach().clone() for p in m.parameters()]

    g1, g2 = grads(), grads()
    for a, b in zip(g1, g2):
        assert torch.equal(a, b)


def test_fused_sgd_two_param_groups():
    from distributeddeeplearning_amd.ops.optim import FusedSGD
    torch.manual_seed(7)
    a = torch.nn.Parameter(torch.randn(512, 64, device=DEV))
    b = torch.nn.Parameter(torch.randn(256, 32, device=DEV))
    a_c = torch.nn.Parameter(a.detach().cpu().clone())
    b_c = torch.nn.Parameter(b.detach().cpu().clone())
    opt = FusedSGD([{"params": [a], "lr": 0.1},
                    {"params": [b], "lr": 0.01}], lr=0.1, momentum=0.9)
    opt_c = FusedSGD([{"params": [a_c], "lr": 0.1},
                      {"params": [b_c], "lr": 0.01}], lr=0.1, momentum=0.9)
    for _ in range(3):
        ga = torch.randn_like(a_c)
        gb = torch.randn_like(b_c)
        a.grad, b.grad = ga.to(DEV), gb.to(DEV)
        a_c.grad, b_c.grad = ga, gb
        opt.step()
        opt_c.step()
    assert relerr(a, a_c) < 1e-6 and relerr(b, b_c) < 1e-6


def test_stem_dgrad_smallc():
    """K2 completeness: dgrad through the C=3 stem (im2col GEMM + col2im
    scatter-add) vs the plain-PyTorch reference."""
    import torch.nn.functional as F
    from distributeddeeplearning_amd.ops import functional as DF
    torch.manual_seed(77)
    x = torch.randn(4, 3, 34, 34).float().requires_grad_(True)
    w = (torch.randn(64, 3, 7, 7) * 0.1).float().requires_grad_(True)
    ref = F.conv2d(x, w, stride=2, padding=3)
    dy = torch.randn_like(ref)
    ref.backward(dy)

    xg = cl(x.detach().to(DEV).bfloat16()).requires_grad_(True)
    wg = cl(w.detach().to(DEV).bfloat16()).requires_grad_(True)
    y = DF.conv2d(xg, wg, stride=2, padding=3)
    y.backward(cl(dy.to(DEV).bfloat16()))
    assert relerr(y, ref) < 0.02
    assert relerr(wg.grad, w.grad) < 0.03
    assert relerr(xg.grad, x.grad) < 0.03
