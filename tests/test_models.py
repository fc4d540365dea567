"""Model construction, shapes, parameter counts and torchvision-compatible
state-dict naming (checkpoint layout contract)."""
import pytest
import torch

from distributeddeeplearning_amd import models


@pytest.mark.parametrize("name,params_m", [
    ("resnet18", 11.69), ("resnet50", 25.56), ("resnet152", 60.19),
    ("vgg16", 138.36),
])
def test_param_counts(name, params_m):
    m = models.build_model(name)
    n = sum(p.numel() for p in m.parameters()) / 1e6
    assert abs(n - params_m) < 0.05, f"{name}: {n:.2f}M params"


@pytest.mark.parametrize("name", ["resnet18", "resnet50"])
def test_forward_shape(name):
    m = models.build_model(name, num_classes=10).eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        y = m(x)
    assert y.shape == (2, 10)


def test_resnet50_state_dict_names():
    sd = models.resnet50().state_dict()
    expected = [
        "conv1.weight", "bn1.weight", "bn1.bias", "bn1.running_mean",
        "bn1.running_var", "bn1.num_batches_tracked",
        "layer1.0.conv1.weight", "layer1.0.bn1.weight",
        "layer1.0.downsample.0.weight", "layer1.0.downsample.1.running_var",
        "layer4.2.bn3.bias", "fc.weight", "fc.bias",
    ]
    for k in expected:
        assert k in sd, f"missing {k}"
    # shapes of a few torchvision-known entries
    assert sd["conv1.weight"].shape == (64, 3, 7, 7)
    assert sd["layer1.0.conv2.weight"].shape == (64, 64, 3, 3)
    assert sd["layer4.2.conv3.weight"].shape == (2048, 512, 1, 1)
    assert sd["fc.weight"].shape == (1000, 2048)


def test_vgg16_state_dict_names():
    sd = models.vgg16().state_dict()
    for k in ["features.0.weight", "features.0.bias", "features.2.weight",
              "features.5.weight", "features.28.weight",
              "classifier.0.weight", "classifier.3.weight",
              "classifier.6.weight"]:
        assert k in sd, f"missing {k}"
    assert sd["features.0.weight"].shape == (64, 3, 3, 3)
    assert sd["classifier.0.weight"].shape == (4096, 25088)


def test_resnet_backward_cpu():
    m = models.resnet18(num_classes=10)
    x = torch.randn(2, 3, 32, 32)
    y = torch.randint(0, 10, (2,))
    from distributeddeeplearning_amd.ops import functional as DF
    loss = DF.softmax_cross_entropy(m(x), y)
    loss.backward()
    assert all(p.grad is not None for p in m.parameters())
    assert torch.isfinite(loss)


def test_resnet152_structure():
    m = models.resnet152()
    assert len(m.layer3) == 36 and len(m.layer2) == 8


def test_vgg16_backward_cpu():
    import torch
    from distributeddeeplearning_amd import models
    from distributeddeeplearning_amd.ops import functional as DF
    torch.manual_seed(0)
    m = models.vgg16(num_classes=10)
    m.train()
    x = torch.randn(1, 3, 224, 224)
    y = torch.randint(0, 10, (1,))
    loss = DF.softmax_cross_entropy(m(x), y)
    loss.backward()
    assert torch.isfinite(loss)
    assert all(p.grad is not None for p in m.parameters())


def test_bn_num_batches_tracked_flushes_to_state_dict():
    """The python-side shadow counter (no per-step GPU kernel) must land
    in state_dict / checkpoints exactly like the eager buffer did."""
    import torch
    from distributeddeeplearning_amd.ops.modules import BatchNorm2d
    bn = BatchNorm2d(8)
    bn.train()
    x = torch.randn(4, 8, 5, 5)
    for _ in range(3):
        bn(x)
    sd = bn.state_dict()
    assert int(sd["num_batches_tracked"]) == 3
    bn2 = BatchNorm2d(8)
    bn2.load_state_dict(sd)
    bn2.train()
    bn2(x)
    assert int(bn2.state_dict()["num_batches_tracked"]) == 4


def test_fork_conv_and_stats_cpu_fallbacks():
    """CPU paths: conv2d_stats -> (y, None); fork_conv2d -> plain fork."""
    import torch
    from distributeddeeplearning_amd.ops import functional as DF
    from distributeddeeplearning_amd.ops.modules import Conv2d
    conv = Conv2d(8, 16, 3, padding=1, emit_bn_stats=True)
    conv.train()
    x = torch.randn(2, 8, 6, 6, requires_grad=True)
    y, partials = DF.conv2d_stats(x, conv.weight, None, 1, 1)
    assert partials is None
    ident, out = DF.fork_conv2d(x, conv)
    assert ident is x
    assert torch.allclose(out, y, atol=1e-6)
    out.sum().backward()
    assert x.grad is not None and conv.weight.grad is not None
