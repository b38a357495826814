"""Reference-path op tests on CPU: the fused functionals must equal the
composition of standard torch ops (these are the numerics baselines the
HIP kernels are later compared against)."""

import torch
import torch.nn.functional as F

from yet_another_mobilenet_series_amd.ops import functional as OF


def test_conv_bn_act_training_matches_composition():
    torch.manual_seed(0)
    x = torch.randn(4, 8, 16, 16)
    w = torch.randn(12, 8, 1, 1) * 0.1
    g = torch.rand(12) + 0.5
    b = torch.randn(12) * 0.1
    rm = torch.zeros(12)
    rv = torch.ones(12)
    rm2, rv2 = rm.clone(), rv.clone()
    y = OF.conv_bn_act(x, w, g, b, rm, rv, stride=1, padding=0, groups=1,
                       act=OF.ACT_RELU6, training=True)
    z = F.conv2d(x, w)
    want = F.relu6(F.batch_norm(z, rm2, rv2, g, b, True, 0.1, 1e-5))
    assert torch.allclose(y, want, atol=1e-6)
    assert torch.allclose(rm, rm2)
    assert torch.allclose(rv, rv2)


def test_conv_bn_act_residual():
    torch.manual_seed(0)
    x = torch.randn(2, 8, 8, 8)
    w = torch.randn(8, 8, 1, 1) * 0.1
    g, b = torch.ones(8), torch.zeros(8)
    rm, rv = torch.zeros(8), torch.ones(8)
    r = torch.randn(2, 8, 8, 8)
    y = OF.conv_bn_act(x, w, g, b, rm, rv, stride=1, padding=0, groups=1,
                       act=OF.ACT_IDENTITY, training=True, residual=r)
    y0 = OF.conv_bn_act(x, w, g, b, rm.clone(), rv.clone(), stride=1,
                        padding=0, groups=1, act=OF.ACT_IDENTITY, training=True)
    assert torch.allclose(y, y0 + r, atol=1e-6)


def test_depthwise_conv_bn_act():
    torch.manual_seed(0)
    x = torch.randn(2, 16, 14, 14)
    w = torch.randn(16, 1, 5, 5) * 0.1
    g, b = torch.rand(16) + 0.5, torch.randn(16) * 0.1
    rm, rv = torch.zeros(16), torch.ones(16)
    y = OF.conv_bn_act(x, w, g, b, rm, rv, stride=2, padding=2, groups=16,
                       act=OF.ACT_HSWISH, training=True)
    z = F.conv2d(x, w, stride=2, padding=2, groups=16)
    want = F.hardswish(F.batch_norm(z, torch.zeros(16), torch.ones(16), g, b,
                                    True, 0.1, 1e-5))
    assert torch.allclose(y, want, atol=1e-6)
    assert y.shape == (2, 16, 7, 7)


def test_squeeze_excite_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(3, 16, 7, 7)
    w1 = torch.randn(8, 16, 1, 1) * 0.2
    b1 = torch.randn(8) * 0.1
    w2 = torch.randn(16, 8, 1, 1) * 0.2
    b2 = torch.randn(16) * 0.1
    y = OF.squeeze_excite(x, w1, b1, w2, b2)
    s = F.adaptive_avg_pool2d(x, 1)
    s = F.relu(F.conv2d(s, w1, b1))
    s = F.hardsigmoid(F.conv2d(s, w2, b2))
    assert torch.allclose(y, x * s, atol=1e-6)


def test_global_avg_pool():
    x = torch.randn(2, 5, 7, 7)
    assert torch.allclose(OF.global_avg_pool(x), x.mean(dim=(2, 3)), atol=1e-6)


def test_linear_act():
    torch.manual_seed(0)
    x = torch.randn(4, 16)
    w = torch.randn(8, 16) * 0.2
    b = torch.randn(8)
    y = OF.linear(x, w, b, OF.ACT_HSWISH)
    assert torch.allclose(y, F.hardswish(F.linear(x, w, b)), atol=1e-6)


def test_hswish_hsigmoid_definitions():
    x = torch.linspace(-5, 5, 101)
    assert torch.allclose(OF.hsigmoid(x), torch.clamp(x + 3, 0, 6) / 6)
    assert torch.allclose(OF.apply_act(x, OF.ACT_HSWISH),
                          x * torch.clamp(x + 3, 0, 6) / 6)


def test_label_smooth_ce_grad():
    torch.manual_seed(0)
    logits = torch.randn(6, 10, requires_grad=True)
    target = torch.randint(0, 10, (6,))
    loss = OF.label_smooth_ce(logits, target, 0.1)
    loss.backward()
    # gradient of CE+smoothing: softmax(p) - ((1-s)·onehot + s/C)
    p = torch.softmax(logits.detach(), 1)
    tgt = F.one_hot(target, 10).float() * 0.9 + 0.1 / 10
    want = (p - tgt) / 6
    assert torch.allclose(logits.grad, want, atol=1e-6)
