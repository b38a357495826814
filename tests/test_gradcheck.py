"""fp64 finite-difference gradcheck of the CPU reference
ops — these are the baselines every HIP kernel's numerics test compares
against, so their own gradients must be exact."""

import torch
from torch.autograd import gradcheck

from yet_another_mobilenet_series_amd.ops import functional as OF


def test_gradcheck_conv_bn_act():
    torch.manual_seed(0)
    x = torch.randn(2, 4, 6, 6, dtype=torch.float64, requires_grad=True)
    w = torch.randn(8, 4, 1, 1, dtype=torch.float64, requires_grad=True)
    g = torch.rand(8, dtype=torch.float64, requires_grad=True) + 0.5
    b = torch.randn(8, dtype=torch.float64, requires_grad=True)
    rm = torch.zeros(8, dtype=torch.float64)
    rv = torch.ones(8, dtype=torch.float64)

    def f(x, w, g, b):
        return OF.conv_bn_act(x, w, g, b, rm.clone(), rv.clone(), stride=1,
                              padding=0, groups=1, act=3, training=True,
                              momentum=0.1, eps=1e-5)

    assert gradcheck(f, (x, w, g, b), eps=1e-6, atol=1e-6)


def test_gradcheck_depthwise():
    torch.manual_seed(1)
    x = torch.randn(2, 4, 6, 6, dtype=torch.float64, requires_grad=True)
    w = torch.randn(4, 1, 3, 3, dtype=torch.float64, requires_grad=True)
    g = torch.rand(4, dtype=torch.float64, requires_grad=True) + 0.5
    b = torch.randn(4, dtype=torch.float64, requires_grad=True)
    rm = torch.zeros(4, dtype=torch.float64)
    rv = torch.ones(4, dtype=torch.float64)

    def f(x, w, g, b):
        return OF.conv_bn_act(x, w, g, b, rm.clone(), rv.clone(), stride=2,
                              padding=1, groups=4, act=1, training=True,
                              momentum=0.1, eps=1e-5)

    assert gradcheck(f, (x, w, g, b), eps=1e-6, atol=1e-6)


def test_gradcheck_squeeze_excite():
    torch.manual_seed(2)
    x = torch.randn(2, 8, 4, 4, dtype=torch.float64, requires_grad=True)
    w1 = (torch.randn(4, 8, 1, 1, dtype=torch.float64) * 0.3
          ).requires_grad_(True)
    b1 = torch.randn(4, dtype=torch.float64, requires_grad=True)
    w2 = (torch.randn(8, 4, 1, 1, dtype=torch.float64) * 0.3
          ).requires_grad_(True)
    b2 = torch.randn(8, dtype=torch.float64, requires_grad=True)

    def f(x, w1, b1, w2, b2):
        return OF.squeeze_excite(x, w1, b1, w2, b2)

    assert gradcheck(f, (x, w1, b1, w2, b2), eps=1e-6, atol=1e-6)


def test_gradcheck_label_smooth_ce():
    torch.manual_seed(3)
    logits = torch.randn(3, 7, dtype=torch.float64, requires_grad=True)
    y = torch.tensor([1, 0, 6])

    def f(logits):
        return OF.label_smooth_ce(logits, y, 0.1)

    assert gradcheck(f, (logits,), eps=1e-6, atol=1e-6)
