"""Model-zoo structure tests: param/MAC ground truth from SURVEY.md App. A."""

import pytest
import torch

from yet_another_mobilenet_series_amd import models

PARAM_COUNTS = [
    ("mobilenet_v1", 1.0, 4_231_976),
    ("mobilenet_v2", 1.0, 3_504_872),
    ("mobilenet_v3_large", 1.0, 5_483_032),
    ("mobilenet_v3_small", 1.0, 2_542_856),
    ("mnasnet_a1", 1.0, 3_904_872),
]


def _params(m):
    return sum(p.numel() for p in m.parameters())


@pytest.mark.parametrize("name,width,expected", PARAM_COUNTS)
def test_param_counts(name, width, expected):
    m = models.build_model(name, width_mult=width, dropout=0.0)
    n = _params(m)
    # SURVEY.md Appendix A verified totals (±0.5% guard for rounding choices)
    assert abs(n - expected) / expected < 0.005, (name, n, expected)


def test_v3_small_075_params():
    m = models.build_model("mobilenet_v3_small", width_mult=0.75, dropout=0.0)
    assert abs(_params(m) - 2_021_000) / 2_021_000 < 0.01


@pytest.mark.parametrize("name", [n for n, _, _ in PARAM_COUNTS])
def test_forward_shape_and_full_grad_coverage(name):
    m = models.build_model(name, dropout=0.0)
    m.train()
    x = torch.randn(2, 3, 224, 224)
    y = m(x)
    assert y.shape == (2, 1000)
    y.sum().backward()
    missing = [n_ for n_, p in m.named_parameters() if p.grad is None]
    assert not missing, f"params without grads: {missing}"


def test_width_mult_rounding():
    # channels round to /8 (SURVEY.md Appendix A conventions)
    m = models.build_model("mobilenet_v2", width_mult=0.75)
    stem = m.features[0].conv
    assert stem.weight.shape[0] % 8 == 0


def test_eval_mode_uses_running_stats():
    m = models.build_model("mobilenet_v2", dropout=0.0)
    m.eval()
    x = torch.randn(2, 3, 224, 224)
    y1 = m(x)
    y2 = m(x)
    assert torch.equal(y1, y2)


def test_se_blocks_present_only_where_expected():
    m = models.build_model("mobilenet_v3_large")
    se_blocks = [b for b in m.features
                 if isinstance(b, models.InvertedResidual) and b.se is not None]
    assert len(se_blocks) == 8  # V3-Large has 8 SE blocks (App. A.3)
    m = models.build_model("mnasnet_a1")
    se_blocks = [b for b in m.features
                 if isinstance(b, models.InvertedResidual) and b.se is not None]
    assert len(se_blocks) == 8  # 3+2+3 SE stages (App. A.6)


def test_custom_num_classes_and_width():
    for name in ("mobilenet_v2", "mobilenet_v3_small"):
        m = models.build_model(name, num_classes=10, width_mult=0.5)
        x = torch.randn(2, 3, 64, 64)
        y = m(x)
        assert y.shape == (2, 10)


def test_dropout_only_in_train_mode():
    torch.manual_seed(0)
    m = models.build_model("mobilenet_v3_large", dropout=0.9)
    m.eval()
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        a, b = m(x), m(x)
    assert torch.equal(a, b)  # dropout inactive in eval
    m.train()
    torch.manual_seed(1)
    c = m(x)
    torch.manual_seed(2)
    d = m(x)
    assert not torch.equal(c, d)  # active (and BN batch stats) in train
