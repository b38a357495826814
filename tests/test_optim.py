import math

import torch
import torch.nn as nn

from yet_another_mobilenet_series_amd.utils.config import default_config
from yet_another_mobilenet_series_amd.utils.optim import (
    CrossEntropyLabelSmooth, ExponentialMovingAverage, LRScheduler, RMSpropTF,
    get_optimizer, split_param_groups)


def test_label_smooth_ce_matches_manual():
    torch.manual_seed(0)
    logits = torch.randn(8, 10)
    target = torch.randint(0, 10, (8,))
    crit = CrossEntropyLabelSmooth(0.1)
    loss = crit(logits, target)
    logp = torch.log_softmax(logits, 1)
    nll = -logp[torch.arange(8), target]
    smooth = -logp.mean(1)
    want = (0.9 * nll + 0.1 * smooth).mean()
    assert torch.allclose(loss, want, atol=1e-6)


def test_label_smooth_zero_equals_ce():
    torch.manual_seed(0)
    logits = torch.randn(8, 10)
    target = torch.randint(0, 10, (8,))
    loss = CrossEntropyLabelSmooth(0.0)(logits, target)
    want = nn.functional.cross_entropy(logits, target)
    assert torch.allclose(loss, want, atol=1e-6)


def test_rmsprop_tf_semantics():
    """TF semantics: ms init 1, eps inside sqrt, momentum accumulates lr-scaled."""
    p = torch.nn.Parameter(torch.tensor([2.0]))
    opt = RMSpropTF([p], lr=0.1, alpha=0.9, eps=1e-3, momentum=0.9)
    p.grad = torch.tensor([1.0])
    opt.step()
    ms = 0.9 * 1.0 + 0.1 * 1.0  # init 1!
    mom = 0.1 * 1.0 / math.sqrt(ms + 1e-3)
    want = 2.0 - mom
    assert torch.allclose(p.data, torch.tensor([want]), atol=1e-6)
    p.grad = torch.tensor([0.5])
    opt.step()
    ms2 = 0.9 * ms + 0.1 * 0.25
    mom2 = 0.9 * mom + 0.1 * 0.5 / math.sqrt(ms2 + 1e-3)
    want2 = want - mom2
    assert torch.allclose(p.data, torch.tensor([want2]), atol=1e-6)


def test_param_group_split():
    m = nn.Sequential(nn.Conv2d(3, 8, 3, bias=False), nn.BatchNorm2d(8),
                      nn.Linear(8, 4))
    groups = split_param_groups(m, weight_decay=1e-4)
    decay, no_decay = groups[0]["params"], groups[1]["params"]
    assert groups[0]["weight_decay"] == 1e-4
    assert groups[1]["weight_decay"] == 0.0
    # conv weight + linear weight decay; BN affine + linear bias don't
    assert len(decay) == 2
    assert len(no_decay) == 3
    assert sum(p.numel() for g in groups for p in g["params"]) == \
        sum(p.numel() for p in m.parameters())


def _sched(schedule, warmup_epochs=0, steps_per_epoch=10, epochs=10, lr=1.0):
    cfg = default_config().optim
    cfg.lr_schedule = schedule
    cfg.warmup_epochs = warmup_epochs
    p = torch.nn.Parameter(torch.zeros(1))
    opt = torch.optim.SGD([p], lr=lr)
    return LRScheduler(opt, cfg, steps_per_epoch, epochs), opt


def test_lr_warmup_then_cosine():
    sched, opt = _sched("cosine", warmup_epochs=2)
    lrs = [sched.step(s) for s in range(100)]
    assert lrs[0] == 0.0
    assert abs(lrs[20] - 1.0) < 1e-6  # warmup done
    assert lrs[-1] < 0.01             # cosine decayed
    assert all(a <= b + 1e-9 for a, b in zip(lrs[:20], lrs[1:21]))


def test_lr_exponential():
    sched, _ = _sched("exp")
    cfg_rate = sched.exp_decay_rate
    lr0 = sched.step(0)
    lr_after = sched.step(sched.exp_decay_steps)
    assert abs(lr_after / lr0 - cfg_rate) < 1e-6


def test_lr_multistep():
    sched, _ = _sched("multistep", epochs=100)
    sched.milestones = [50, 80]
    assert sched.step(10) == 1.0
    assert abs(sched.step(60) - 0.1) < 1e-9
    assert abs(sched.step(90) - 0.01) < 1e-9


def test_ema_update_and_swap():
    m = nn.Sequential(nn.Linear(4, 4), nn.BatchNorm1d(4))
    ema = ExponentialMovingAverage(m, decay=0.5)
    w0 = m[0].weight.detach().clone()
    with torch.no_grad():
        m[0].weight += 1.0
        m[1].running_mean += 2.0
    ema.update()
    want = 0.5 * w0 + 0.5 * (w0 + 1.0)
    assert torch.allclose(ema.shadow["0.weight"], want)
    assert torch.allclose(ema.shadow["1.running_mean"],
                          torch.full((4,), 1.0))
    live = m[0].weight.detach().clone()
    ema.swap()
    assert torch.allclose(m[0].weight, want)
    ema.swap()
    assert torch.allclose(m[0].weight, live)


def test_get_optimizer_variants():
    m = nn.Linear(4, 4)
    cfg = default_config().optim
    assert isinstance(get_optimizer(m, cfg), torch.optim.SGD)
    cfg.name = "rmsprop"
    assert isinstance(get_optimizer(m, cfg), RMSpropTF)
