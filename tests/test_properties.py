"""Property-based tests (hypothesis) for foundational utilities."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from yet_another_mobilenet_series_amd.utils.common import make_divisible
from yet_another_mobilenet_series_amd.utils.config import (default_config,
                                                           setup_config)
from yet_another_mobilenet_series_amd.utils.optim import LRScheduler


@given(v=st.floats(min_value=0.1, max_value=4096),
       d=st.sampled_from([4, 8, 16]))
def test_make_divisible_properties(v, d):
    """Divisor-rounding per the papers: multiple of d, >= 0.9*v, >= d."""
    r = make_divisible(v, d)
    assert r % d == 0
    assert r >= d
    assert r >= 0.9 * v


@given(warmup=st.integers(min_value=0, max_value=5),
       epochs=st.integers(min_value=1, max_value=20),
       spe=st.integers(min_value=1, max_value=50),
       lr=st.floats(min_value=1e-4, max_value=10.0))
@settings(max_examples=30, deadline=None)
def test_lr_cosine_schedule_properties(warmup, epochs, spe, lr):
    """Warmup rises monotonically to base lr; cosine decays monotonically
    to ~0 at the end; lr never negative."""
    cfg = default_config().optim
    cfg.lr = lr
    cfg.lr_schedule = "cosine"
    cfg.warmup_epochs = warmup
    cfg.warmup_lr = 0.0
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=lr)
    sched = LRScheduler(opt, cfg, steps_per_epoch=spe, epochs=epochs)
    total = epochs * spe
    prev = None
    wsteps = warmup * spe
    for step in range(total + 1):
        cur = sched.lr_at(step, lr)
        assert cur >= 0.0
        assert cur <= lr + 1e-9
        if prev is not None:
            if step <= wsteps and wsteps > 0:
                assert cur >= prev - 1e-9  # warmup monotone up
            elif step > wsteps:
                assert cur <= prev + 1e-9  # cosine monotone down
        prev = cur
    if warmup < epochs:  # warmup >= epochs: the whole run is warmup
        assert abs(sched.lr_at(total, lr)) < 1e-6 * max(lr, 1.0) + 1e-9


@given(keys=st.lists(
    st.sampled_from(["optim.lr", "train.epochs", "model.width_mult",
                     "data.batch_size_per_gpu"]),
    min_size=1, max_size=4, unique=True),
    vals=st.lists(st.integers(min_value=1, max_value=100), min_size=4,
                  max_size=4))
@settings(max_examples=25, deadline=None)
def test_config_override_roundtrip(keys, vals):
    """Dotted CLI overrides land exactly where addressed."""
    args = [f"{k}={v}" for k, v in zip(keys, vals)]
    cfg = setup_config(args)
    for k, v in zip(keys, vals):
        node = cfg
        for part in k.split("."):
            node = node[part]
        assert node == v


@given(n=st.integers(min_value=1, max_value=64),
       world=st.integers(min_value=1, max_value=9))
def test_eval_sampler_partition(n, world):
    """The eval sampler is an exact partition for ANY (n, world)."""
    from yet_another_mobilenet_series_amd.utils.dataflow import (
        EvalDistributedSampler)
    ds = list(range(n))
    shards = [list(EvalDistributedSampler(ds, world, r))
              for r in range(world)]
    flat = sorted(i for s in shards for i in s)
    assert flat == list(range(n))
    for r, s in enumerate(shards):
        assert len(EvalDistributedSampler(ds, world, r)) == len(s)


@given(seed=st.integers(min_value=0, max_value=2**31 - 1),
       n=st.integers(min_value=1, max_value=64))
def test_synthetic_dataset_deterministic(seed, n):
    from yet_another_mobilenet_series_amd.utils.dataflow import (
        SyntheticDataset)
    a = SyntheticDataset(size=64, image_size=8, seed=seed)
    b = SyntheticDataset(size=64, image_size=8, seed=seed)
    i = n - 1
    xa, ya = a[i]
    xb, yb = b[i]
    assert ya == yb and torch.equal(xa, xb)
