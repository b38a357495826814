"""Tests for the round-2 ADVICE/VERDICT fixes (all CPU)."""

import torch

from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.models import build_model
from yet_another_mobilenet_series_amd.ops.hip_ops import _packed
from yet_another_mobilenet_series_amd.utils.dataflow import (
    EvalDistributedSampler)
from yet_another_mobilenet_series_amd.utils.optim import _mark_updated

from tests.test_trainer import _smoke_cfg


def test_pack_cache_invalidated_after_raw_update():
    """ADVICE round-1 HIGH: a fused-kernel (raw data_ptr) param update must
    invalidate the bf16 pack cache, else conv weights freeze."""
    p = torch.nn.Parameter(torch.ones(8))
    packed1 = _packed(p, "k", lambda t: t.detach() * 2)
    # cache hit while untouched
    assert _packed(p, "k", lambda t: t.detach() * 2) is packed1
    # simulate a raw-pointer write: mutate storage without the dispatcher
    p.detach().numpy()[:] = 3.0
    assert _packed(p, "k", lambda t: t.detach() * 2) is packed1  # stale!
    _mark_updated([p])  # what FusedSGD/FusedRMSpropTF now call
    packed2 = _packed(p, "k", lambda t: t.detach() * 2)
    assert packed2 is not packed1
    assert torch.equal(packed2, torch.full((8,), 6.0))


def test_lr_schedule_counts_optimizer_steps(tmp_path):
    """ADVICE round-1 medium: with grad accumulation the schedule spans
    optimizer steps, not microbatches."""
    cfg = _smoke_cfg(tmp_path, **{
        "train.grad_accum_steps": 2, "data.batch_size_per_gpu": 2,
        "data.synthetic_size": 8, "optim.lr_schedule": "cosine",
        "optim.lr": 1.0, "ema.enabled": False})
    t = Trainer(cfg)
    # 4 microbatches/epoch, accum 2 -> 2 optimizer steps/epoch, 1 epoch
    assert t.lr_sched.total_steps == 2
    # cosine reaches its floor exactly at the last optimizer step
    assert abs(t.lr_sched.lr_at(2, 1.0)) < 1e-9
    assert t.lr_sched.lr_at(1, 1.0) > 0.4


def test_trailing_microbatches_step(tmp_path):
    """ADVICE round-1 low: an epoch whose length is not divisible by accum
    still applies the trailing microbatches."""
    cfg = _smoke_cfg(tmp_path, **{
        "train.grad_accum_steps": 3, "data.batch_size_per_gpu": 2,
        "data.synthetic_size": 8, "train.epochs": 1, "ema.enabled": False})
    t = Trainer(cfg)
    t.fit()
    # 4 microbatches, accum 3 -> boundary at it=2 AND the trailing it=3
    assert t.global_step == 2


def test_eval_sampler_no_padding():
    """ADVICE round-1 low: val shards are uneven, never padded."""
    ds = list(range(10))  # len() is all the sampler needs
    shards = [list(EvalDistributedSampler(ds, 4, r)) for r in range(4)]
    flat = sorted(i for s in shards for i in s)
    assert flat == list(range(10))  # exact cover, no duplicates
    assert [len(s) for s in shards] == [3, 3, 2, 2]
    for r, s in enumerate(shards):
        assert len(EvalDistributedSampler(ds, 4, r)) == len(s)


def test_checkpoint_weights_only_safe(tmp_path):
    """Our checkpoint format loads fully under weights_only=True."""
    from yet_another_mobilenet_series_amd.utils import checkpoint as ckpt
    m = build_model("mobilenet_v2", dropout=0.0, width_mult=0.35)
    path = str(tmp_path / "c.pt")
    ckpt.save_checkpoint(path, m, epoch=3)
    state = torch.load(path, weights_only=True)  # must not raise
    assert state["epoch"] == 3
    # and the RNG state round-trips
    torch.manual_seed(5)
    import numpy as np
    import random
    random.seed(5)
    np.random.seed(5)
    ckpt.save_checkpoint(path, m, epoch=4)
    want = (random.random(), np.random.rand(), torch.rand(()).item())
    ckpt.load_checkpoint(path, m, restore_rng=True)
    got = (random.random(), np.random.rand(), torch.rand(()).item())
    assert want == got


def test_bn_counters_batched():
    """num_batches_tracked advances once per forward for every BN, via a
    single foreach op (VERDICT round-1 weak #5)."""
    m = build_model("mobilenet_v3_small", dropout=0.0)
    m.train()
    x = torch.randn(2, 3, 64, 64)
    m(x)
    m(x)
    counts = {mm.num_batches_tracked.item() for mm in m.modules()
              if isinstance(mm, torch.nn.BatchNorm2d)}
    assert counts == {2}
    m.eval()
    with torch.no_grad():
        m(x)
    counts = {mm.num_batches_tracked.item() for mm in m.modules()
              if isinstance(mm, torch.nn.BatchNorm2d)}
    assert counts == {2}
