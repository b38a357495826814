"""End-to-end trainer tests on CPU (BASELINE.json config[0] class)."""

import json
import os

import torch

from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.utils.config import setup_config


def _smoke_cfg(tmp_path, **over):
    args = [
        "model.name=mobilenet_v2", "model.dropout=0.0",
        "data.dataset=synthetic", "data.batch_size_per_gpu=4",
        "data.workers=0", "data.synthetic_size=8", "data.image_size=64",
        "optim.name=sgd", "optim.lr=0.01", "optim.warmup_epochs=0",
        "optim.lr_schedule=const",
        "train.epochs=1", "train.dtype=fp32", "train.log_interval=1",
        "train.seed=7", "ema.decay=0.99",
        f"log.dir={tmp_path}/run",
        "val.interval_epochs=1",
    ]
    args += [f"{k}={v}" for k, v in over.items()]
    return setup_config(args)


def test_trainer_smoke_loss_decreases(tmp_path):
    cfg = _smoke_cfg(tmp_path, **{"train.epochs": 8})
    t = Trainer(cfg)
    t.fit()
    metrics = [json.loads(l) for l in
               open(os.path.join(cfg.log.dir, "metrics.jsonl"))]
    losses = [m["loss"] for m in metrics if m["kind"] == "train"]
    assert len(losses) >= 2
    assert losses[-1] < losses[0]
    assert os.path.exists(os.path.join(cfg.log.dir, "checkpoint.pt"))


def test_trainer_resume_exact(tmp_path):
    # train 2 epochs straight
    cfg = _smoke_cfg(tmp_path / "a", **{"train.epochs": 2})
    ta = Trainer(cfg)
    ta.fit()
    ref = {k: v.clone() for k, v in ta.model.state_dict().items()}

    # train 1 epoch, checkpoint, resume for epoch 2
    cfg1 = _smoke_cfg(tmp_path / "b", **{"train.epochs": 1})
    tb = Trainer(cfg1)
    tb.fit()
    cfg2 = _smoke_cfg(tmp_path / "b", **{"train.epochs": 2})
    cfg2.train.resume = os.path.join(cfg1.log.dir, "checkpoint.pt")
    tc = Trainer(cfg2)
    assert tc.start_epoch == 1
    tc.fit()
    got = tc.model.state_dict()
    for k in ref:
        assert torch.allclose(ref[k].float(), got[k].float(),
                              atol=1e-5), f"mismatch at {k}"


def test_trainer_validate_runs(tmp_path):
    cfg = _smoke_cfg(tmp_path)
    t = Trainer(cfg)
    top1 = t.validate(0)
    assert 0.0 <= top1 <= 100.0


def test_trainer_determinism(tmp_path):
    """Fixed seed ⇒ identical loss sequence across runs."""
    def losses(d):
        cfg = _smoke_cfg(d)
        t = Trainer(cfg)
        t.fit()
        ms = [json.loads(l) for l in
              open(os.path.join(cfg.log.dir, "metrics.jsonl"))]
        return [m["loss"] for m in ms if m["kind"] == "train"]

    assert losses(tmp_path / "r1") == losses(tmp_path / "r2")


def test_trainer_trace_flag(tmp_path):
    """train.trace writes a chrome trace for the first steps (§5.1c)."""
    cfg = _smoke_cfg(tmp_path, **{"train.trace": True, "train.trace_steps": 2,
                                  "data.synthetic_size": 32})
    Trainer(cfg).fit()
    import glob
    assert glob.glob(str(tmp_path / "run" / "trace_rank*.json")), \
        "no chrome trace"


def test_trainer_grad_accum_two(tmp_path):
    """grad_accum_steps=2: the optimizer steps once per 2 micro-batches
    (note: with BatchNorm, accumulation is NOT numerically identical to a
    bigger batch — each micro-batch normalizes with its own statistics —
    so this checks the step mechanics, while the gradient math itself is
    covered by tests/test_ddp.py::test_grad_accumulation_matches_big_batch)."""
    import torch
    cfg = _smoke_cfg(tmp_path, **{
        "train.grad_accum_steps": 2, "data.batch_size_per_gpu": 2,
        "data.synthetic_size": 8, "train.epochs": 1, "ema.enabled": False})
    t = Trainer(cfg)
    before = [p.detach().clone() for p in t.model.parameters()]
    t.fit()
    # 4 micro-batches of 2 -> 2 optimizer steps
    assert t.global_step == 2
    changed = sum((a != b).any().item()
                  for a, b in zip(before, t.model.parameters()))
    assert changed > 0


def test_trainer_with_packed_dataset(tmp_path):
    """End-to-end Trainer over the packed single-file fast path."""
    import sys

    import numpy as np

    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    try:
        from PIL import Image
    except ImportError:
        import pytest
        pytest.skip("PIL required")
    from tools.make_packed_dataset import pack_folder

    rng = np.random.default_rng(0)
    for split in ("train", "val"):
        for c in ("a", "b"):
            d = tmp_path / "imgs" / split / c
            d.mkdir(parents=True)
            for i in range(3):
                arr = rng.integers(0, 255, (72, 72, 3), dtype=np.uint8)
                Image.fromarray(arr).save(d / f"{i}.jpg")
        pack_folder(str(tmp_path / "imgs" / split),
                    str(tmp_path / f"{split}.pack"))
    cfg = _smoke_cfg(tmp_path, **{
        "data.dataset": "packed", "data.root": str(tmp_path),
        "data.image_size": 64, "data.batch_size_per_gpu": 2,
        "train.epochs": 1, "ema.enabled": False})
    t = Trainer(cfg)
    t.fit()
    assert t.global_step == 3  # 6 train samples / batch 2


def test_trainer_step_interval_checkpoints(tmp_path):
    """SURVEY §5.3 every-N-step checkpoints: a mid-epoch save exists even
    when the per-epoch save never fires, and records the CURRENT epoch as
    the resume point (the interrupted epoch replays from its start)."""
    cfg = _smoke_cfg(tmp_path, **{
        "train.save_interval_steps": 1, "train.save_interval_epochs": 1000,
        "val.interval_epochs": 1000})
    t = Trainer(cfg)
    t.fit()
    path = os.path.join(cfg.log.dir, "checkpoint.pt")
    assert os.path.exists(path)
    state = torch.load(path, map_location="cpu", weights_only=True)
    assert state["epoch"] == 0          # replay epoch 0 on resume
    assert state["global_step"] == t.global_step > 0


def test_train_cli_torchrun_two_ranks(tmp_path):
    """train.py end to end under torchrun (gloo world 2): init_dist,
    param broadcast, bucketed DP engine, cross-rank metric reduce and the
    master-only checkpoint all exercised through the real CLI."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    logdir = tmp_path / "ddp_run"
    env = dict(os.environ, GLOO_SOCKET_IFNAME="lo")
    p = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "train.py",
         "model.name=mobilenet_v2", "model.dropout=0.0",
         "data.dataset=synthetic", "data.batch_size_per_gpu=2",
         "data.workers=0", "data.synthetic_size=8", "data.image_size=64",
         "optim.name=sgd", "optim.lr=0.01", "optim.warmup_epochs=0",
         "optim.lr_schedule=const", "train.epochs=1", "train.dtype=fp32",
         "train.log_interval=1", f"log.dir={logdir}",
         "val.interval_epochs=1"],
        cwd=root, capture_output=True, text=True, timeout=900, env=env)
    assert p.returncode == 0, (p.stdout[-1500:], p.stderr[-1500:])
    assert (logdir / "checkpoint.pt").exists()
    metrics = [json.loads(l) for l in open(logdir / "metrics.jsonl")]
    vals = [m for m in metrics if m["kind"] == "val"]
    assert vals, "no val record from the master rank"


def test_trainer_resume_auto(tmp_path):
    """train.resume=auto: fresh start when no checkpoint exists, then a
    relaunch with the same log.dir resumes from this run's checkpoint —
    the watchdog/torchrun crash-loop restart story."""
    cfg = _smoke_cfg(tmp_path, **{"train.resume": "auto"})
    ta = Trainer(cfg)
    assert ta.start_epoch == 0       # no checkpoint yet -> fresh
    ta.fit()
    cfg2 = _smoke_cfg(tmp_path, **{"train.resume": "auto",
                                   "train.epochs": 2})
    tb = Trainer(cfg2)
    assert tb.start_epoch == 1       # picked up epoch-1 checkpoint
    assert tb.global_step == ta.global_step
