"""End-to-end Trainer on the HIP engine (@gpu): full loop with synthetic
data, fused optimizers, EMA, checkpointing."""

import json
import os

import pytest
import torch

from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.utils.config import setup_config

pytestmark = pytest.mark.gpu


def test_trainer_gpu_end_to_end(tmp_path):
    cfg = setup_config([
        "model.name=mobilenet_v3_large", "model.dropout=0.0",
        "data.dataset=synthetic", "data.batch_size_per_gpu=16",
        "data.workers=0", "data.synthetic_size=48", "data.image_size=224",
        "optim.name=rmsprop", "optim.lr=0.005", "optim.warmup_epochs=0",
        "optim.lr_schedule=const",
        "train.epochs=2", "train.dtype=bf16", "train.log_interval=1",
        "train.seed=3", "ema.decay=0.99",
        f"log.dir={tmp_path}/run", "val.interval_epochs=1",
    ])
    t = Trainer(cfg)
    from yet_another_mobilenet_series_amd.utils.optim import FusedRMSpropTF
    assert isinstance(t.optimizer, FusedRMSpropTF)
    t.fit()
    metrics = [json.loads(l) for l in
               open(os.path.join(cfg.log.dir, "metrics.jsonl"))]
    losses = [m["loss"] for m in metrics if m["kind"] == "train"]
    assert all(torch.isfinite(torch.tensor(l)) for l in losses)
    assert losses[-1] < losses[0]
    assert os.path.exists(os.path.join(cfg.log.dir, "checkpoint.pt"))

    # resume continues without error
    cfg2 = setup_config([
        "model.name=mobilenet_v3_large", "model.dropout=0.0",
        "data.dataset=synthetic", "data.batch_size_per_gpu=16",
        "data.workers=0", "data.synthetic_size=48", "data.image_size=224",
        "optim.name=rmsprop", "optim.lr=0.005", "optim.warmup_epochs=0",
        "optim.lr_schedule=const",
        "train.epochs=3", "train.dtype=bf16", "train.log_interval=1",
        "train.seed=3", "ema.decay=0.99",
        f"log.dir={tmp_path}/run", "val.interval_epochs=1",
        f"train.resume={tmp_path}/run/checkpoint.pt",
    ])
    t2 = Trainer(cfg2)
    assert t2.start_epoch == 2
    t2.fit()
