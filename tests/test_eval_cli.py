"""eval.py CLI: validate a saved checkpoint end-to-end on CPU."""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import eval as eval_cli
from yet_another_mobilenet_series_amd.engine import Trainer
from yet_another_mobilenet_series_amd.utils.config import setup_config


def _args(tmp_path, **over):
    base = [
        "model.name=mobilenet_v2", "model.dropout=0.0",
        "data.dataset=synthetic", "data.batch_size_per_gpu=4",
        "data.workers=0", "data.synthetic_size=8", "data.image_size=64",
        "optim.lr=0.01", "optim.warmup_epochs=0",
        "train.epochs=1", "train.dtype=fp32", "train.log_interval=1",
        f"log.dir={tmp_path}/run",
    ]
    return base + [f"{k}={v}" for k, v in over.items()]


def test_eval_cli_runs_on_checkpoint(tmp_path, capsys):
    args = _args(tmp_path)
    Trainer(setup_config(args)).fit()
    ckpt = os.path.join(str(tmp_path), "run", "checkpoint.pt")
    assert os.path.exists(ckpt)
    eval_cli.main(_args(tmp_path, **{"train.resume": ckpt,
                                     "log.dir": f"{tmp_path}/eval"}))
    # the logger is a process-wide singleton, so assert on the eval run's
    # own metrics stream instead of its log file
    lines = open(os.path.join(str(tmp_path), "eval",
                              "metrics.jsonl")).read().splitlines()
    assert any('"kind": "val"' in ln for ln in lines)
