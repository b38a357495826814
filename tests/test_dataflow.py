"""Data pipeline tests: imagefolder loader (PIL path) and transforms."""

import numpy as np
import pytest
import torch

from yet_another_mobilenet_series_amd.utils import dataflow
from yet_another_mobilenet_series_amd.utils.config import default_config


def _make_imagefolder(root, classes=("cat", "dog"), per=3, size=80):
    try:
        from PIL import Image
    except ImportError:
        pytest.skip("PIL not available")
    rng = np.random.default_rng(0)
    for split in ("train", "val"):
        for c in classes:
            d = root / split / c
            d.mkdir(parents=True)
            for i in range(per):
                arr = rng.integers(0, 255, (size, size, 3), dtype=np.uint8)
                Image.fromarray(arr).save(d / f"{i}.jpg")


def test_imagefolder_loaders(tmp_path):
    _make_imagefolder(tmp_path)
    cfg = default_config().data
    cfg.dataset = "imagefolder"
    cfg.root = str(tmp_path)
    cfg.image_size = 64
    cfg.batch_size_per_gpu = 2
    cfg.workers = 0
    train, val = dataflow.get_data_loaders(cfg, seed=0)
    xb, yb = next(iter(train))
    assert xb.shape == (2, 3, 64, 64) and xb.dtype == torch.float32
    assert yb.max().item() <= 1
    xv, yv = next(iter(val))
    assert xv.shape == (2, 3, 64, 64)
    # normalization applied: values roughly centered
    assert xb.abs().mean().item() < 3.0


def test_imagefolder_deterministic_val(tmp_path):
    _make_imagefolder(tmp_path)
    cfg = default_config().data
    cfg.dataset = "imagefolder"
    cfg.root = str(tmp_path)
    cfg.image_size = 64
    cfg.batch_size_per_gpu = 4
    cfg.workers = 0
    _, val1 = dataflow.get_data_loaders(cfg, seed=0)
    _, val2 = dataflow.get_data_loaders(cfg, seed=0)
    a = next(iter(val1))[0]
    b = next(iter(val2))[0]
    assert torch.equal(a, b)  # center-crop eval path has no randomness
