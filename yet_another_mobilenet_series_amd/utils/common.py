"""Common utilities: seeding, meters, accuracy, logging (SURVEY.md §2.2 "Common utils")."""

from __future__ import annotations

import json
import logging
import os
import random
import sys
import time
from typing import Iterable, List, Optional, Sequence

import numpy as np
import torch


def set_random_seed(seed: int, deterministic: bool = False) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    if deterministic:
        torch.use_deterministic_algorithms(True, warn_only=True)


class AverageMeter:
    """Tracks a running average of a scalar metric."""

    def __init__(self, name: str = "") -> None:
        self.name = name
        self.reset()

    def reset(self) -> None:
        self.val = 0.0
        self.sum = 0.0
        self.count = 0

    def update(self, val: float, n: int = 1) -> None:
        self.val = float(val)
        self.sum += float(val) * n
        self.count += n

    @property
    def avg(self) -> float:
        return self.sum / max(self.count, 1)


def accuracy(output: torch.Tensor, target: torch.Tensor,
             topk: Sequence[int] = (1,)) -> List[torch.Tensor]:
    """Top-k accuracy (in percent) of `output` logits vs integer `target`."""
    maxk = max(topk)
    _, pred = output.float().topk(maxk, dim=1, largest=True, sorted=True)
    pred = pred.t()
    correct = pred.eq(target.view(1, -1).expand_as(pred))
    res = []
    for k in topk:
        correct_k = correct[:k].reshape(-1).float().sum(0)
        res.append(correct_k.mul_(100.0 / output.size(0)))
    return res


def setup_logger(log_dir: str = "", rank: int = 0, name: str = "yams") -> logging.Logger:
    logger = logging.getLogger(name)
    logger.setLevel(logging.INFO if rank == 0 else logging.WARNING)
    logger.propagate = False
    if logger.handlers:
        return logger
    fmt = logging.Formatter(
        "%(asctime)s %(levelname)s %(message)s", datefmt="%H:%M:%S")
    sh = logging.StreamHandler(sys.stdout)
    sh.setFormatter(fmt)
    logger.addHandler(sh)
    if log_dir and rank == 0:
        os.makedirs(log_dir, exist_ok=True)
        fh = logging.FileHandler(os.path.join(log_dir, "train.log"))
        fh.setFormatter(fmt)
        logger.addHandler(fh)
    return logger


class JsonlWriter:
    """Machine-readable metrics stream: one JSON line per log step."""

    def __init__(self, path: str, enabled: bool = True) -> None:
        self.enabled = enabled
        self._f = None
        if enabled and path:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._f = open(path, "a", buffering=1)

    def write(self, **kv: object) -> None:
        if self._f is None:
            return
        kv.setdefault("time", time.time())
        self._f.write(json.dumps(kv) + "\n")

    def close(self) -> None:
        if self._f is not None:
            self._f.close()
            self._f = None


def make_divisible(v: float, divisor: int = 8, min_value: Optional[int] = None) -> int:
    """Round channel count to the nearest multiple of `divisor`, never dropping >10%.

    Matches the MobileNet papers' channel rounding (SURVEY.md Appendix A
    `mk_div`; arXiv:1801.04381 / 1905.02244 reference implementations).
    """
    if min_value is None:
        min_value = divisor
    new_v = max(min_value, int(v + divisor / 2) // divisor * divisor)
    if new_v < 0.9 * v:
        new_v += divisor
    return new_v


def count_params(module: torch.nn.Module) -> int:
    return sum(p.numel() for p in module.parameters())
