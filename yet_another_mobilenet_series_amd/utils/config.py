"""YAML config system.

Mirrors the reference's UX (SURVEY.md §5.6 [LINEAGE]): configs live in
``apps/*.yml``; a run is selected with a positional ``app:<path>`` argument
and overridden with ``dotted.key=value`` CLI arguments, e.g.::

    python train.py app:apps/mobilenet_v3_large.yml optim.lr=0.8

Implemented as an attribute-access dict (``AttrDict``) plus a typed loader.
Unknown override keys raise (schema guard), unlike the reference, so typos
fail fast; new keys can still be *introduced* with ``+key=value``.
"""

from __future__ import annotations

import ast
import copy
import os
from typing import Any, Dict, List, Optional

import yaml


class AttrDict(dict):
    """dict with attribute access, recursively wrapping nested dicts."""

    def __init__(self, d: Optional[Dict[str, Any]] = None, **kw: Any) -> None:
        super().__init__()
        d = dict(d or {}, **kw)
        for k, v in d.items():
            self[k] = self._wrap(v)

    @classmethod
    def _wrap(cls, v: Any) -> Any:
        if isinstance(v, dict) and not isinstance(v, AttrDict):
            return cls(v)
        if isinstance(v, (list, tuple)):
            return type(v)(cls._wrap(x) for x in v)
        return v

    def __setitem__(self, k: str, v: Any) -> None:
        super().__setitem__(k, self._wrap(v))

    def __getattr__(self, k: str) -> Any:
        try:
            return self[k]
        except KeyError:
            raise AttributeError(k) from None

    def __setattr__(self, k: str, v: Any) -> None:
        self[k] = v

    def __deepcopy__(self, memo: Dict[int, Any]) -> "AttrDict":
        return AttrDict({k: copy.deepcopy(v, memo) for k, v in self.items()})

    def to_dict(self) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        for k, v in self.items():
            if isinstance(v, AttrDict):
                out[k] = v.to_dict()
            elif isinstance(v, (list, tuple)):
                out[k] = [x.to_dict() if isinstance(x, AttrDict) else x for x in v]
            else:
                out[k] = v
        return out


def _parse_value(s: str) -> Any:
    """Parse a CLI override value: YAML-style scalars, lists, dicts."""
    try:
        v = yaml.safe_load(s)
    except yaml.YAMLError:
        try:
            return ast.literal_eval(s)
        except (ValueError, SyntaxError):
            return s
    if isinstance(v, str):
        # YAML 1.1 leaves "1e-2" (no dot) as a string; accept it as float
        try:
            return float(v)
        except ValueError:
            return v
    return v


def _set_dotted(cfg: AttrDict, key: str, value: Any, allow_new: bool) -> None:
    parts = key.split(".")
    node: Any = cfg
    for p in parts[:-1]:
        if p not in node:
            if not allow_new:
                raise KeyError(
                    f"unknown config key {key!r} (use +{key}=... to introduce a new key)"
                )
            node[p] = AttrDict()
        node = node[p]
        if not isinstance(node, AttrDict):
            raise KeyError(f"config key {key!r} descends through non-dict value")
    leaf = parts[-1]
    if leaf not in node and not allow_new:
        raise KeyError(
            f"unknown config key {key!r} (use +{key}=... to introduce a new key)"
        )
    node[leaf] = value


def load_yaml(path: str) -> AttrDict:
    with open(path, "r") as f:
        data = yaml.safe_load(f) or {}
    if not isinstance(data, dict):
        raise ValueError(f"config file {path!r} must contain a mapping at top level")
    return AttrDict(data)


def _merge(base: AttrDict, other: AttrDict) -> AttrDict:
    for k, v in other.items():
        if k in base and isinstance(base[k], AttrDict) and isinstance(v, AttrDict):
            _merge(base[k], v)
        else:
            base[k] = v
    return base


DEFAULTS: Dict[str, Any] = {
    "model": {"name": "mobilenet_v3_large", "width_mult": 1.0, "num_classes": 1000,
              "dropout": 0.2, "bn_momentum": 0.1, "bn_epsilon": 1e-5},
    "data": {"dataset": "synthetic", "root": "", "image_size": 224,
             "batch_size_per_gpu": 256, "workers": 4, "synthetic_size": 12800,
             "color_jitter": 0.4},
    "optim": {"name": "sgd", "lr": 0.4, "momentum": 0.9, "nesterov": True,
              "weight_decay": 4e-5, "no_wd_on_bn_bias": True,
              "rmsprop_alpha": 0.9, "rmsprop_eps": 1e-3,
              "label_smoothing": 0.1,
              "lr_schedule": "cosine", "warmup_epochs": 5, "warmup_lr": 0.0,
              "exp_decay_rate": 0.97, "exp_decay_epochs": 2.4,
              "multistep_milestones": [30, 60, 90], "multistep_gamma": 0.1},
    "ema": {"enabled": True, "decay": 0.9999},
    "train": {"epochs": 150, "seed": 42, "log_interval": 50,
              "grad_accum_steps": 1, "dtype": "bf16",
              "resume": "", "pretrained": "",
              "save_interval_epochs": 1, "save_interval_steps": 0,
              "channels_last": True,
              "deterministic": False, "perf_breakdown": False,
              "trace": False, "trace_steps": 3,
              "engine": "auto", "use_cuda_graph": True},
    "dist": {"backend": "auto", "bucket_mb": None, "overlap": True,
             "grad_dtype": "fp32", "reduce_dtype": "fp32",
             "timeout_sec": 300, "watchdog_sec": 0, "sync_bn": False},
    "log": {"dir": "runs/default", "jsonl": True},
    "val": {"interval_epochs": 1, "batch_size_per_gpu": 256, "use_ema": True},
}


def default_config() -> AttrDict:
    return AttrDict(copy.deepcopy(DEFAULTS))


def setup_config(argv: List[str]) -> AttrDict:
    """Build a config from CLI args: ``[app:<yml>] [dotted.key=value ...]``."""
    cfg = default_config()
    overrides: List[str] = []
    for a in argv:
        if a.startswith("app:"):
            path = a[len("app:"):]
            if not os.path.exists(path):
                raise FileNotFoundError(f"config file not found: {path}")
            _merge(cfg, load_yaml(path))
        elif "=" in a:
            overrides.append(a)
        else:
            raise ValueError(f"unrecognized argument {a!r} (expected app:<yml> or key=value)")
    for ov in overrides:
        key, _, val = ov.partition("=")
        allow_new = key.startswith("+")
        if allow_new:
            key = key[1:]
        _set_dotted(cfg, key, _parse_value(val), allow_new)
    return cfg
