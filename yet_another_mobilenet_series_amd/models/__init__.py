"""Model zoo: MobileNetV1/V2/V3-Large/V3-Small, MNasNet-A1."""

from __future__ import annotations

from typing import Callable, Dict

import torch.nn as nn

from . import mnasnet, mobilenet_base, mobilenet_v1, mobilenet_v2, mobilenet_v3  # noqa: F401
from .mobilenet_base import ConvBNAct, InvertedResidual, SqueezeExcite  # noqa: F401

_REGISTRY: Dict[str, Callable[..., nn.Module]] = {
    "mobilenet_v1": mobilenet_v1.Model,
    "mobilenet_v2": mobilenet_v2.Model,
    "mobilenet_v3_large": lambda **kw: mobilenet_v3.Model(mode="large", **kw),
    "mobilenet_v3_small": lambda **kw: mobilenet_v3.Model(mode="small", **kw),
    "mnasnet_a1": mnasnet.Model,
}


def model_names():
    return sorted(_REGISTRY)


def build_model(name: str, **kwargs) -> nn.Module:
    try:
        factory = _REGISTRY[name]
    except KeyError:
        raise ValueError(
            f"unknown model {name!r}; available: {model_names()}") from None
    return factory(**kwargs)
