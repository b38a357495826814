"""MobileNetV3 Large & Small (arXiv:1905.02244 Tables 1–2).

Large 1.0x: 5.483 M params, 216.6 M MACs @224.
Small 1.0x: 2.543 M params,  56.5 M MACs @224.
SE reduce width = make_divisible(expanded/4, 8) (SURVEY.md Appendix A
caveat 1).  Head: pw→(960|576) HS → pool → FC(1280|1024) HS no-BN → FC.
"""

from __future__ import annotations

from typing import List, Tuple

import torch
import torch.nn as nn

from .mobilenet_base import (ConvBNAct, GlobalAvgPool, InvertedResidual,
                             LinearAct, Dropout, ZooModel, make_divisible)

# (kernel, expand_ch, out_ch, use_se, act, stride)
_LARGE: List[Tuple[int, int, int, bool, str, int]] = [
    (3, 16, 16, False, "relu", 1),
    (3, 64, 24, False, "relu", 2),
    (3, 72, 24, False, "relu", 1),
    (5, 72, 40, True, "relu", 2),
    (5, 120, 40, True, "relu", 1),
    (5, 120, 40, True, "relu", 1),
    (3, 240, 80, False, "hswish", 2),
    (3, 200, 80, False, "hswish", 1),
    (3, 184, 80, False, "hswish", 1),
    (3, 184, 80, False, "hswish", 1),
    (3, 480, 112, True, "hswish", 1),
    (3, 672, 112, True, "hswish", 1),
    (5, 672, 160, True, "hswish", 2),
    (5, 960, 160, True, "hswish", 1),
    (5, 960, 160, True, "hswish", 1),
]
_SMALL: List[Tuple[int, int, int, bool, str, int]] = [
    (3, 16, 16, True, "relu", 2),
    (3, 72, 24, False, "relu", 2),
    (3, 88, 24, False, "relu", 1),
    (5, 96, 40, True, "hswish", 2),
    (5, 240, 40, True, "hswish", 1),
    (5, 240, 40, True, "hswish", 1),
    (5, 120, 48, True, "hswish", 1),
    (5, 144, 48, True, "hswish", 1),
    (5, 288, 96, True, "hswish", 2),
    (5, 576, 96, True, "hswish", 1),
    (5, 576, 96, True, "hswish", 1),
]


class MobileNetV3(ZooModel):
    def __init__(self, mode: str = "large", num_classes: int = 1000,
                 width_mult: float = 1.0, dropout: float = 0.2,
                 bn_momentum: float = 0.1, bn_epsilon: float = 1e-5,
                 **_: object) -> None:
        super().__init__()
        bn_kw = dict(bn_momentum=bn_momentum, bn_epsilon=bn_epsilon)
        cfg = _LARGE if mode == "large" else _SMALL
        last_pw = 960 if mode == "large" else 576
        head_ch = 1280 if mode == "large" else 1024
        c = lambda v: make_divisible(v * width_mult)  # noqa: E731
        cin = c(16)
        layers = [ConvBNAct(3, cin, 3, 2, act="hswish", **bn_kw)]
        for k, exp, out, se, act, s in cfg:
            exp_c, out_c = c(exp), c(out)
            se_rd = make_divisible(exp_c // 4) if se else 0
            layers.append(InvertedResidual(cin, out_c, k, s, expand_ch=exp_c,
                                           se_reduced_ch=se_rd, act=act,
                                           bn_momentum=bn_momentum,
                                           bn_epsilon=bn_epsilon))
            cin = out_c
        lp = c(last_pw)
        layers.append(ConvBNAct(cin, lp, 1, act="hswish", **bn_kw))
        self.features = nn.Sequential(*layers)
        self.pool = GlobalAvgPool()
        # paper: "conv 1x1, NBN" head at 1x1 spatial == Linear with bias
        self.head = LinearAct(lp, head_ch, act="hswish")
        self.dropout = Dropout(dropout) if dropout > 0 else nn.Identity()
        self.classifier = LinearAct(head_ch, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self.bump_bn_counters()
        x = self.features(x)
        x = self.pool(x)
        x = self.head(x)
        x = self.dropout(x)
        return self.classifier(x)


def Model(**kwargs) -> MobileNetV3:
    return MobileNetV3(**kwargs)
