"""MobileNetV2 (arXiv:1801.04381 Table 2).

1.0x: 3.505 M params, 300.8 M MACs @224.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .mobilenet_base import (ConvBNAct, GlobalAvgPool, InvertedResidual,
                             LinearAct, Dropout, ZooModel, make_divisible)

# (expand_ratio t, out_channels c, repeats n, first stride s)
_CFG = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
        (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]


class MobileNetV2(ZooModel):
    def __init__(self, num_classes: int = 1000, width_mult: float = 1.0,
                 dropout: float = 0.0, bn_momentum: float = 0.1,
                 bn_epsilon: float = 1e-5, **_: object) -> None:
        super().__init__()
        bn_kw = dict(bn_momentum=bn_momentum, bn_epsilon=bn_epsilon)
        c = lambda v: make_divisible(v * width_mult)  # noqa: E731
        cin = c(32)
        layers = [ConvBNAct(3, cin, 3, 2, act="relu6", **bn_kw)]
        for t, ch, n, s in _CFG:
            cout = c(ch)
            for i in range(n):
                layers.append(InvertedResidual(
                    cin, cout, 3, s if i == 0 else 1, expand_ch=cin * t,
                    act="relu6", bn_momentum=bn_momentum,
                    bn_epsilon=bn_epsilon))
                cin = cout
        last = make_divisible(1280 * max(width_mult, 1.0))
        layers.append(ConvBNAct(cin, last, 1, act="relu6", **bn_kw))
        self.features = nn.Sequential(*layers)
        self.pool = GlobalAvgPool()
        self.dropout = Dropout(dropout) if dropout > 0 else nn.Identity()
        self.classifier = LinearAct(last, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self.bump_bn_counters()
        x = self.features(x)
        x = self.pool(x)
        x = self.dropout(x)
        return self.classifier(x)


def Model(**kwargs) -> MobileNetV2:
    return MobileNetV2(**kwargs)
