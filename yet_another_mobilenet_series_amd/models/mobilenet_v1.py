"""MobileNetV1 (arXiv:1704.04861 Table 1).

1.0x: 4.232 M params, 568.7 M MACs @224 (verified against the paper in
SURVEY.md Appendix A).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .mobilenet_base import (ConvBNAct, GlobalAvgPool, LinearAct,
                             Dropout, ZooModel, make_divisible)

# (out_channels, stride) for each depthwise-separable stage
_CFG = [(64, 1), (128, 2), (128, 1), (256, 2), (256, 1), (512, 2),
        (512, 1), (512, 1), (512, 1), (512, 1), (512, 1), (1024, 2),
        (1024, 1)]


class MobileNetV1(ZooModel):
    def __init__(self, num_classes: int = 1000, width_mult: float = 1.0,
                 dropout: float = 0.0, bn_momentum: float = 0.1,
                 bn_epsilon: float = 1e-5, **_: object) -> None:
        super().__init__()
        bn_kw = dict(bn_momentum=bn_momentum, bn_epsilon=bn_epsilon)
        c = lambda v: make_divisible(v * width_mult)  # noqa: E731
        layers = [ConvBNAct(3, c(32), 3, 2, act="relu", **bn_kw)]
        cin = c(32)
        for cout, s in _CFG:
            cout = c(cout)
            layers.append(ConvBNAct(cin, cin, 3, s, groups=cin, act="relu",
                                    **bn_kw))
            layers.append(ConvBNAct(cin, cout, 1, act="relu", **bn_kw))
            cin = cout
        self.features = nn.Sequential(*layers)
        self.pool = GlobalAvgPool()
        self.dropout = Dropout(dropout) if dropout > 0 else nn.Identity()
        self.classifier = LinearAct(cin, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self.bump_bn_counters()
        x = self.features(x)
        x = self.pool(x)
        x = self.dropout(x)
        return self.classifier(x)


def Model(**kwargs) -> MobileNetV1:
    return MobileNetV1(**kwargs)
