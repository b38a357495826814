"""MNasNet-A1 (arXiv:1807.11626 Fig 7, A1 variant with SE ratio 0.25;
SURVEY.md Appendix A.6).  1.0x: 3.905 M params, 311.6 M MACs @224.

SE reduce width is computed from the block INPUT channels (A1 convention,
SURVEY.md Appendix A caveat 2); torchvision's MNasNet is the no-SE B1
family and is not this model.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .mobilenet_base import (ConvBNAct, GlobalAvgPool, InvertedResidual,
                             LinearAct, Dropout, ZooModel, make_divisible)

# (expand_ratio t, out_ch c, repeats n, stride s, kernel k, se_ratio_of_cin)
_CFG = [(6, 24, 2, 2, 3, 0.0), (3, 40, 3, 2, 5, 0.25), (6, 80, 4, 2, 3, 0.0),
        (6, 112, 2, 1, 3, 0.25), (6, 160, 3, 2, 5, 0.25), (6, 320, 1, 1, 3, 0.0)]


class MNasNetA1(ZooModel):
    def __init__(self, num_classes: int = 1000, width_mult: float = 1.0,
                 dropout: float = 0.0, bn_momentum: float = 0.1,
                 bn_epsilon: float = 1e-5, **_: object) -> None:
        super().__init__()
        bn_kw = dict(bn_momentum=bn_momentum, bn_epsilon=bn_epsilon)
        c = lambda v: make_divisible(v * width_mult)  # noqa: E731
        cin = c(32)
        layers = [ConvBNAct(3, cin, 3, 2, act="relu", **bn_kw)]
        # SepConv 3x3 -> 16 (dw + linear pw)
        layers += [ConvBNAct(cin, cin, 3, 1, groups=cin, act="relu", **bn_kw),
                   ConvBNAct(cin, c(16), 1, act="identity", **bn_kw)]
        cin = c(16)
        for t, ch, n, s, k, se in _CFG:
            cout = c(ch)
            for i in range(n):
                se_rd = make_divisible(int(cin * se)) if se else 0
                layers.append(InvertedResidual(
                    cin, cout, k, s if i == 0 else 1, expand_ch=cin * t,
                    se_reduced_ch=se_rd, act="relu",
                    bn_momentum=bn_momentum, bn_epsilon=bn_epsilon))
                cin = cout
        layers.append(ConvBNAct(cin, 1280, 1, act="relu", **bn_kw))
        self.features = nn.Sequential(*layers)
        self.pool = GlobalAvgPool()
        self.dropout = Dropout(dropout) if dropout > 0 else nn.Identity()
        self.classifier = LinearAct(1280, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        self.bump_bn_counters()
        x = self.features(x)
        x = self.pool(x)
        x = self.dropout(x)
        return self.classifier(x)


def Model(**kwargs) -> MNasNetA1:
    return MNasNetA1(**kwargs)
