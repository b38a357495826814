"""Generic MobileNet-series building blocks (SURVEY.md §2.2 "Generic blocks").

Modules hold standard PyTorch parameters (OIHW fp32 conv weights,
BatchNorm2d affine + running stats) so checkpoints stay compatible with the
reference's ``torch.save`` dict convention; compute routes
through :mod:`..ops.functional`, which dispatches CPU→pure-PyTorch
reference, GPU→CDNA4 HIP kernels.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from ..ops import functional as OF
from ..utils.common import make_divisible


class ZooModel(nn.Module):
    """Base class for zoo models: one batched BN-counter bump per forward.

    BatchNorm's ``num_batches_tracked`` counters are bookkeeping, but
    incrementing them per-module launched ~46 eager ``add<long>`` kernels
    every training step (~0.2 ms of pure launch overhead on MI355X).  The
    zoo models instead call :meth:`bump_bn_counters` once per forward —
    a single ``torch._foreach_add_`` over every counter in the model.
    """

    def bump_bn_counters(self) -> None:
        if not self.training:
            return
        counters = getattr(self, "_bn_counters", None)
        if counters is None:
            counters = [m.num_batches_tracked for m in self.modules()
                        if isinstance(m, nn.BatchNorm2d)
                        and m.track_running_stats
                        and m.num_batches_tracked is not None]
            self._bn_counters = counters
        if counters:
            torch._foreach_add_(counters, 1)

    def _apply(self, fn, recurse=True):
        # .to()/.cuda() replace buffer tensors; drop the cached list
        self._bn_counters = None
        return super()._apply(fn, recurse)


class ConvBNAct(nn.Module):
    """conv2d → BatchNorm → activation, fused on the HIP path.

    Covers the zoo's K1 stem (3x3 s2 dense), K2 pointwise 1x1 and K3/K4
    depthwise 3x3/5x5 conv shapes.
    """

    def __init__(self, cin: int, cout: int, kernel_size: int = 1,
                 stride: int = 1, groups: int = 1, act: str = "relu6",
                 bn_momentum: float = 0.1, bn_epsilon: float = 1e-5) -> None:
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, kernel_size, stride,
                              padding=kernel_size // 2, groups=groups,
                              bias=False)
        self.bn = nn.BatchNorm2d(cout, eps=bn_epsilon, momentum=bn_momentum)
        self.act = OF.act_code(act)
        self.stride = stride
        self.padding = kernel_size // 2
        self.groups = groups

    def forward(self, x: torch.Tensor,
                residual: Optional[torch.Tensor] = None) -> torch.Tensor:
        # num_batches_tracked is bumped once per model forward by
        # ZooModel.bump_bn_counters (batched), not per module
        return OF.conv_bn_act(
            x, self.conv.weight, self.bn.weight, self.bn.bias,
            self.bn.running_mean, self.bn.running_var,
            stride=self.stride, padding=self.padding, groups=self.groups,
            act=self.act, training=self.training,
            momentum=self.bn.momentum, eps=self.bn.eps, residual=residual)


class SqueezeExcite(nn.Module):
    """SE module: pool → 1x1 reduce → ReLU → 1x1 expand → h-sigmoid → scale.

    ``reduced_ch`` is the bottleneck width (already divisor-rounded by the
    caller, per-paper convention; SURVEY.md Appendix A caveat 1).
    """

    def __init__(self, channels: int, reduced_ch: int) -> None:
        super().__init__()
        self.fc1 = nn.Conv2d(channels, reduced_ch, 1, bias=True)
        self.fc2 = nn.Conv2d(reduced_ch, channels, 1, bias=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return OF.squeeze_excite(x, self.fc1.weight, self.fc1.bias,
                                 self.fc2.weight, self.fc2.bias)


class InvertedResidual(nn.Module):
    """MobileNetV2-style inverted residual (arXiv:1801.04381 §3.3) with the
    V3/MNasNet extensions: configurable kernel size, SE, activation
    (arXiv:1905.02244 §5; 1807.11626 Fig 7).

    ``expand_ch`` is the absolute expanded width (the zoo's tables give
    absolute expansion, not ratios — SURVEY.md Appendix A).
    """

    def __init__(self, cin: int, cout: int, kernel_size: int, stride: int,
                 expand_ch: int, se_reduced_ch: int = 0, act: str = "relu6",
                 bn_momentum: float = 0.1, bn_epsilon: float = 1e-5) -> None:
        super().__init__()
        self.use_residual = (stride == 1 and cin == cout)
        bn_kw = dict(bn_momentum=bn_momentum, bn_epsilon=bn_epsilon)
        self.expand = (ConvBNAct(cin, expand_ch, 1, act=act, **bn_kw)
                       if expand_ch != cin else None)
        self.dw = ConvBNAct(expand_ch, expand_ch, kernel_size, stride,
                            groups=expand_ch, act=act, **bn_kw)
        self.se = (SqueezeExcite(expand_ch, se_reduced_ch)
                   if se_reduced_ch else None)
        self.project = ConvBNAct(expand_ch, cout, 1, act="identity", **bn_kw)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from ..ops.backend import use_fused_blocks, use_hip
        if use_hip(x) and use_fused_blocks():
            from ..ops.fused_block import fused_inverted_residual
            return fused_inverted_residual(x, self)
        y = self.expand(x) if self.expand is not None else x
        y = self.dw(y)
        if self.se is not None:
            y = self.se(y)
        return self.project(y, residual=x if self.use_residual else None)


class GlobalAvgPool(nn.Module):
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return OF.global_avg_pool(x)


class Dropout(nn.Module):
    """Head dropout routed through the op layer (K14): HIP counter-hash
    kernel on GPU, torch dropout on CPU.  State-dict compatible with
    nn.Dropout (both are parameter-free)."""

    def __init__(self, p: float) -> None:
        super().__init__()
        self.p = float(p)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return OF.dropout(x, self.p, self.training)

    def extra_repr(self) -> str:
        return f"p={self.p}"


class LinearAct(nn.Module):
    """FC layer with optional fused activation (the V3 head's post-pool
    1x1-no-BN convs are expressed as Linear: spatial is 1x1 there)."""

    def __init__(self, cin: int, cout: int, act: str = "identity",
                 bias: bool = True) -> None:
        super().__init__()
        self.fc = nn.Linear(cin, cout, bias=bias)
        self.act = OF.act_code(act)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return OF.linear(x, self.fc.weight, self.fc.bias, self.act)


__all__ = ["ZooModel", "ConvBNAct", "SqueezeExcite", "InvertedResidual",
           "GlobalAvgPool", "LinearAct", "Dropout", "make_divisible"]
