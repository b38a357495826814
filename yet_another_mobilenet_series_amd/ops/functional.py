"""Functional op layer: the kernel surface of the framework.

Every op the reference causes PyTorch to dispatch (SURVEY.md §2.5 K1–K14)
is expressed here as one fused functional.  Dispatch is two-way only:

- CPU tensors (or ``engine_mode() == "reference"``): a pure-PyTorch fp32
  composition — the numerics reference for the kernel tests.
- CUDA(=ROCm) tensors: hand-written CDNA4 HIP kernels via custom
  ``torch.autograd.Function``s (NHWC bf16 compute, fp32 accumulation).

Logical tensor layout is NCHW (PyTorch convention, so checkpoints stay
reference-compatible); on the HIP path tensors are kept in
``torch.channels_last`` memory format, which is exactly the NHWC layout the
gfx950 kernels want.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import hip_ops
from .backend import use_hip, use_syncbn

ACT_IDENTITY = 0
ACT_RELU = 1
ACT_RELU6 = 2
ACT_HSWISH = 3

_ACT_NAMES = {"identity": ACT_IDENTITY, "none": ACT_IDENTITY,
              "relu": ACT_RELU, "relu6": ACT_RELU6,
              "hswish": ACT_HSWISH, "h_swish": ACT_HSWISH}


def act_code(name) -> int:
    if isinstance(name, int):
        return name
    try:
        return _ACT_NAMES[name]
    except KeyError:
        raise ValueError(f"unknown activation {name!r}") from None


def apply_act(x: torch.Tensor, act: int) -> torch.Tensor:
    if act == ACT_IDENTITY:
        return x
    if act == ACT_RELU:
        return F.relu(x)
    if act == ACT_RELU6:
        return F.relu6(x)
    if act == ACT_HSWISH:
        return F.hardswish(x)
    raise ValueError(f"bad act code {act}")


def hsigmoid(x: torch.Tensor) -> torch.Tensor:
    """h-sigmoid(x) = clamp(x+3, 0, 6)/6 (arXiv:1905.02244 §5.2)."""
    return F.hardsigmoid(x)


def conv_bn_act(
    x: torch.Tensor,
    weight: torch.Tensor,
    bn_weight: Optional[torch.Tensor],
    bn_bias: Optional[torch.Tensor],
    running_mean: Optional[torch.Tensor],
    running_var: Optional[torch.Tensor],
    *,
    stride: int = 1,
    padding: int = 0,
    groups: int = 1,
    act: int = ACT_IDENTITY,
    training: bool = False,
    momentum: float = 0.1,
    eps: float = 1e-5,
    residual: Optional[torch.Tensor] = None,
    conv_bias: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Fused conv2d → [BatchNorm] → activation [+ residual].

    Covers the reference's K1 (stem 3x3 s2), K2 (1x1 pointwise), K3/K4
    (depthwise 3x3/5x5), K5 (BN), K6 (activations), K8 (residual add)
    (SURVEY.md §2.5).  BN is skipped when ``bn_weight is None`` (the V3
    head's "NBN" 1x1, which instead carries ``conv_bias``).
    """
    if use_hip(x):
        return hip_ops.conv_bn_act(
            x, weight, bn_weight, bn_bias, running_mean, running_var,
            stride=stride, padding=padding, groups=groups, act=act,
            training=training, momentum=momentum, eps=eps,
            residual=residual, conv_bias=conv_bias)
    z = F.conv2d(x, weight, conv_bias, stride=stride, padding=padding,
                 groups=groups)
    if bn_weight is not None:
        if training and use_syncbn():
            z = _SyncBNFn.apply(z, bn_weight, bn_bias, running_mean,
                                running_var, momentum, eps)
        else:
            z = F.batch_norm(z, running_mean, running_var, bn_weight,
                             bn_bias, training=training, momentum=momentum,
                             eps=eps)
    y = apply_act(z, act)
    if residual is not None:
        y = y + residual
    return y


class _SyncBNFn(torch.autograd.Function):
    """Exact cross-rank BatchNorm (SURVEY.md §2.4 SyncBN option).

    Forward all-reduces per-rank {sum, sumsq, count}; backward
    all-reduces {sum dy, sum dy*zhat} so dz matches single-process BN
    over the concatenated global batch bit-for-math.  Runs on the
    reference engine (``dist.sync_bn: true`` in config forces it); the
    HIP fused path keeps BN local per GPU — the standard recipe at
    batch 256/GPU (SURVEY.md §2.4 "keep local by default").
    """

    @staticmethod
    def forward(ctx, z, gamma, beta, running_mean, running_var, momentum,
                eps):
        import torch.distributed as dist
        n_local = z.numel() // z.shape[1]
        sums = torch.empty(2, z.shape[1], dtype=torch.float64,
                           device=z.device)
        zf = z.detach().double()
        sums[0] = zf.sum(dim=(0, 2, 3))
        sums[1] = (zf * zf).sum(dim=(0, 2, 3))
        dist.all_reduce(sums)
        world = dist.get_world_size()
        cnt = n_local * world
        mean = (sums[0] / cnt).float()
        var = (sums[1] / cnt).float() - mean * mean
        var = var.clamp_min(0)
        invstd = (var + eps).rsqrt()
        if running_mean is not None:
            with torch.no_grad():
                unbiased = var * (cnt / max(cnt - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased,
                                                    alpha=momentum)
        sh = (1, -1, 1, 1)
        zhat = (z - mean.view(sh)) * invstd.view(sh)
        ctx.save_for_backward(zhat, gamma, invstd)
        ctx.cnt = cnt
        return zhat * gamma.view(sh) + beta.view(sh)

    @staticmethod
    def backward(ctx, dy):
        import torch.distributed as dist
        zhat, gamma, invstd = ctx.saved_tensors
        sh = (1, -1, 1, 1)
        red = torch.empty(2, dy.shape[1], dtype=torch.float64,
                          device=dy.device)
        red[0] = dy.double().sum(dim=(0, 2, 3))
        red[1] = (dy.double() * zhat.double()).sum(dim=(0, 2, 3))
        dist.all_reduce(red)
        dbeta = red[0].float()
        dgamma = red[1].float()
        m = ctx.cnt
        dz = (gamma.view(sh) * invstd.view(sh)) * (
            dy - dbeta.view(sh) / m - zhat * dgamma.view(sh) / m)
        return (dz.to(dy.dtype), dgamma.to(gamma.dtype),
                dbeta.to(gamma.dtype), None, None, None, None)


def squeeze_excite(
    x: torch.Tensor,
    w1: torch.Tensor, b1: torch.Tensor,
    w2: torch.Tensor, b2: torch.Tensor,
) -> torch.Tensor:
    """Squeeze-and-excite (K7): global-pool → FC(reduce) → ReLU → FC(expand)
    → h-sigmoid → channel scale.  w1: [R, C, 1, 1], w2: [C, R, 1, 1]."""
    if use_hip(x):
        return hip_ops.squeeze_excite(x, w1, b1, w2, b2)
    s = F.adaptive_avg_pool2d(x, 1)
    s = F.relu(F.conv2d(s, w1, b1))
    s = hsigmoid(F.conv2d(s, w2, b2))
    return x * s


def global_avg_pool(x: torch.Tensor) -> torch.Tensor:
    """Global average pool (K9): [N, C, H, W] → [N, C]."""
    if use_hip(x):
        return hip_ops.global_avg_pool(x)
    return F.adaptive_avg_pool2d(x, 1).flatten(1)


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None,
           act: int = ACT_IDENTITY) -> torch.Tensor:
    """FC layer (K10) with optional fused activation. weight: [out, in]."""
    if use_hip(x):
        return hip_ops.linear(x, weight, bias, act)
    return apply_act(F.linear(x, weight, bias), act)


def dropout(x: torch.Tensor, p: float, training: bool) -> torch.Tensor:
    """Dropout (K14).  On the HIP path: a stateless counter-hash kernel —
    the mask is a pure function of (seed, index), regenerated bit-exactly
    in backward, nothing stored (VERDICT round-1
    item 10 — the last eager torch kernel in the training hot path)."""
    if not training or p <= 0.0:
        return x
    if use_hip(x):
        return hip_ops.dropout(x, p)
    return F.dropout(x, p, training=True)


def label_smooth_ce(logits: torch.Tensor, target: torch.Tensor,
                    smoothing: float = 0.1) -> torch.Tensor:
    """Label-smoothed cross-entropy (K11), mean over the batch.

    loss_i = (1-s) * (-log p[y_i]) + s * mean_c(-log p[c])
    (the reference's CrossEntropyLabelSmooth semantics).
    """
    if use_hip(logits):
        return hip_ops.label_smooth_ce(logits, target, smoothing)
    if logits.dtype not in (torch.float32, torch.float64):
        logits = logits.float()  # upcast bf16; keep fp64 for gradcheck
    logp = F.log_softmax(logits, dim=1)
    nll = -logp.gather(1, target.view(-1, 1)).squeeze(1)
    smooth = -logp.mean(dim=1)
    return ((1.0 - smoothing) * nll + smoothing * smooth).mean()
