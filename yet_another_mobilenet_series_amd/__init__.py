"""yet_another_mobilenet_series_amd — MI355X-native MobileNet-series training framework.

A from-scratch AMD MI355X (gfx950, CDNA4) training framework with the
capabilities of the reference `meijieru/yet_another_mobilenet_series`
(see SURVEY.md §1): ImageNet-class training of MobileNetV1/V2/V3/MNasNet
with label smoothing, weight EMA, SGD/RMSpropTF recipes, YAML configs and
PyTorch-compatible checkpoints.

Compute path: PyTorch-ROCm host/autograd layer + hand-written CDNA4 HIP
kernels (NHWC bf16, pointwise convs as MFMA GEMMs, LDS-tiled depthwise,
fused BN/activation/SE) + RCCL collectives over xGMI.  On CPU every op runs
a pure-PyTorch fp32 reference implementation (the numerics baseline for the
kernel tests).
"""

__version__ = "0.1.0"

from . import models, ops, parallel, utils  # noqa: F401
