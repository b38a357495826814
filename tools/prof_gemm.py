#!/usr/bin/env python3
"""Tight loop over the hot pw-GEMM shapes for PMC counter collection."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from yet_another_mobilenet_series_amd.ops.backend import ext

e = ext()
dev = torch.device("cuda:0")
N = 256
for (hw, cin, cout) in [(14, 160, 960), (14, 112, 672), (28, 40, 120),
                        (112, 16, 64)]:
    x = (torch.randn(N, cin, hw, hw, device=dev).bfloat16()
         .contiguous(memory_format=torch.channels_last))
    w = (torch.randn(cout, cin, device=dev) * 0.2).bfloat16()
    z = torch.empty(N, cout, hw, hw, device=dev, dtype=torch.bfloat16,
                    memory_format=torch.channels_last)
    for _ in range(10):
        e.pw_conv(x, w, z)
torch.cuda.synchronize()
print("PROF GEMM DONE")
