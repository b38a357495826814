#!/usr/bin/env python3
"""Tiny PMC probe: one hot dw_fwd shape (c120 h28 k5 s1), few dispatches."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from yet_another_mobilenet_series_amd.ops.backend import ext

e = ext()
dev = torch.device("cuda:0")
N, C, H, K, S = 256, 120, 28, 5, 1
x = (torch.randn(N, C, H, H, device=dev).bfloat16()
     .contiguous(memory_format=torch.channels_last))
w = (torch.randn(K * K, C, device=dev) * 0.2).bfloat16()
y = torch.empty_like(x)
for _ in range(10):
    e.dw_conv_fwd(x, w, y, K, S, K // 2)
torch.cuda.synchronize()
print("PROF DW DONE")
