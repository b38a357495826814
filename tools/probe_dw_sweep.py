#!/usr/bin/env python3
"""Roadmap §3 probe: dw_fwd time vs chunks (work per wave) at fixed shape.

If time is ~flat while chunks drops 4-8x, the kernel is launch/ramp
bound, not stream bound. Run with YAMS_DW_CHUNKS_DIV swept by the shell.
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from yet_another_mobilenet_series_amd.ops.backend import ext

e = ext()
dev = torch.device("cuda:0")
N = 256
for (c, h, k, s) in [(120, 28, 5, 1), (480, 14, 3, 1), (64, 112, 3, 1)]:
    x = (torch.randn(N, c, h, h, device=dev).bfloat16()
         .contiguous(memory_format=torch.channels_last))
    w = (torch.randn(k * k, c, device=dev) * 0.2).bfloat16()
    y = torch.empty_like(x)
    for _ in range(5):
        e.dw_conv_fwd(x, w, y, k, s, k // 2)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(20):
        e.dw_conv_fwd(x, w, y, k, s, k // 2)
    t1.record()
    torch.cuda.synchronize()
    print(f"div={os.environ.get('YAMS_DW_CHUNKS_DIV','1'):>2} "
          f"dwfwd c{c:<4d}h{h:<3d}k{k}: {t0.elapsed_time(t1)/20*1000:7.1f} us")
