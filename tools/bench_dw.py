#!/usr/bin/env python3
"""Micro-benchmark of the depthwise kernels over V3-Large's dw shapes.

Prints us/call and achieved GB/s vs the stream floor (read x + write y).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from yet_another_mobilenet_series_amd.ops.backend import load_extension

# (C, H, k, s) — V3-Large's 15 dw layers
SHAPES = [
    (16, 112, 3, 1), (64, 112, 3, 2), (72, 56, 3, 1), (72, 56, 5, 2),
    (120, 28, 5, 1), (120, 28, 5, 1), (240, 28, 3, 2), (200, 14, 3, 1),
    (184, 14, 3, 1), (184, 14, 3, 1), (480, 14, 3, 1), (672, 14, 3, 1),
    (672, 14, 5, 2), (960, 7, 5, 1), (960, 7, 5, 1),
]


def t(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    q = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    q.record()
    torch.cuda.synchronize()
    return s.elapsed_time(q) * 1000.0 / iters


def main():
    e = load_extension(required=True)
    dev = torch.device("cuda:0")
    n = 256
    torch.manual_seed(0)
    tot = {"fwd": 0.0, "dgrad": 0.0, "wgrad": 0.0}
    print(f"{'C':>4} {'H':>4} {'k':>2} {'s':>2} | {'fwd_us':>7} {'GB/s':>6}"
          f" | {'dg_us':>7} {'GB/s':>6} | {'wg_us':>7}")
    for c, h, k, s_ in SHAPES:
        p = k // 2
        ho = (h + 2 * p - k) // s_ + 1
        x = (torch.randn(n, c, h, h, device=dev).bfloat16()
             .contiguous(memory_format=torch.channels_last))
        w = (torch.randn(k * k, c, device=dev) * 0.2).bfloat16().contiguous()
        y = torch.empty(n, c, ho, ho, device=dev, dtype=torch.bfloat16,
                        memory_format=torch.channels_last)
        dy = torch.randn_like(y)
        dx = torch.empty_like(x)
        dw = torch.empty(c, k * k, device=dev, dtype=torch.float32)
        xb = n * c * h * h * 2
        yb = n * c * ho * ho * 2
        tf = t(lambda: e.dw_conv_fwd(x, w, y, k, s_, p))
        td = t(lambda: e.dw_conv_dgrad(dy, w, dx, k, s_, p))
        tw = t(lambda: e.dw_conv_wgrad(x, dy, dw, k, s_, p))
        gf = (xb + yb) / tf / 1e3   # GB/s
        gd = (xb + yb) / td / 1e3
        tot["fwd"] += tf
        tot["dgrad"] += td
        tot["wgrad"] += tw
        print(f"{c:4d} {h:4d} {k:2d} {s_:2d} | {tf:7.1f} {gf:6.0f}"
              f" | {td:7.1f} {gd:6.0f} | {tw:7.1f}")
    print(f"TOTAL fwd {tot['fwd']:.0f} us  dgrad {tot['dgrad']:.0f} us  "
          f"wgrad {tot['wgrad']:.0f} us")


if __name__ == "__main__":
    main()
