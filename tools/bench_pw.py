#!/usr/bin/env python3
"""Micro-benchmark of the pointwise MFMA GEMM over V3-Large's pw shapes.

Prints us/call and achieved GB/s (A read + C write; B negligible) for
forward, dgrad (transposed weight) and the atb wgrad.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from yet_another_mobilenet_series_amd.ops.backend import load_extension

# (Cin, Cout, H) for N=256 — V3-Large's pw convs, one per
# distinct shape with the head GEMMs
SHAPES = [
    (16, 16, 112), (16, 64, 112), (64, 24, 56), (24, 72, 56), (72, 24, 56),
    (72, 40, 28), (40, 120, 28), (120, 40, 28), (40, 240, 28),
    (240, 80, 14), (80, 200, 14), (200, 80, 14), (80, 184, 14),
    (184, 80, 14), (80, 480, 14), (480, 112, 14), (112, 672, 14),
    (672, 112, 14), (672, 160, 7), (160, 960, 7), (960, 160, 7),
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
    print(f"{'K':>4} {'N':>4} {'H':>4} {'M':>8} | {'fwd_us':>7} {'GB/s':>6}"
          f" | {'dg_us':>7} {'GB/s':>6} | {'wg_us':>7} {'GB/s':>6}")
    for cin, cout, hh in SHAPES:
        m = n * hh * hh
        x = (torch.randn(n, cin, hh, hh, device=dev).bfloat16()
             .contiguous(memory_format=torch.channels_last))
        w = (torch.randn(cout, cin, device=dev) / cin ** 0.5).bfloat16()
        wt = w.t().contiguous()
        y = torch.empty(n, cout, hh, hh, device=dev, dtype=torch.bfloat16,
                        memory_format=torch.channels_last)
        dy = torch.randn_like(y)
        dx = torch.empty_like(x)
        dw = torch.empty(cout, cin, device=dev, dtype=torch.float32)
        ab = m * cin * 2
        cb = m * cout * 2
        tf = t(lambda: e.pw_conv(x, w.contiguous(), y))
        td = t(lambda: e.pw_conv(dy, wt, dx))
        tw = t(lambda: e.pw_conv_wgrad(dy, x, dw))
        tot["fwd"] += tf
        tot["dgrad"] += td
        tot["wgrad"] += tw
        print(f"{cin:4d} {cout:4d} {hh:4d} {m:8d} | {tf:7.1f} "
              f"{(ab + cb)/tf/1e3:6.0f} | {td:7.1f} "
              f"{(ab + cb)/td/1e3:6.0f} | {tw:7.1f} "
              f"{(ab + cb)/tw/1e3:6.0f}")
    print(f"TOTAL fwd {tot['fwd']:.0f} us  dgrad {tot['dgrad']:.0f} us  "
          f"wgrad {tot['wgrad']:.0f} us")


if __name__ == "__main__":
    main()
