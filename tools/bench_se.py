#!/usr/bin/env python3
"""Micro-benchmark of the fused SE kernels per zoo shape (GPU).

Usage: python tools/bench_se.py [iters]
Respects YAMS_SE_BS (256|1024). Prints us/call per (C, R, HW) and kernel.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from yet_another_mobilenet_series_amd.ops.backend import load_extension

SHAPES = [  # (C, R, HW_side) — V3-Large's 8 SE blocks
    (72, 24, 28), (120, 32, 28), (120, 32, 28), (480, 120, 14),
    (672, 168, 14), (672, 168, 7), (960, 240, 7), (960, 240, 7),
]


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    e = load_extension(required=True)
    dev = torch.device("cuda:0")
    n = 256
    torch.manual_seed(0)
    rows = []
    tot = {"fwd": 0.0, "bwd": 0.0, "wgrad": 0.0}
    for c, r, side in SHAPES:
        hw = side * side
        x = (torch.randn(n, c, side, side, device=dev).bfloat16()
             .contiguous(memory_format=torch.channels_last))
        dy = torch.randn_like(x)
        w1 = torch.randn(r, c, device=dev).bfloat16().contiguous() / c
        b1 = torch.randn(r, device=dev) * 0.1
        w2 = torch.randn(c, r, device=dev).bfloat16().contiguous() / r
        b2 = torch.randn(c, device=dev) * 0.1
        y = torch.empty_like(x)
        dx = torch.empty_like(x)
        pooled = torch.empty(n, c, dtype=torch.float32, device=dev)
        ha = torch.empty(n, r, dtype=torch.float32, device=dev)
        sz = torch.empty(n, c, dtype=torch.float32, device=dev)
        dsz = torch.empty(n, c, dtype=torch.float32, device=dev)
        dhz = torch.empty(n, r, dtype=torch.float32, device=dev)
        dw1 = torch.empty(r, c, dtype=torch.float32, device=dev)
        dw2 = torch.empty(c, r, dtype=torch.float32, device=dev)
        db1 = torch.empty(r, dtype=torch.float32, device=dev)
        db2 = torch.empty(c, dtype=torch.float32, device=dev)

        def t(fn):
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
            return s.elapsed_time(q) * 1000.0 / iters  # us

        w1t = w1.t().contiguous()
        w2t = w2.t().contiguous()
        tf = t(lambda: e.se_fwd(x, w1t, b1, w2t, b2, y, pooled, ha, sz))
        tb = t(lambda: e.se_bwd(dy, x, w1, w2, ha, sz, dx, dsz, dhz))
        tw = t(lambda: e.se_wgrad(dsz, ha, dhz, pooled, dw1, dw2, db1,
                                  db2))
        rows.append((c, r, side, tf, tb, tw))
        tot["fwd"] += tf
        tot["bwd"] += tb
        tot["wgrad"] += tw
    print(f"{'C':>5} {'R':>4} {'HW':>4} {'fwd_us':>8} {'bwd_us':>8} "
          f"{'wgrad_us':>9}")
    for c, r, side, tf, tb, tw in rows:
        print(f"{c:5d} {r:4d} {side:4d} {tf:8.1f} {tb:8.1f} {tw:9.1f}")
    print(f"TOTAL fwd {tot['fwd']:.0f} us  bwd {tot['bwd']:.0f} us  "
          f"wgrad {tot['wgrad']:.0f} us  all {sum(tot.values()):.0f} us")


if __name__ == "__main__":
    main()
