#!/usr/bin/env python3
"""Inference (eval-mode) throughput over the zoo on one MI355X.

Forward-only, BN running stats, bf16 NHWC, synthetic input — the
serving-shaped counterpart of bench.py's training step.  Prints one row
per (model, batch): ms/batch and img/s.

Usage: python tools/bench_infer.py [--models m1,m2] [--batches 1,8,256]
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from yet_another_mobilenet_series_amd.models import build_model


@torch.no_grad()
def time_model(model, batch, iters=50, warmup=10, graph=False):
    x = torch.randn(batch, 3, 224, 224, device="cuda") \
        .bfloat16().contiguous(memory_format=torch.channels_last)
    fn = lambda: model(x)  # noqa: E731
    if graph:
        # static shapes + eval mode (no stats writes, no side streams):
        # the ~190-launch forward collapses to one graph replay, which
        # is where small-batch serving latency lives
        s0 = torch.cuda.Stream()
        s0.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s0):
            for _ in range(3):
                model(x)
        torch.cuda.current_stream().wait_stream(s0)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            model(x)
        fn = g.replay
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    q = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    q.record()
    torch.cuda.synchronize()
    ms = s.elapsed_time(q) / iters
    return ms, batch / ms * 1000.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--models", default="mobilenet_v1,mobilenet_v2,"
                    "mobilenet_v3_large,mobilenet_v3_small,mnasnet_a1")
    ap.add_argument("--batches", default="1,8,64,256")
    ap.add_argument("--graph", action="store_true",
                    help="capture the forward in a hipGraph, time replay")
    args = ap.parse_args()
    batches = [int(b) for b in args.batches.split(",")]
    print(f"{'model':22s} " + " ".join(f"{('b' + str(b)):>14s}" for b in batches))
    for name in args.models.split(","):
        torch.manual_seed(0)
        model = build_model(name).to("cuda") \
            .to(memory_format=torch.channels_last).eval()
        cols = []
        for b in batches:
            ms, ips = time_model(model, b, graph=args.graph)
            cols.append(f"{ms:6.2f}ms {ips:>7.0f}/s" if ips < 99999
                        else f"{ms:6.2f}ms {ips / 1000:>6.1f}k/s")
        print(f"{name:22s} " + " ".join(f"{c:>14s}" for c in cols))


if __name__ == "__main__":
    main()
