#!/usr/bin/env python3
"""Per-kernel A/B microbench on real V3-Large layer shapes (GPU box).

Times fused pieces vs their unfused equivalents with hipEvents so the
fusion win/loss per kernel is explicit.
"""

import sys

import torch

sys.path.insert(0, ".")
from yet_another_mobilenet_series_amd.ops.backend import ext

BF16 = torch.bfloat16
CL = torch.channels_last
dev = torch.device("cuda:0")
e = ext()
N = 256


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    t = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    t.record()
    torch.cuda.synchronize()
    return s.elapsed_time(t) / iters * 1000  # us


def nhwc(n, c, h, w):
    return (torch.randn(n, c, h, w, device=dev).to(BF16)
            .contiguous(memory_format=CL))


print(f"{'case':46s} {'us':>9s}")

# ---- dw conv shapes (V3-Large hot layers) ------------------------------
for (c, h, k, s) in [(72, 56, 5, 2), (120, 28, 5, 1), (240, 28, 3, 2),
                     (480, 14, 3, 1), (672, 14, 5, 2), (960, 7, 5, 1),
                     (64, 112, 3, 2), (16, 112, 3, 1)]:
    p = k // 2
    ho = (h + 2 * p - k) // s + 1
    x = nhwc(N, c, h, h)
    w = (torch.randn(k * k, c, device=dev) * 0.3).to(BF16)
    y = torch.empty(N, c, ho, ho, device=dev, dtype=BF16, memory_format=CL)
    ab = torch.rand(2 * c, device=dev)
    chunks = e.dw_stats_chunks(N, ho, ho, c, s)
    ws = torch.empty(2 * chunks * c, device=dev)
    base = timeit(lambda: e.dw_conv_fwd(x, w, y, k, s, p))
    lop = timeit(lambda: e.dw_conv_fwd(x, w, y, k, s, p, lop_kind=1,
                                       lop_act=3, lop_coef=ab))
    both = timeit(lambda: e.dw_conv_fwd(x, w, y, k, s, p, lop_kind=1,
                                        lop_act=3, lop_coef=ab,
                                        stats_ws=ws))
    st = timeit(lambda: e.dw_conv_fwd(x, w, y, k, s, p, stats_ws=ws))
    # unfused equivalent of +lop+stats: base + separate act pass + stats
    mean = torch.empty(c, device=dev)
    invstd = torch.empty(c, device=dev)
    ab2 = torch.empty(2 * c, device=dev)
    g1 = torch.rand(c, device=dev)
    b1 = torch.rand(c, device=dev)
    yact = torch.empty_like(y)
    actp = timeit(lambda: e.bn_act_fwd(y, ab, 3, yact, None))
    statp = timeit(lambda: e.bn_stats(y, mean, invstd, ab2, g1, b1,
                                      None, None, 0.1, 1e-5))
    print(f"dw c{c:<4d}h{h:<3d}k{k}s{s}: base {base:7.1f} +lop {lop:7.1f} "
          f"+stats {st:7.1f} +both {both:7.1f} | act-pass {actp:7.1f} "
          f"stats-pass {statp:7.1f} -> fused {both:7.1f} vs "
          f"unfused {base+actp+statp:7.1f}")

# ---- pw conv: stats epilogue cost on expand shapes ---------------------
for (m_hw, cin, cout) in [(112, 16, 64), (56, 24, 72), (28, 40, 120),
                          (14, 80, 480), (14, 112, 672), (7, 160, 960)]:
    x = nhwc(N, cin, m_hw, m_hw)
    w = (torch.randn(cout, cin, device=dev) * 0.2).to(BF16)
    z = torch.empty(N, cout, m_hw, m_hw, device=dev, dtype=BF16,
                    memory_format=CL)
    m = N * m_hw * m_hw
    mb = e.pw_stats_mblocks(m)
    ws = torch.empty(2 * mb * cout, device=dev)
    base = timeit(lambda: e.pw_conv(x, w, z))
    st = timeit(lambda: e.pw_conv(x, w, z, stats_ws=ws))
    mean = torch.empty(cout, device=dev)
    invstd = torch.empty(cout, device=dev)
    ab2 = torch.empty(2 * cout, device=dev)
    g1 = torch.rand(cout, device=dev)
    b1 = torch.rand(cout, device=dev)
    statp = timeit(lambda: e.bn_stats(z, mean, invstd, ab2, g1, b1,
                                      None, None, 0.1, 1e-5))
    ab = torch.rand(2 * cin, device=dev)
    lop = timeit(lambda: e.pw_conv(x, w, z, lop_kind=1, lop_act=3,
                                   lop_coef=ab))
    print(f"pw {cin:>4d}->{cout:<4d}@{m_hw:<3d}: base {base:7.1f} "
          f"+stats-epi {st:7.1f} +bnact-lop {lop:7.1f} | "
          f"stats-pass {statp:7.1f}")

# ---- dw wgrad lopx cost ------------------------------------------------
for (c, h, k, s) in [(120, 28, 5, 1), (480, 14, 3, 1), (960, 7, 5, 1),
                     (72, 56, 5, 2), (672, 14, 5, 2), (64, 112, 3, 2)]:
    p = k // 2
    ho = (h + 2 * p - k) // s + 1
    x = nhwc(N, c, h, h)
    dy = nhwc(N, c, ho, ho)
    dw = torch.empty(k * k, c, device=dev)
    ab = torch.rand(2 * c, device=dev)
    base = timeit(lambda: e.dw_conv_wgrad(x, dy, dw, k, s, p))
    lop = timeit(lambda: e.dw_conv_wgrad(x, dy, dw, k, s, p, lopx_kind=1,
                                         lopx_act=3, lopx_coef=ab))
    print(f"dwwg c{c:<4d}h{h:<3d}k{k}s{s}: base {base:7.1f} "
          f"+lopx {lop:7.1f}")
print("MB DONE")
