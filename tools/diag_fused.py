#!/usr/bin/env python3
"""Isolate fused-path pieces (run on GPU box): epilogue stats vs the
standalone bn_stats kernel; lop-applied dgrad vs manual dz."""

import sys

import torch

sys.path.insert(0, ".")
from yet_another_mobilenet_series_amd.ops.backend import ext

BF16 = torch.bfloat16
CL = torch.channels_last
dev = torch.device("cuda:0")
e = ext()


def rel(a, b):
    return ((a.float() - b.float()).norm() /
            b.float().norm().clamp(min=1e-9)).item()


def nhwc(n, c, h, w, seed=0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(n, c, h, w, generator=g).to(dev).to(BF16)
            .contiguous(memory_format=CL))


print("== pw epilogue stats vs torch ==")
for (m_shape, k, cout) in [((4, 24, 28, 28), 24, 72), ((2, 16, 9, 9), 16, 24),
                           ((4, 160, 14, 14), 160, 960)]:
    n, cin, h, w = m_shape
    x = nhwc(n, cin, h, w, seed=1)
    wgt = (torch.randn(cout, k, device=dev) * 0.2).to(BF16)
    z = torch.empty(n, cout, h, w, device=dev, dtype=BF16, memory_format=CL)
    m = n * h * w
    mb = e.pw_stats_mblocks(m)
    ws = torch.full((2 * mb * cout,), float("nan"), device=dev)
    e.pw_conv(x, wgt, z, stats_ws=ws)
    mean = torch.empty(cout, device=dev)
    invstd = torch.empty_like(mean)
    ab = torch.empty(2 * cout, device=dev)
    g1 = torch.rand(cout, device=dev) + 0.5
    b1 = torch.randn(cout, device=dev) * 0.1
    e.bn_finalize(ws, mb, mean, invstd, ab, g1, b1, None, None, 0.1, 1e-5, m)
    zf = z.float()
    wm = zf.mean(dim=(0, 2, 3))
    wv = zf.var(dim=(0, 2, 3), unbiased=False)
    print(f"  shape {m_shape} cout={cout}: mean rel {rel(mean, wm):.5f} "
          f"invstd rel {rel(invstd, (wv+1e-5).rsqrt()):.5f}")

print("== dw epilogue stats vs torch ==")
import torch.nn.functional as F
for (n, c, h, k, s) in [(4, 72, 28, 3, 1), (4, 120, 28, 5, 1),
                        (2, 16, 12, 3, 2)]:
    x = nhwc(n, c, h, h, seed=2)
    wgt = (torch.randn(k * k, c, device=dev) * 0.3).to(BF16)
    p = k // 2
    ho = (h + 2 * p - k) // s + 1
    z = torch.empty(n, c, ho, ho, device=dev, dtype=BF16, memory_format=CL)
    chunks = e.dw_stats_chunks(n, ho, ho, c)
    ws = torch.full((2 * chunks * c,), float("nan"), device=dev)
    e.dw_conv_fwd(x, wgt, z, k, s, p, stats_ws=ws)
    m = n * ho * ho
    mean = torch.empty(c, device=dev)
    invstd = torch.empty_like(mean)
    ab = torch.empty(2 * c, device=dev)
    g1 = torch.rand(c, device=dev) + 0.5
    b1 = torch.randn(c, device=dev) * 0.1
    e.bn_finalize(ws, chunks, mean, invstd, ab, g1, b1, None, None, 0.1,
                  1e-5, m)
    want = F.conv2d(x.float(), wgt.float().t().reshape(c, 1, k, k),
                    stride=s, padding=p, groups=c)
    wm = want.mean(dim=(0, 2, 3))
    wv = want.var(dim=(0, 2, 3), unbiased=False)
    print(f"  c={c} k={k} s={s}: mean rel {rel(mean, wm):.5f} "
          f"invstd rel {rel(invstd, (wv+1e-5).rsqrt()):.5f} "
          f"fwd rel {rel(z, want):.5f}")

print("== lop bnact on pw A-stream vs manual ==")
n, cin, h = 4, 72, 28
cout = 24
z1 = nhwc(n, cin, h, h, seed=3)
ab = torch.cat([torch.rand(cin, device=dev) + 0.5,
                torch.randn(cin, device=dev) * 0.2]).contiguous()
wgt = (torch.randn(cout, cin, device=dev) * 0.2).to(BF16)
y1 = torch.empty(n, cout, h, h, device=dev, dtype=BF16, memory_format=CL)
e.pw_conv(z1, wgt, y1, lop_kind=1, lop_act=2, lop_coef=ab)
manual = F.relu6(z1.float() * ab[:cin].view(1, -1, 1, 1)
                 + ab[cin:].view(1, -1, 1, 1)).to(BF16)
y2 = torch.empty_like(y1)
e.pw_conv(manual.contiguous(memory_format=CL), wgt, y2)
print(f"  pw bnact lop rel {rel(y1, y2):.5f}")

print("== lop bnbwd on dw dgrad vs manual dz ==")
for (k, s) in [(3, 1), (5, 1), (3, 2)]:
    c = 72
    p = k // 2
    hi_ = 28
    ho = (hi_ + 2 * p - k) // s + 1
    dy = nhwc(n, c, ho, ho, seed=4)
    z2 = nhwc(n, c, ho, ho, seed=5)
    coef = torch.cat([torch.rand(c, device=dev) + 0.5,
                      torch.randn(c, device=dev) * 0.2,
                      torch.randn(c, device=dev) * 0.05,
                      torch.randn(c, device=dev) * 0.05]).contiguous()
    wgt = (torch.randn(k * k, c, device=dev) * 0.3).to(BF16)
    dx1 = torch.empty(n, c, hi_, hi_, device=dev, dtype=BF16,
                      memory_format=CL)
    e.dw_conv_dgrad(dy, wgt, dx1, k, s, p, lop_kind=2, lop_act=3,
                    lop_coef=coef, lop_z=z2)
    # manual dz
    a = coef[:c].view(1, -1, 1, 1)
    b = coef[c:2*c].view(1, -1, 1, 1)
    k0 = coef[2*c:3*c].view(1, -1, 1, 1)
    k2 = coef[3*c:].view(1, -1, 1, 1)
    zf = z2.float()
    v0 = zf * a + b
    dact = torch.where(v0 <= -3, torch.zeros_like(v0),
                       torch.where(v0 >= 3, torch.ones_like(v0),
                                   (2 * v0 + 3) / 6))
    dz = (a * (dy.float() * dact) + k0 - k2 * zf).to(BF16)
    dx2 = torch.empty_like(dx1)
    e.dw_conv_dgrad(dz.contiguous(memory_format=CL), wgt, dx2, k, s, p)
    print(f"  k{k}s{s}: dgrad bnbwd lop rel {rel(dx1, dx2):.5f}")

print("== atb lop bnbwd vs manual ==")
m = 3136
n1, n2 = 72, 24
dyf = torch.randn(m, n1, device=dev).to(BF16)
z1f = torch.randn(m, n1, device=dev).to(BF16)
bmat = torch.randn(m, n2, device=dev).to(BF16)
coef = torch.cat([torch.rand(n1, device=dev) + 0.5,
                  torch.randn(n1, device=dev) * 0.2,
                  torch.randn(n1, device=dev) * 0.05,
                  torch.randn(n1, device=dev) * 0.05]).contiguous()
dw1 = torch.empty(n1, n2, device=dev, dtype=torch.float32)
e.pw_conv_wgrad(dyf.view(m, n1, 1, 1), bmat.view(m, n2, 1, 1), dw1,
                lopa_kind=2, lopa_act=2, lopa_coef=coef, lopa_z=z1f)
a = coef[:n1]; b = coef[n1:2*n1]; k0 = coef[2*n1:3*n1]; k2 = coef[3*n1:]
v0 = z1f.float() * a + b
dact = ((v0 > 0) & (v0 < 6)).float()
dz = (a * (dyf.float() * dact) + k0 - k2 * z1f.float()).to(BF16)
dw2 = torch.empty_like(dw1)
e.pw_conv_wgrad(dz.view(m, n1, 1, 1), bmat.view(m, n2, 1, 1), dw2)
print(f"  atb bnbwd lopA rel {rel(dw1, dw2):.5f}")
print("DIAG DONE")
