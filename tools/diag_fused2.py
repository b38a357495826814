#!/usr/bin/env python3
"""Replay the fused block cfg0 (no-SE, relu6, residual) step by step and
compare every intermediate with fp32 autograd."""

import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from yet_another_mobilenet_series_amd.ops.backend import ext

BF16 = torch.bfloat16
CL = torch.channels_last
dev = torch.device("cuda:0")
e = ext()
ACT = 2  # relu6


def rel(a, b):
    return ((a.float() - b.float()).norm() /
            b.float().norm().clamp(min=1e-9)).item()


torch.manual_seed(0)
n, cin, h = 4, 24, 28
ce, cd, cp = 72, 72, 24
k, s, p = 3, 1, 1

x = (torch.randn(n, cin, h, h, device=dev).bfloat16()
     .contiguous(memory_format=CL))
ew = ((torch.randn(ce, cin, device=dev) * 0.2).bfloat16().float()
      .requires_grad_(True))
eg = (torch.rand(ce, device=dev) + 0.5).requires_grad_(True)
ebb = (torch.randn(ce, device=dev) * 0.1).requires_grad_(True)
dww = ((torch.randn(cd, k * k, device=dev) * 0.3).bfloat16().float()
       .requires_grad_(True))
dg = (torch.rand(cd, device=dev) + 0.5).requires_grad_(True)
db = (torch.randn(cd, device=dev) * 0.1).requires_grad_(True)
pw = ((torch.randn(cp, cd, device=dev) * 0.2).bfloat16().float()
      .requires_grad_(True))
pg = (torch.rand(cp, device=dev) + 0.5).requires_grad_(True)
pb = (torch.randn(cp, device=dev) * 0.1).requires_grad_(True)
eps = 1e-5

# ---------------- fp32 reference with retained intermediates -----------
xr = x.float().clone().requires_grad_(True)
z1r = F.conv2d(xr, ew.view(ce, cin, 1, 1))
z1r.retain_grad()
y1r = F.relu6(F.batch_norm(z1r, None, None, eg, ebb, True, 0.0, eps))
y1r.retain_grad()
z2r = F.conv2d(y1r, dww.view(cd, 1, k, k), stride=s, padding=p, groups=cd)
z2r.retain_grad()
y2r = F.relu6(F.batch_norm(z2r, None, None, dg, db, True, 0.0, eps))
y2r.retain_grad()
z3r = F.conv2d(y2r, pw.view(cp, cd, 1, 1))
z3r.retain_grad()
ybr = F.batch_norm(z3r, None, None, pg, pb, True, 0.0, eps) + xr
torch.manual_seed(9)
dy = (torch.randn_like(ybr).bfloat16().contiguous(memory_format=CL))
ybr.backward(dy.float())

# ---------------- fused path, step by step ------------------------------
def bn_bufs(c):
    return (torch.empty(c, device=dev), torch.empty(c, device=dev),
            torch.empty(2 * c, device=dev))

ewp = ew.detach().view(ce, cin).bfloat16().contiguous()
z1 = torch.empty(n, ce, h, h, device=dev, dtype=BF16, memory_format=CL)
m1 = n * h * h
mb = e.pw_stats_mblocks(m1)
ws = torch.empty(2 * mb * ce, device=dev)
e.pw_conv(x, ewp, z1, stats_ws=ws)
mean1, invstd1, ab1 = bn_bufs(ce)
e.bn_finalize(ws, mb, mean1, invstd1, ab1, eg.detach(), ebb.detach(), None, None, 0.1, eps, m1)
print("z1 rel", rel(z1, z1r), " mean1 rel",
      rel(mean1, z1r.detach().mean(dim=(0, 2, 3))))

dwp = dww.detach().view(cd, k * k).t().bfloat16().contiguous()
z2 = torch.empty(n, cd, h, h, device=dev, dtype=BF16, memory_format=CL)
chunks = e.dw_stats_chunks(n, h, h, cd)
ws2 = torch.empty(2 * chunks * cd, device=dev)
e.dw_conv_fwd(z1, dwp, z2, k, s, p, lop_kind=1, lop_act=ACT, lop_coef=ab1,
              stats_ws=ws2)
mean2, invstd2, ab2 = bn_bufs(cd)
m2 = n * h * h
e.bn_finalize(ws2, chunks, mean2, invstd2, ab2, dg.detach(), db.detach(), None, None, 0.1,
              eps, m2)
print("z2 rel", rel(z2, z2r), " mean2 rel",
      rel(mean2, z2r.detach().mean(dim=(0, 2, 3))))

pwp = pw.detach().view(cp, cd).bfloat16().contiguous()
z3 = torch.empty(n, cp, h, h, device=dev, dtype=BF16, memory_format=CL)
mb3 = e.pw_stats_mblocks(m2)
ws3 = torch.empty(2 * mb3 * cp, device=dev)
e.pw_conv(z2, pwp, z3, lop_kind=1, lop_act=ACT, lop_coef=ab2, stats_ws=ws3)
mean3, invstd3, ab3 = bn_bufs(cp)
e.bn_finalize(ws3, mb3, mean3, invstd3, ab3, pg.detach(), pb.detach(), None, None, 0.1, eps, m2)
print("z3 rel", rel(z3, z3r))

yb = torch.empty_like(z3)
e.bn_act_fwd(z3, ab3, 0, yb, x)
print("yb rel", rel(yb, ybr.detach()))

# backward
dg3 = torch.empty(cp, device=dev)
db3 = torch.empty(cp, device=dev)
coef3 = torch.empty(4 * cp, device=dev)
e.bn_bwd_stats(dy, z3, mean3, invstd3, pg.detach(), pb.detach(), 0, True, dg3, db3, coef3)
print("dg3 rel", rel(dg3, pg.grad), " db3 rel", rel(db3, pb.grad)
      if pb.grad is not None else None)

pwt = pw.detach().view(cp, cd).t().bfloat16().contiguous()
d_in3 = torch.empty_like(z2)
e.pw_conv(dy, pwt, d_in3, lop_kind=2, lop_act=0, lop_coef=coef3, lop_z=z3)
print("d_y2 rel", rel(d_in3, y2r.grad))

dwp3 = torch.empty(cp, cd, device=dev)
e.pw_conv_wgrad(dy, z2, dwp3, lopa_kind=2, lopa_act=0, lopa_coef=coef3,
                lopa_z=z3, lopb_kind=1, lopb_act=ACT, lopb_coef=ab2)
print("dWp rel", rel(dwp3, pw.grad.view(cp, cd)))

dg2 = torch.empty(cd, device=dev)
db2_ = torch.empty(cd, device=dev)
coef2 = torch.empty(4 * cd, device=dev)
e.bn_bwd_stats(d_in3, z2, mean2, invstd2, dg.detach(), db.detach(), ACT, True, dg2, db2_,
               coef2)
print("dg2 rel", rel(dg2, dg.grad), " db2 rel", rel(db2_, db.grad))

d_y1 = torch.empty_like(z1)
e.dw_conv_dgrad(d_in3, dwp, d_y1, k, s, p, lop_kind=2, lop_act=ACT,
                lop_coef=coef2, lop_z=z2)
print("d_y1 rel", rel(d_y1, y1r.grad))

dwgt = torch.empty(k * k, cd, device=dev)
e.dw_conv_wgrad(z1, d_in3, dwgt, k, s, p, lopx_kind=1, lopx_act=ACT,
                lopx_coef=ab1, lopdy_kind=2, lopdy_act=ACT, lopdy_coef=coef2,
                lopdy_z=z2)
print("dWd rel", rel(dwgt.t(), dww.grad.view(cd, k * k)))

dg1 = torch.empty(ce, device=dev)
db1_ = torch.empty(ce, device=dev)
coef1 = torch.empty(4 * ce, device=dev)
e.bn_bwd_stats(d_y1, z1, mean1, invstd1, eg.detach(), ebb.detach(), ACT, True, dg1, db1_,
               coef1)
print("dg1 rel", rel(dg1, eg.grad), " db1 rel", rel(db1_, ebb.grad))

dwp1 = torch.empty(ce, cin, device=dev)
e.pw_conv_wgrad(d_y1, x, dwp1, lopa_kind=2, lopa_act=ACT, lopa_coef=coef1,
                lopa_z=z1)
print("dW1 rel", rel(dwp1, ew.grad.view(ce, cin)))
print("DONE")

# ---------------- same via the actual module + Function -----------------
print("== module path ==")
from yet_another_mobilenet_series_amd.models.mobilenet_base import InvertedResidual
from yet_another_mobilenet_series_amd.ops.backend import set_engine_mode
torch.manual_seed(0)
block = InvertedResidual(cin=24, cout=24, kernel_size=3, stride=1,
                         expand_ch=72, se_reduced_ch=0, act="relu6").to(dev)
block = block.to(memory_format=torch.channels_last)
block.train()
x2 = (torch.randn(4, 24, 28, 28, device=dev).bfloat16()
      .contiguous(memory_format=CL))
state0 = {kk: v.clone() for kk, v in block.state_dict().items()}
xh = x2.clone().requires_grad_(True)
y = block(xh)
torch.manual_seed(11)
dy2 = torch.randn_like(y.float()).bfloat16().contiguous(memory_format=CL)
y.backward(dy2)
g_fused = {kk: p.grad.float().clone() for kk, p in block.named_parameters()}

block.load_state_dict(state0)
for p_ in block.parameters():
    p_.grad = None
set_engine_mode("reference")
xr2 = x2.float().clone().requires_grad_(True)
yr2 = block(xr2)
yr2.backward(dy2.float())
set_engine_mode("hip")
for kk, p_ in block.named_parameters():
    print(f"  {kk:28s} rel {rel(g_fused[kk], p_.grad):8.4f} "
          f"ref_norm {p_.grad.norm().item():10.4f} "
          f"fused_norm {g_fused[kk].norm().item():10.4f}")
print("fwd rel", rel(y.detach(), yr2.detach()), " dx rel",
      rel(xh.grad, xr2.grad))
