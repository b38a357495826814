#!/usr/bin/env python3
"""Layer-local backward parity diagnostic (run on GPU box).

For each ConvBNAct layer of MobileNetV2: feed the SAME bf16 input and
output-grad to (a) the HIP fused Function and (b) a plain fp32 torch
composition computed from the identical bf16-rounded values.  Layer-local
agreement ⇒ model-level drift is accumulated bf16 noise, not a kernel bug.
"""

import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")

from yet_another_mobilenet_series_amd import models
from yet_another_mobilenet_series_amd.models.mobilenet_base import ConvBNAct
from yet_another_mobilenet_series_amd.ops import functional as OF

ACTS = {0: lambda t: t, 1: F.relu, 2: F.relu6, 3: F.hardswish}


def rel(a, b):
    return ((a - b).norm() / b.norm().clamp(min=1e-12)).item()


def cos(a, b):
    return torch.nn.functional.cosine_similarity(
        a.flatten(), b.flatten(), dim=0).item()


def main():
    dev = torch.device("cuda:0")
    torch.manual_seed(0)
    model = models.build_model("mobilenet_v2", dropout=0.0).to(dev)
    model = model.to(memory_format=torch.channels_last)
    model.train()

    # capture each ConvBNAct's input via hooks on a HIP forward
    layers = [m for m in model.modules() if isinstance(m, ConvBNAct)]
    inputs = {}

    def mk_hook(mod):
        def hook(m, args, kwargs=None):
            inputs[mod] = args[0].detach().clone()
        return hook

    hs = [m.register_forward_pre_hook(mk_hook(m)) for m in layers]
    x = torch.randn(4, 3, 224, 224, device=dev).bfloat16()
    x = x.contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        model(x)
    for h in hs:
        h.remove()

    print(f"{'layer':38s} {'kind':5s} {'dx_rel':>8s} {'dw_rel':>8s} "
          f"{'dg_rel':>8s} {'dg_cos':>7s} {'db_rel':>8s} {'y_rel':>8s}")
    for i, m in enumerate(layers):
        xin = inputs[m].bfloat16().contiguous(
            memory_format=torch.channels_last)
        kind = ("stem" if m.conv.weight.shape[1] == 3 and
                m.conv.weight.shape[2] == 3 else
                "dw" if m.groups > 1 else "pw")
        # --- HIP fused path
        xh = xin.clone().requires_grad_(kind != "stem")
        w = m.conv.weight.detach().clone().requires_grad_(True)
        g = m.bn.weight.detach().clone().requires_grad_(True)
        b = m.bn.bias.detach().clone().requires_grad_(True)
        rm, rv = m.bn.running_mean.clone(), m.bn.running_var.clone()
        y = OF.conv_bn_act(xh, w, g, b, rm, rv, stride=m.stride,
                           padding=m.padding, groups=m.groups, act=m.act,
                           training=True, momentum=0.1, eps=m.bn.eps)
        torch.manual_seed(100 + i)
        dy = torch.randn_like(y.float()).bfloat16().contiguous(
            memory_format=torch.channels_last)
        y.backward(dy)
        # --- fp32 reference from the same bf16 values
        xr = xin.float().clone().requires_grad_(kind != "stem")
        wr = w.detach().float().clone().requires_grad_(True)
        gr = g.detach().clone().requires_grad_(True)
        br = b.detach().clone().requires_grad_(True)
        z = F.conv2d(xr, wr, stride=m.stride, padding=m.padding,
                     groups=m.groups)
        zn = F.batch_norm(z, rm.clone(), rv.clone(), gr, br, True, 0.1,
                          m.bn.eps)
        yr = ACTS[m.act](zn)
        yr.backward(dy.float())
        name = f"L{i:02d} c{m.conv.weight.shape[0]}x{m.conv.weight.shape[1]}k{m.conv.weight.shape[2]}s{m.stride}"
        dx_rel = (rel(xh.grad.float(), xr.grad.float())
                  if kind != "stem" else float("nan"))
        print(f"{name:38s} {kind:5s} {dx_rel:8.4f} "
              f"{rel(w.grad.float(), wr.grad):8.4f} "
              f"{rel(g.grad, gr.grad):8.4f} {cos(g.grad, gr.grad):7.3f} "
              f"{rel(b.grad, br.grad):8.4f} "
              f"{rel(y.detach().float(), yr.detach().float()):8.4f}")


if __name__ == "__main__":
    main()
