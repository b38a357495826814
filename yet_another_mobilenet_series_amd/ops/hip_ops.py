"""GPU op path: custom autograd.Functions over the CDNA4 HIP kernels.

Layout contract with the extension (`ops/csrc`): activations are NCHW
logical / ``torch.channels_last`` memory (== NHWC rows, C fastest), compute
dtype bf16, accumulation and statistics fp32.  Weight tensors are re-packed
per call into the kernel-friendly layouts:

- pointwise 1x1 [Cout, Cin, 1, 1]   -> GEMM B operand [Cout, Cin] bf16
  (K-major: the MFMA GEMM reads both operands along the reduction axis,
  so the natural [Cout, Cin] order needs no transpose staging)
- depthwise  [C, 1, k, k]           -> [k*k, C] bf16 (C fastest)
- stem dense [Cout, 3, k, k]        -> [Cout, k*k*3 padded to /8] bf16,
  consumed by im2row patches through the same MFMA GEMM

Gradients w.r.t. parameters are produced in fp32 (master-weight dtype);
gradients w.r.t. activations in bf16.  All reductions (BN stats, wgrad)
are two-stage slab reductions — deterministic, no atomics.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from .backend import ext

BF16 = torch.bfloat16
CL = torch.channels_last

# activation codes shared with csrc/common.h
ACT_IDENTITY = 0
ACT_RELU = 1
ACT_RELU6 = 2
ACT_HSWISH = 3
ACT_HSIGMOID = 4



def _packed(param: torch.Tensor, key: str, fn, spec: Optional[str] = None):
    """Cache a packed/cast weight on the tensor, keyed by data version.

    ``spec`` records how the pack derives from the param so the fused
    optimizers can refresh EVERY pack with one multi-tensor kernel per
    step (:func:`refresh_packs`) instead of ~95 per-tensor torch ops:

    - ``"cast"``: same element order, bf16 cast (pw/lin/SE packs — the
      param is [A, B, 1, 1] or [A, B] and the pack its flat bf16 copy)
    - ``"t"``: 2D transpose-cast — the pack is [A, B] bf16 from the
      param's flat [B, A] fp32 order (dw [k*k,C], dgrad W^T packs)
    - ``None``: custom (stem pad) — dropped on refresh, rebuilt lazily.

    Packs re-run whenever the param version changes, and ALWAYS during
    CUDA-graph capture (so replay re-reads the updated fp32 master
    through the baked cast kernels)."""
    if param.is_cuda and torch.cuda.is_current_stream_capturing():
        return fn(param)
    cache = getattr(param, "_yams_pack", None)
    if cache is None:
        cache = {}
        try:
            param._yams_pack = cache
        except AttributeError:
            return fn(param)
    ver = param._version
    hit = cache.get(key)
    if hit is not None and hit[0] == ver:
        return hit[1]
    t = fn(param)
    cache[key] = (ver, t, spec)
    return t


def refresh_packs(params) -> None:
    """Refresh every cached bf16 pack of ``params`` in place (one fused
    kernel) and re-stamp the entries at the params' CURRENT versions.

    Called by the fused optimizers right after they bump versions: without
    it the next forward would rebuild ~95 packs through per-tensor torch
    ops (measured ~2 ms/step on V3-Large)."""
    srcs, dsts, adims = [], [], []
    for p in params:
        cache = getattr(p, "_yams_pack", None)
        if not cache:
            continue
        ver = p._version
        drop = []
        for key, entry in cache.items():
            _, t, spec = entry
            if spec == "cast":
                srcs.append(p)
                dsts.append(t)
                adims.append(0)
            elif spec == "t":
                srcs.append(p)
                dsts.append(t)
                adims.append(t.shape[0])
            else:
                drop.append(key)
                continue
            cache[key] = (ver, t, spec)
        for key in drop:
            del cache[key]
    if srcs:
        ext().fused_repack(srcs, dsts, adims)

def _nhwc(x: torch.Tensor) -> torch.Tensor:
    """bf16 + channels_last view of a 4D activation tensor."""
    if x.dim() == 2:
        x = x.view(x.shape[0], x.shape[1], 1, 1)
    return x.to(BF16).contiguous(memory_format=CL)


def _empty_like_nhwc(x: torch.Tensor, c: Optional[int] = None,
                     h: Optional[int] = None, w: Optional[int] = None) -> torch.Tensor:
    n = x.shape[0]
    c = c if c is not None else x.shape[1]
    h = h if h is not None else x.shape[2]
    w = w if w is not None else x.shape[3]
    return torch.empty((n, c, h, w), dtype=BF16, device=x.device,
                       memory_format=CL)


def _conv_out_hw(h: int, w: int, k: int, s: int, p: int) -> Tuple[int, int]:
    return (h + 2 * p - k) // s + 1, (w + 2 * p - k) // s + 1


def _conv_forward(x: torch.Tensor, w_packed: torch.Tensor, kind: str,
                  k: int, s: int, p: int) -> torch.Tensor:
    """Raw conv z = conv(x) with packed bf16 weights; returns bf16 NHWC."""
    e = ext()
    n, cin, h, wdt = x.shape
    if kind == "pw":
        cout = w_packed.shape[0]
        z = _empty_like_nhwc(x, c=cout)
        e.pw_conv(x, w_packed, z)
    elif kind == "dw":
        ho, wo = _conv_out_hw(h, wdt, k, s, p)
        z = _empty_like_nhwc(x, h=ho, w=wo)
        e.dw_conv_fwd(x, w_packed, z, k, s, p)
    else:
        # the stem never reaches here: it runs im2row + pw_conv in
        # _ConvBNActFn.forward directly (needs the patches for wgrad)
        raise ValueError(kind)
    return z


class _ConvBNActFn(torch.autograd.Function):
    """Fused conv → BN(train|eval) → act [+ residual] on the HIP engine."""

    @staticmethod
    def forward(ctx, x, weight, bn_w, bn_b, residual,
                running_mean, running_var, kind, k, s, p,
                act, training, momentum, eps):
        e = ext()
        x = _nhwc(x)
        cout = weight.shape[0]
        cin = x.shape[1]
        # pack weights into kernel layout (bf16)
        if kind == "pw":
            w_packed = _packed(weight, "pw", lambda w: w.view(
                cout, cin).to(BF16).contiguous(), spec="cast")
        elif kind == "dw":
            w_packed = _packed(weight, "dw", lambda w: w.view(
                cout, k * k).t().to(BF16).contiguous(), spec="t")
        elif kind == "stem":
            # [Cout, 3, k, k] -> [Cout, k*k*3 padded to /8] matching im2row
            kp = (k * k * cin + 7) // 8 * 8
            def _pack_stem(w):
                flat = w.permute(0, 2, 3, 1).reshape(cout, k * k * cin)
                out = torch.zeros(cout, kp, dtype=BF16, device=w.device)
                out[:, :k * k * cin] = flat.to(BF16)
                return out
            w_packed = _packed(weight, "stem", _pack_stem, spec=None)
        else:
            raise ValueError(kind)

        if kind == "stem":
            n, _, h, wdt = x.shape
            ho, wo = _conv_out_hw(h, wdt, k, s, p)
            m = n * ho * wo
            patches = torch.empty(m, w_packed.shape[1], dtype=BF16,
                                  device=x.device)
            e.im2row(x, patches, k, s, p)
            z = _empty_like_nhwc(x, c=cout, h=ho, w=wo)
            e.pw_conv(patches.view(m, -1, 1, 1), w_packed,
                      z.permute(0, 2, 3, 1).reshape(m, cout)
                      .view(m, cout, 1, 1))
            ctx.stem_patches = patches
        else:
            z = _conv_forward(x, w_packed, kind, k, s, p)

        mean = torch.empty(cout, dtype=torch.float32, device=x.device)
        invstd = torch.empty_like(mean)
        ab = torch.empty(2 * cout, dtype=torch.float32, device=x.device)
        if training:
            e.bn_stats(z, mean, invstd, ab, bn_w, bn_b,
                       running_mean, running_var, momentum, eps)
        else:
            mean.copy_(running_mean)
            invstd.copy_((running_var.float() + eps).rsqrt())
            a = bn_w.float() * invstd
            ab[:cout] = a
            ab[cout:] = bn_b.float() - mean * a

        y = torch.empty_like(z)
        res = residual
        if res is not None:
            res = _nhwc(res)
        e.bn_act_fwd(z, ab, act, y, res)

        ctx.save_for_backward(x, z, mean, invstd, bn_w, bn_b, weight)
        ctx.kind, ctx.k, ctx.s, ctx.p = kind, k, s, p
        ctx.act, ctx.training, ctx.eps = act, training, eps
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        e = ext()
        x, z, mean, invstd, bn_w, bn_b, weight = ctx.saved_tensors
        kind, k, s, p = ctx.kind, ctx.k, ctx.s, ctx.p
        dy = dy.to(BF16).contiguous(memory_format=CL)
        cout = weight.shape[0]
        cin = x.shape[1]

        dgamma = torch.empty(cout, dtype=torch.float32, device=x.device)
        dbeta = torch.empty_like(dgamma)
        dz = torch.empty_like(z)
        e.bn_act_bwd(dy, z, mean, invstd, bn_w, bn_b, ctx.act,
                     bool(ctx.training), dgamma, dbeta, dz)

        dx = None
        if ctx.needs_input_grad[0]:
            if kind == "pw":
                w_t = _packed(weight, "pw_t", lambda w: w.view(
                    cout, cin).t().to(BF16).contiguous(), spec="t")
                dx = _empty_like_nhwc(x)
                e.pw_conv(dz, w_t, dx)
            elif kind == "dw":
                w_packed = _packed(weight, "dw", lambda w: w.view(
                    cout, k * k).t().to(BF16).contiguous(), spec="t")
                dx = _empty_like_nhwc(x)
                e.dw_conv_dgrad(dz, w_packed, dx, k, s, p)
            else:
                raise NotImplementedError(
                    "stem conv input gradient is not implemented "
                    "(the stem is the first layer; its input needs no grad)")

        dweight = None
        if ctx.needs_input_grad[1]:
            if kind == "pw":
                dwp = torch.empty(cout, cin, dtype=torch.float32,
                                  device=x.device)
                e.pw_conv_wgrad(dz, x, dwp)
                dweight = dwp.reshape(cout, cin, 1, 1)
            elif kind == "dw":
                dwp = torch.empty(cout, k * k, dtype=torch.float32,
                                  device=x.device)
                e.dw_conv_wgrad(x, dz, dwp, k, s, p)
                dweight = dwp.reshape(cout, 1, k, k)
            else:  # stem: wgrad on the split-M GEMM over saved patches
                patches = ctx.stem_patches
                m = patches.shape[0]
                kp = patches.shape[1]
                dwp = torch.empty(cout, kp, dtype=torch.float32,
                                  device=x.device)
                e.pw_conv_wgrad(
                    dz.permute(0, 2, 3, 1).reshape(m, cout)
                    .view(m, cout, 1, 1),
                    patches.view(m, kp, 1, 1), dwp)
                dweight = (dwp[:, :k * k * cin]
                           .view(cout, k, k, cin)
                           .permute(0, 3, 1, 2).contiguous())

        dres = dy if ctx.has_residual else None
        return (dx, dweight, dgamma, dbeta, dres) + (None,) * 10


def conv_bn_act(x, weight, bn_weight, bn_bias, running_mean, running_var, *,
                stride, padding, groups, act, training, momentum, eps,
                residual=None, conv_bias=None):
    cout, cin_g, kh, kw = weight.shape
    if bn_weight is None or conv_bias is not None:
        raise NotImplementedError(
            "HIP conv_bn_act requires BN and no conv bias; the zoo's only "
            "bias convs are post-pool 1x1s, expressed as linear()")
    if kh != kw:
        raise ValueError("square kernels only")
    if groups == 1 and kh == 1:
        kind = "pw"
    elif groups == cout and cin_g == 1:
        kind = "dw"
    elif groups == 1 and cin_g == 3 and kh == 3:
        kind = "stem"
    else:
        raise NotImplementedError(
            f"conv shape not in the zoo: k={kh} groups={groups} cin={cin_g}")
    return _ConvBNActFn.apply(
        x, weight, bn_weight, bn_bias, residual, running_mean, running_var,
        kind, kh, stride, padding, act, training, momentum, eps)


class _LinearActFn(torch.autograd.Function):
    """y = act(x @ W^T + b) via the pointwise MFMA GEMM (M = batch)."""

    @staticmethod
    def forward(ctx, x, weight, bias, act):
        e = ext()
        x2 = x.to(BF16).contiguous()
        if x2.dim() != 2:
            x2 = x2.view(x2.shape[0], -1)
        nout, nin = weight.shape
        wt = _packed(weight, "lin", lambda w: w.to(BF16).contiguous(), spec="cast")
        z = torch.empty(x2.shape[0], nout, dtype=BF16, device=x.device)
        e.pw_conv(x2.view(x2.shape[0], nin, 1, 1), wt,
                  z.view(z.shape[0], nout, 1, 1))
        y = torch.empty_like(z)
        e.bias_act_fwd(z, bias.float() if bias is not None else None, act, y)
        ctx.save_for_backward(x2, z, weight,
                              bias if bias is not None else torch.empty(0))
        ctx.act = act
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        e = ext()
        x2, z, weight, bias = ctx.saved_tensors
        dy = dy.to(BF16).contiguous()
        nout, nin = weight.shape
        dz = torch.empty_like(dy)
        dbias = (torch.empty(nout, dtype=torch.float32, device=dy.device)
                 if ctx.has_bias else None)
        e.bias_act_bwd(dy, z, bias.float() if ctx.has_bias else None,
                       ctx.act, dz, dbias)
        dx = None
        if ctx.needs_input_grad[0]:
            w = _packed(weight, "lin_t",
                        lambda wt_: wt_.t().to(BF16).contiguous(), spec="t")
            dx = torch.empty_like(x2)
            e.pw_conv(dz.view(-1, nout, 1, 1), w, dx.view(-1, nin, 1, 1))
        dweight = None
        if ctx.needs_input_grad[1]:
            dwp = torch.empty(nout, nin, dtype=torch.float32, device=dy.device)
            e.pw_conv_wgrad(dz.view(-1, nout, 1, 1), x2.view(-1, nin, 1, 1), dwp)
            dweight = dwp
        return dx, dweight, dbias, None


def linear(x, weight, bias=None, act=ACT_IDENTITY):
    return _LinearActFn.apply(x, weight, bias, act)


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        e = ext()
        x = _nhwc(x)
        out = torch.empty(x.shape[0], x.shape[1], dtype=BF16, device=x.device)
        e.avgpool_fwd(x, out)
        ctx.in_shape = x.shape
        ctx.dev = x.device
        return out

    @staticmethod
    def backward(ctx, dy):
        e = ext()
        n, c, h, w = ctx.in_shape
        dy = dy.to(BF16).contiguous()
        dx = torch.empty((n, c, h, w), dtype=BF16, device=ctx.dev,
                         memory_format=CL)
        e.avgpool_bwd(dy, dx)
        return dx


def global_avg_pool(x):
    return _GlobalAvgPoolFn.apply(x)


def se_pack(w1, w2):
    """bf16 kernel packs, natural layout: w1 [R,C], w2 [C,R] (backward)."""
    r, c = w1.shape[0], w1.shape[1]
    w1m = _packed(w1, "se", lambda w: w.view(r, c).to(BF16).contiguous(),
                  spec="cast")
    w2m = _packed(w2, "se2", lambda w: w.view(c, r).to(BF16).contiguous(),
                  spec="cast")
    return w1m, w2m


def se_pack_t(w1, w2):
    """Transposed packs w1t [C,R], w2t [R,C] — the forward kernel's FC
    loops read these coalesced across lanes (se.hip)."""
    r, c = w1.shape[0], w1.shape[1]
    w1t = _packed(w1, "se_t",
                  lambda w: w.view(r, c).t().to(BF16).contiguous(),
                  spec="t")
    w2t = _packed(w2, "se2_t",
                  lambda w: w.view(c, r).t().to(BF16).contiguous(),
                  spec="t")
    return w1t, w2t


def se_forward(e, x, w1, b1, w2, b2, lop=None, ymat=None):
    """One-kernel fused SE forward (K7, se.hip).  Returns
    (y, pooled, ha, sz) with fp32 saved intermediates.

    ``lop=(1, act, ab)`` applies the producing BN+act while loading x and
    streams the activated input to ``ymat``.
    """
    n, c = x.shape[0], x.shape[1]
    r = w1.shape[0]
    w1t, w2t = se_pack_t(w1, w2)
    y = torch.empty_like(x)
    pooled = torch.empty(n, c, dtype=torch.float32, device=x.device)
    ha = torch.empty(n, r, dtype=torch.float32, device=x.device)
    sz = torch.empty(n, c, dtype=torch.float32, device=x.device)
    if lop is None:
        e.se_fwd(x, w1t, b1.float(), w2t, b2.float(), y, pooled, ha, sz)
    else:
        e.se_fwd(x, w1t, b1.float(), w2t, b2.float(), y, pooled, ha, sz,
                 lop_kind=lop[0], lop_act=lop[1], lop_coef=lop[2],
                 ymat=ymat)
    return y, pooled, ha, sz


def se_backward(e, dy, x, w1, w2, pooled, ha, sz, bn=None):
    """Fused SE backward: one per-image kernel (dx including the pool
    gradient) + one deterministic cross-batch wgrad kernel.  All
    SE-internal gradients stay fp32.  ``bn=(z2, ab2, act, stats_ws)``
    additionally emits the dw BN's backward partials while writing dx.
    Returns (dx, dw1, db1, dw2, db2)."""
    n, c = pooled.shape
    r = ha.shape[1]
    w1m, w2m = se_pack(w1, w2)
    dx = torch.empty_like(x)
    dsz = torch.empty(n, c, dtype=torch.float32, device=x.device)
    dhz = torch.empty(n, r, dtype=torch.float32, device=x.device)
    if bn is not None:
        e.se_bwd(dy, x, w1m, w2m, ha, sz, dx, dsz, dhz, bn_z=bn[0],
                 bn_ab=bn[1], bn_act=bn[2], bn_stats_ws=bn[3])
    else:
        e.se_bwd(dy, x, w1m, w2m, ha, sz, dx, dsz, dhz)
    dw1 = torch.empty(r, c, dtype=torch.float32, device=x.device)
    dw2 = torch.empty(c, r, dtype=torch.float32, device=x.device)
    db1 = torch.empty(r, dtype=torch.float32, device=x.device)
    db2 = torch.empty(c, dtype=torch.float32, device=x.device)
    e.se_wgrad(dsz, ha, dhz, pooled, dw1, dw2, db1, db2)
    return dx, dw1, db1, dw2, db2


class _SqueezeExciteFn(torch.autograd.Function):
    """SE block (K7): ONE fused kernel per direction (se.hip) — pool,
    FC1+ReLU, FC2+h-sigmoid and the channel scale in a single launch,
    fp32 internal precision (VERDICT round-1 item 7).
    """

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        e = ext()
        x = _nhwc(x)
        y, pooled, ha, sz = se_forward(e, x, w1, b1, w2, b2)
        ctx.save_for_backward(x, pooled, ha, sz, w1, w2)
        return y

    @staticmethod
    def backward(ctx, dy):
        e = ext()
        x, pooled, ha, sz, w1, w2 = ctx.saved_tensors
        dy = dy.to(BF16).contiguous(memory_format=CL)
        dx, dw1, db1, dw2, db2 = se_backward(e, dy, x, w1, w2, pooled,
                                             ha, sz)
        return dx, dw1.reshape_as(w1), db1, dw2.reshape_as(w2), db2


def squeeze_excite(x, w1, b1, w2, b2):
    return _SqueezeExciteFn.apply(x, w1, b1, w2, b2)


_dropout_counter = [0]


def _dropout_seed() -> int:
    """Deterministic per-call seed: process seed x call counter.  Each
    dropout call gets a fresh mask; a fixed torch seed reproduces the
    whole sequence (determinism test)."""
    _dropout_counter[0] += 1
    return (torch.initial_seed() * 1000003 + _dropout_counter[0]) \
        & 0x7FFFFFFFFFFFFFFF


class _DropoutFn(torch.autograd.Function):
    """K14: y = x*mask/(1-p); the mask is recomputed from the saved seed in
    backward (counter-hash RNG in misc.hip) — no mask tensor stored."""

    @staticmethod
    def forward(ctx, x, p):
        e = ext()
        x = x.to(BF16)
        x = (x.contiguous() if x.dim() == 2
             else x.contiguous(memory_format=CL))
        seed = _dropout_seed()
        y = torch.empty_like(x)
        e.dropout(x, y, seed, p, forward=True)
        ctx.seed = seed
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        e = ext()
        dy = dy.to(BF16)
        dy = (dy.contiguous() if dy.dim() == 2
              else dy.contiguous(memory_format=CL))
        dx = torch.empty_like(dy)
        e.dropout(dy, dx, ctx.seed, ctx.p, forward=False)
        return dx, None


def dropout(x, p):
    return _DropoutFn.apply(x, p)


class _LabelSmoothCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, smoothing):
        e = ext()
        logits = logits.to(BF16).contiguous()
        loss = torch.empty((), dtype=torch.float32, device=logits.device)
        dlogits = torch.empty(logits.shape, dtype=torch.float32,
                              device=logits.device)
        e.label_smooth_ce(logits, target.to(torch.int64).contiguous(),
                          float(smoothing), loss, dlogits)
        ctx.save_for_backward(dlogits)
        ctx.logits_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, dloss):
        (dlogits,) = ctx.saved_tensors
        return (dlogits * dloss).to(ctx.logits_dtype), None, None


def label_smooth_ce(logits, target, smoothing):
    return _LabelSmoothCEFn.apply(logits, target, smoothing)
