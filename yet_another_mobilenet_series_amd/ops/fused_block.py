"""Fused inverted-residual block (SURVEY.md §7 stage 4: block-level fusion).

One autograd Function runs the whole block {pw expand -> BN+act -> dw ->
BN+act -> [SE] -> pw project -> BN [+residual]}.  The fusion set below is
the one that MEASURED fastest on MI355X (each candidate was A/B'd per
kernel and per step; losers are documented in profiles/README.md):

- BN statistics always ride the conv epilogues (per-block partial slabs
  combined by ``bn_finalize``) — no separate stats pass over any z;
- the no-SE projection conv applies the dw-BN+act as a LOAD-OP, so its
  activated input y2 is never written; y1 (expand activation) IS
  materialized — the depthwise window-load lop costs more registers
  (occupancy) than this memory-bound elementwise pass costs time;
- backward materializes dz per BN via the fused
  partial→finalize→apply chain (``bn_act_bwd``); computing dz as a
  BNBWD load-op inside the dgrad/wgrad GEMMs, or emitting BN-backward
  partials from dgrad epilogues (``YAMS_BWD_EPI``), both measured
  net-negative.  The SE pool-grad add IS fused into the BN partial pass
  (``bn_act_bwd_addpool``) — one less full pass over d_y2.
"""

from __future__ import annotations

from typing import Optional

import os

import torch

from .backend import ext
from .hip_ops import (ACT_IDENTITY, BF16, CL, _empty_like_nhwc, _nhwc,
                      _packed)

F32 = torch.float32


def _bn_bufs(c: int, dev) -> tuple:
    mean = torch.empty(c, dtype=F32, device=dev)
    invstd = torch.empty(c, dtype=F32, device=dev)
    ab = torch.empty(2 * c, dtype=F32, device=dev)
    return mean, invstd, ab


def _eval_ab(mean, invstd, gamma, beta, ab) -> None:
    a = gamma.float() * invstd
    ab[:gamma.numel()] = a
    ab[gamma.numel():] = beta.float() - mean * a


# backward-fusion knobs (A/B measured on MI355X, V3-Large B256 bf16):
#   YAMS_BWD_EPI=1   emit BN-backward partials from the conv-dgrad
#                    epilogues (pw EPI==2 / dw FLIP EPI==2) instead of the
#                    separate bn_bwd_partial pass.  Measured NET-NEGATIVE
#                    (+0.9 ms/step): the scalar z loads + per-frag
#                    reductions slow the stream-optimal producer kernels
#                    more than the removed partial pass costs.  Default off.
_EPI_ON = os.environ.get("YAMS_BWD_EPI", "0") != "0"
#   YAMS_WGRAD_STREAM=0  disable the side-stream weight-grad overlap
#                    (default on: the per-block wgrad chain — two atb
#                    GEMMs + dw wgrad + slab combines, ~0.25 ms/block of
#                    latency-bound work — runs on a second HIP stream
#                    under the bandwidth-bound dgrad/BN chain, joined at
#                    the end of each block's backward)
_WGRAD_STREAM_ON = os.environ.get("YAMS_WGRAD_STREAM", "1") != "0"
#   YAMS_BWD_LOPDZ=1 never-materialize dz for the pw BNs: reduce-only
#                    backward + LOP_BNBWD dz recompute in both GEMMs.
#                    Measured NET-NEGATIVE again in round 2 (+1.6 ms
#                    same-box A/B) even with the wgrads on the side
#                    stream: the dual-stream staging slows the MFMA
#                    GEMMs more than the removed apply pass costs.
#                    Default off; kept as a measured experiment.
_LOPDZ_ON = os.environ.get("YAMS_BWD_LOPDZ", "0") != "0"
#   YAMS_DW_LOP=1    never materialize y1: the dw forward AND the dw
#                    wgrad read z1 with the expand BN+act as a load-op.
#                    Round-1 measured this negative (register pressure);
#                    re-tried after the packed-f32 dw rewrite.
_DW_LOP_ON = os.environ.get("YAMS_DW_LOP", "0") != "0"
#   YAMS_GRAPH_OVERLAP=1  keep the side-stream wgrads during hipGraph
#                    capture (stream-capture propagates across the event
#                    dependencies; measured: graph+overlap 14.41 vs
#                    graph-only 14.83 vs eager+overlap 13.77 ms — graphs
#                    stay off by default, but --graph runs capture the
#                    overlapped step)
_GRAPH_OVERLAP_ON = os.environ.get("YAMS_GRAPH_OVERLAP", "1") != "0"

_wgrad_stream = [None]


def wgrad_stream():
    if _wgrad_stream[0] is None:
        _wgrad_stream[0] = torch.cuda.Stream()
    return _wgrad_stream[0]


class FusedInvertedResidualFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, ew, eg, ebb, erm, erv, dww, dg, db, drm, drv,
                sw1, sb1, sw2, sb2, pw, pg, pb, prm, prv,
                k, stride, act, training, momentum, eps, use_res):
        e = ext()
        x = _nhwc(x)
        dev = x.device
        n, cin, h, wdt = x.shape
        pad = k // 2
        has_expand = ew is not None
        has_se = sw1 is not None

        # ---- expand pw ------------------------------------------------
        if has_expand:
            ce = ew.shape[0]
            ewp = _packed(ew, "pw", lambda w: w.view(ce, cin).to(BF16)
                          .contiguous(), spec="cast")
            z1 = _empty_like_nhwc(x, c=ce)
            m1 = n * h * wdt
            mean1, invstd1, ab1 = _bn_bufs(ce, dev)
            if training:
                mb = e.pw_stats_mblocks(m1)
                ws = torch.empty(2 * mb * ce, dtype=F32, device=dev)
                e.pw_conv(x, ewp, z1, stats_ws=ws)
                e.bn_finalize(ws, mb, mean1, invstd1, ab1, eg, ebb,
                              erm, erv, momentum, eps, m1)
            else:
                e.pw_conv(x, ewp, z1)
                mean1.copy_(erm)
                invstd1.copy_((erv.float() + eps).rsqrt())
                _eval_ab(mean1, invstd1, eg, ebb, ab1)
            if _DW_LOP_ON:
                # y1 never exists: dw fwd/wgrad apply BN+act on load
                y1 = None
                dw_in = z1
                dw_lop = (1, act, ab1)
            else:
                # materialize y1: the dw window-load lop costs more
                # registers than this memory-bound pass costs time
                # (round-1 measurement; YAMS_DW_LOP re-tests it)
                y1 = torch.empty_like(z1)
                e.bn_act_fwd(z1, ab1, act, y1, None)
                dw_in = y1
                dw_lop = (0, 0, None)
        else:
            z1, mean1, invstd1, ab1 = x, None, None, None
            y1 = None
            dw_in = x
            dw_lop = (0, 0, None)
            ce = cin

        # ---- depthwise ------------------------------------------------
        cd = dww.shape[0]
        dwp = _packed(dww, "dw", lambda w: w.view(cd, k * k).t().to(BF16)
                      .contiguous(), spec="t")
        ho = (h + 2 * pad - k) // stride + 1
        wo = (wdt + 2 * pad - k) // stride + 1
        z2 = torch.empty((n, cd, ho, wo), dtype=BF16, device=dev,
                         memory_format=CL)
        m2 = n * ho * wo
        mean2, invstd2, ab2 = _bn_bufs(cd, dev)
        if training:
            chunks = e.dw_stats_chunks(n, ho, wo, cd, stride, k)
            ws = torch.empty(2 * chunks * cd, dtype=F32, device=dev)
            e.dw_conv_fwd(dw_in, dwp, z2, k, stride, pad,
                          lop_kind=dw_lop[0], lop_act=dw_lop[1],
                          lop_coef=dw_lop[2], stats_ws=ws)
            e.bn_finalize(ws, chunks, mean2, invstd2, ab2, dg, db,
                          drm, drv, momentum, eps, m2)
        else:
            e.dw_conv_fwd(dw_in, dwp, z2, k, stride, pad,
                          lop_kind=dw_lop[0], lop_act=dw_lop[1],
                          lop_coef=dw_lop[2])
            mean2.copy_(drm)
            invstd2.copy_((drv.float() + eps).rsqrt())
            _eval_ab(mean2, invstd2, dg, db, ab2)

        # ---- squeeze-excite -------------------------------------------
        if has_se:
            # ONE fused kernel (se.hip): BN+act applied on load (the
            # activated y2 is streamed out for project-wgrad/backward),
            # pool->FC1+ReLU->FC2+hsig->scale inside; fp32 internals
            from .hip_ops import se_forward
            y2 = torch.empty_like(z2)
            y2s, pooled, ha, sz = se_forward(
                e, z2, sw1, sb1, sw2, sb2, lop=(1, act, ab2), ymat=y2)
            pj_in, pj_lop = y2s, (0, 0, None)
        else:
            # no-SE: project reads z2 with the BN+act applied on load —
            # y2 is never materialized (single consumer)
            y2 = pooled = ha = sz = y2s = None
            pj_in, pj_lop = z2, (1, act, ab2)

        # ---- project pw -----------------------------------------------
        cp = pw.shape[0]
        pwp = _packed(pw, "pw", lambda w: w.view(cp, cd).to(BF16)
                      .contiguous(), spec="cast")
        z3 = torch.empty((n, cp, ho, wo), dtype=BF16, device=dev,
                         memory_format=CL)
        mean3, invstd3, ab3 = _bn_bufs(cp, dev)
        if training:
            mb = e.pw_stats_mblocks(m2)
            ws = torch.empty(2 * mb * cp, dtype=F32, device=dev)
            e.pw_conv(pj_in, pwp, z3, lop_kind=pj_lop[0], lop_act=pj_lop[1],
                      lop_coef=pj_lop[2], stats_ws=ws)
            e.bn_finalize(ws, mb, mean3, invstd3, ab3, pg, pb, prm, prv,
                          momentum, eps, m2)
        else:
            e.pw_conv(pj_in, pwp, z3, lop_kind=pj_lop[0], lop_act=pj_lop[1],
                      lop_coef=pj_lop[2])
            mean3.copy_(prm)
            invstd3.copy_((prv.float() + eps).rsqrt())
            _eval_ab(mean3, invstd3, pg, pb, ab3)

        # ---- block output: BN (linear) + residual ----------------------
        yb = torch.empty_like(z3)
        e.bn_act_fwd(z3, ab3, ACT_IDENTITY, yb, x if use_res else None)

        ctx.save_for_backward(
            x, z1 if has_expand else None, z2, z3,
            mean1, invstd1, ab1, mean2, invstd2, ab2, mean3, invstd3,
            pooled, ha, sz, y2s,
            y1 if has_expand else None, y2,
            ew, eg, ebb, dww, dg, db, sw1, sb1, sw2, sb2, pw, pg, pb)
        ctx.cfg = (k, stride, act, training, has_expand, has_se, use_res,
                   n, cin, h, wdt, ho, wo)
        return yb

    @staticmethod
    def backward(ctx, dyb):
        e = ext()
        (x, z1, z2, z3, mean1, invstd1, ab1, mean2, invstd2, ab2, mean3,
         invstd3, pooled, ha, sz, y2s, y1, y2, ew, eg, ebb, dww,
         dg, db, sw1, sb1, sw2, sb2, pw, pg, pb) = ctx.saved_tensors
        (k, stride, act, training, has_expand, has_se, use_res, n, cin, h,
         wdt, ho, wo) = ctx.cfg
        dev = x.device
        pad = k // 2
        dyb = dyb.to(BF16).contiguous(memory_format=CL)
        cd = dww.shape[0]
        cp = pw.shape[0]
        ce = ew.shape[0] if has_expand else cin
        # side-stream wgrad overlap: wgrads only read (dz, activations),
        # so they run concurrently with the rest of the backward chain;
        # a hard join before returning keeps allocator/stream semantics
        # trivially safe (nothing outlives the block's backward)
        overlap = _WGRAD_STREAM_ON and (
            _GRAPH_OVERLAP_ON
            or not torch.cuda.is_current_stream_capturing())
        if overlap:
            sw = wgrad_stream()
            cur = torch.cuda.current_stream()

        def _on_wgrad_stream():
            # make the side stream wait for everything main has queued,
            # then return the context that routes the wgrads onto it
            ev = torch.cuda.Event()
            ev.record(cur)
            sw.wait_event(ev)
            return torch.cuda.stream(sw)

        # ---- project bn backward ---------------------------------------
        dg3 = torch.empty(cp, dtype=F32, device=dev)
        db3 = torch.empty_like(dg3)
        pwt = _packed(pw, "pw_t", lambda w: w.view(cp, cd).t().to(BF16)
                      .contiguous(), spec="t")
        d_in3 = torch.empty_like(z2)
        lopdz = _LOPDZ_ON and not _EPI_ON
        if lopdz:
            # dz3 is NEVER materialized: reduce-only backward, then both
            # GEMMs recompute dz on load (LOP_BNBWD) — kills the apply
            # pass (read dy + read z + write dz) on the critical path
            coef3 = torch.empty(4 * cp, dtype=F32, device=dev)
            e.bn_bwd_stats(dyb, z3, mean3, invstd3, pg, pb, ACT_IDENTITY,
                           bool(training), dg3, db3, coef3)
            ws2 = sp2 = None
            e.pw_conv(dyb, pwt, d_in3, lop_kind=2, lop_act=ACT_IDENTITY,
                      lop_coef=coef3, lop_z=z3)
            dz3 = dyb  # the wgrad A-stream applies the same lop
        else:
            dz3 = torch.empty_like(z3)
            e.bn_act_bwd(dyb, z3, mean3, invstd3, pg, pb, ACT_IDENTITY,
                         bool(training), dg3, db3, dz3)
            if not has_se and _EPI_ON:
                # d_in3 IS the dw-BN's backward dy: emit that BN's partial
                # sums from the GEMM epilogue (EPI==2)
                m2 = n * ho * wo
                sp2 = int(e.pw_stats_mblocks(m2))
                ws2 = torch.empty(2 * sp2 * cd, dtype=F32, device=dev)
                e.pw_conv(dz3, pwt, d_in3, stats_ws=ws2, bstats_act=act,
                          bstats_coef=ab2, bstats_z=z2)
            else:
                ws2 = sp2 = None
                e.pw_conv(dz3, pwt, d_in3)

        def _wgrad3():
            dwp3 = torch.empty(cp, cd, dtype=F32, device=dev)
            la = dict(lopa_kind=2, lopa_act=ACT_IDENTITY, lopa_coef=coef3,
                      lopa_z=z3) if lopdz else {}
            if has_se:
                e.pw_conv_wgrad(dz3, y2s, dwp3, **la)
            else:
                e.pw_conv_wgrad(dz3, z2, dwp3, lopb_kind=1, lopb_act=act,
                                lopb_coef=ab2, **la)
            return dwp3

        if overlap:
            with _on_wgrad_stream():
                dwp3 = _wgrad3()
        else:
            dwp3 = _wgrad3()
        d_pw = dwp3.reshape(cp, cd, 1, 1)

        # ---- SE backward: fused per-image + wgrad kernels (se.hip) ------
        if has_se:
            from .hip_ops import se_backward
            # the dw BN's backward partials ride se_bwd's dx-writing
            # phase (one less full pass over d_y2 + z2, one less launch)
            se_ws = torch.empty(2 * n * cd, dtype=F32, device=dev)
            d_y2, dw1p, db1se, dw2p, db2se = se_backward(
                e, d_in3, y2, sw1, sw2, pooled, ha, sz,
                bn=(z2, ab2, act, se_ws))
            d_sw1 = dw1p.reshape_as(sw1)
            d_sw2 = dw2p.reshape_as(sw2)
        else:
            d_y2 = d_in3
            se_ws = None
            d_sw1 = d_sw2 = db1se = db2se = None

        # ---- depthwise bn backward + grads (dz2 materialized) -----------
        dg2 = torch.empty(cd, dtype=F32, device=dev)
        db2_ = torch.empty_like(dg2)
        dz2 = torch.empty_like(z2)
        if ws2 is not None:
            e.bn_act_bwd(d_y2, z2, mean2, invstd2, dg, db, act,
                         bool(training), dg2, db2_, dz2,
                         partial_ws=ws2, partial_splits=sp2)
        elif se_ws is not None:
            # partial sums already emitted by se_bwd's dx phase
            e.bn_act_bwd(d_y2, z2, mean2, invstd2, dg, db, act,
                         bool(training), dg2, db2_, dz2,
                         partial_ws=se_ws, partial_splits=n)
        else:
            e.bn_act_bwd(d_y2, z2, mean2, invstd2, dg, db, act,
                         bool(training), dg2, db2_, dz2)
        dwpk = _packed(dww, "dw", lambda w: w.view(cd, k * k).t().to(BF16)
                       .contiguous(), spec="t")
        dw_lop_b = {}
        if has_expand and y1 is None:
            dw_in = z1  # YAMS_DW_LOP: wgrad re-applies BN+act on load
            dw_lop_b = dict(lopx_kind=1, lopx_act=act, lopx_coef=ab1)
        else:
            dw_in = y1 if has_expand else x
        d_z1out = torch.empty_like(dw_in)
        if has_expand and stride == 1 and _EPI_ON:
            # d_z1out feeds the expand BN's backward: fuse its partials
            sp1 = int(e.dw_stats_chunks(n, h, wdt, ce, 1, k))
            ws1 = torch.empty(2 * sp1 * ce, dtype=F32, device=dev)
            e.dw_conv_dgrad(dz2, dwpk, d_z1out, k, stride, pad,
                            stats_ws=ws1, bstats_act=act, bstats_coef=ab1,
                            bstats_z=z1)
        else:
            ws1 = sp1 = None
            e.dw_conv_dgrad(dz2, dwpk, d_z1out, k, stride, pad)
        if overlap:
            with _on_wgrad_stream():
                dwgt = torch.empty(cd, k * k, dtype=F32, device=dev)
                e.dw_conv_wgrad(dw_in, dz2, dwgt, k, stride, pad,
                                **dw_lop_b)
        else:
            dwgt = torch.empty(cd, k * k, dtype=F32, device=dev)
            e.dw_conv_wgrad(dw_in, dz2, dwgt, k, stride, pad, **dw_lop_b)
        d_dww = dwgt.reshape(cd, 1, k, k)

        # ---- expand backward --------------------------------------------
        if has_expand:
            dg1 = torch.empty(ce, dtype=F32, device=dev)
            db1_ = torch.empty_like(dg1)
            ewt = _packed(ew, "pw_t", lambda w: w.view(ce, cin).t().to(BF16)
                          .contiguous(), spec="t")
            d_x = _empty_like_nhwc(x)
            if lopdz and ws1 is None:
                coef1 = torch.empty(4 * ce, dtype=F32, device=dev)
                e.bn_bwd_stats(d_z1out, z1, mean1, invstd1, eg, ebb, act,
                               bool(training), dg1, db1_, coef1)
                # residual grad + BNBWD load-op fused into the dgrad
                e.pw_conv(d_z1out, ewt, d_x, lop_kind=2, lop_act=act,
                          lop_coef=coef1, lop_z=z1,
                          residual=dyb if use_res else None)
                dz1 = d_z1out
                la1 = dict(lopa_kind=2, lopa_act=act, lopa_coef=coef1,
                           lopa_z=z1)
            else:
                dz1 = torch.empty_like(z1)
                if ws1 is not None:
                    e.bn_act_bwd(d_z1out, z1, mean1, invstd1, eg, ebb, act,
                                 bool(training), dg1, db1_, dz1,
                                 partial_ws=ws1, partial_splits=sp1)
                else:
                    e.bn_act_bwd(d_z1out, z1, mean1, invstd1, eg, ebb, act,
                                 bool(training), dg1, db1_, dz1)
                # residual grad fused into the dgrad epilogue (zero extra
                # passes; the eager add cost 3 passes x 10 blocks/step)
                e.pw_conv(dz1, ewt, d_x, residual=dyb if use_res else None)
                la1 = {}
            if overlap:
                with _on_wgrad_stream():
                    dwp1 = torch.empty(ce, cin, dtype=F32, device=dev)
                    e.pw_conv_wgrad(dz1, x, dwp1, **la1)
            else:
                dwp1 = torch.empty(ce, cin, dtype=F32, device=dev)
                e.pw_conv_wgrad(dz1, x, dwp1, **la1)
            d_ew = dwp1.reshape(ce, cin, 1, 1)
        else:
            d_x = d_z1out
            d_ew = dg1 = db1_ = None

        if use_res and not has_expand:
            d_x = d_x + dyb  # rare no-expand residual block (V3-L #1)

        if overlap:
            ev_w = torch.cuda.Event()
            ev_w.record(sw)
            cur.wait_event(ev_w)

        return (d_x, d_ew, dg1, db1_, None, None,
                d_dww, dg2, db2_, None, None,
                d_sw1, db1se, d_sw2, db2se,
                d_pw, dg3, db3, None, None,
                None, None, None, None, None, None, None)


def fused_inverted_residual(x, block):
    """Run one InvertedResidual module through the fused HIP path."""
    ew = eg = ebb = erm = erv = None
    if block.expand is not None:
        ew = block.expand.conv.weight
        eg, ebb = block.expand.bn.weight, block.expand.bn.bias
        erm, erv = block.expand.bn.running_mean, block.expand.bn.running_var
    sw1 = sb1 = sw2 = sb2 = None
    if block.se is not None:
        sw1, sb1 = block.se.fc1.weight, block.se.fc1.bias
        sw2, sb2 = block.se.fc2.weight, block.se.fc2.bias
    # num_batches_tracked is bumped once per model forward (ZooModel)
    dwm, pjm = block.dw, block.project
    return FusedInvertedResidualFn.apply(
        x, ew, eg, ebb, erm, erv,
        dwm.conv.weight, dwm.bn.weight, dwm.bn.bias,
        dwm.bn.running_mean, dwm.bn.running_var,
        sw1, sb1, sw2, sb2,
        pjm.conv.weight, pjm.bn.weight, pjm.bn.bias,
        pjm.bn.running_mean, pjm.bn.running_var,
        dwm.conv.weight.shape[2], dwm.stride, dwm.act, block.training,
        dwm.bn.momentum, dwm.bn.eps, block.use_residual)
