"""Data-parallel engine: bucketed gradient all-reduce over RCCL/xGMI.

Replaces the reference's thin ``AllReduceDistributedDataParallel``
 with an MI355X-first design:

- Parameters are bucketed in reverse registration order (the order their
  grads are produced during backward) into flat fp32 buffers; ``p.grad``
  is a *view* into its bucket, so autograd accumulates directly into
  contiguous memory and each all-reduce is one flat message — no
  gather/scatter copies.
- Default bucket count ≈ the 7 xGMI point-to-point links per GPU: the node
  is fully connected (no NVSwitch), so several in-flight ring all-reduces
  spread across links; with overlap on, each bucket's reduce launches as
  soon as backward finishes its last parameter and runs on RCCL's comm
  stream under the remaining backward.
- ``finish()`` waits on all handles and averages.  The same code runs the
  CPU fake-world tests over gloo (world_size 2–8, no GPUs).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .distributed import broadcast_module, get_world_size


#: number of point-to-point xGMI links per GPU on an 8-OAM MI355X node.
#: The node is fully connected (no switch): a single ring all-reduce is
#: bound by ONE link (~153 GB/s), so the engine keeps ~XGMI_LINKS buckets
#: in flight — RCCL pipelines them across links/channels — and each bucket
#: launches as soon as backward produces its last gradient.
XGMI_LINKS = 7

#: floor on the bucket size: below ~1 MB the per-collective launch/sync
#: overhead (~20 us) exceeds the wire time of the message itself
#: (1 MB over one 153 GB/s link ≈ 7 us x 2(8-1)/8 steps ≈ 12 us).
MIN_BUCKET_MB = 1.0


def auto_bucket_mb(total_bytes: int) -> float:
    """Bucket size targeting ~XGMI_LINKS in-flight reduces per step.

    V3-Large (21.9 MB fp32 grads) -> ~3.1 MB buckets; small zoo models
    degrade gracefully to fewer, MIN_BUCKET_MB-sized buckets.  The wire
    math (xGMI 7x153 GB/s, ring per-link bound) is laid out in
    docs/COMM.md with the measured tuning table.
    """
    mb = total_bytes / (1024.0 * 1024.0)
    return max(mb / XGMI_LINKS, MIN_BUCKET_MB)


class _Bucket:
    __slots__ = ("params", "flat", "stage", "pending", "handle",
                 "ready_event")

    def __init__(self, params: List[torch.nn.Parameter], device, dtype,
                 reduce_dtype) -> None:
        self.params = params
        total = sum(p.numel() for p in params)
        self.flat = torch.zeros(total, device=device, dtype=dtype)
        # optional reduced-precision wire buffer (bf16 halves the per-link
        # bytes; local accumulation stays fp32)
        self.stage = (torch.zeros(total, device=device, dtype=reduce_dtype)
                      if reduce_dtype is not None and reduce_dtype != dtype
                      else None)
        off = 0
        for p in params:
            p.grad = self.flat[off:off + p.numel()].view_as(p)
            off += p.numel()
        self.pending = 0
        self.handle = None

    def launch(self) -> None:
        if self.stage is not None:
            self.stage.copy_(self.flat)
            self.handle = dist.all_reduce(self.stage, async_op=True)
        else:
            self.handle = dist.all_reduce(self.flat, async_op=True)

    def wait(self) -> None:
        self.handle.wait()
        self.handle = None
        if self.stage is not None:
            self.flat.copy_(self.stage)


class DataParallelEngine(nn.Module):
    """Wrap a module for single-node data parallelism.

    Usage::

        engine = DataParallelEngine(model, bucket_mb=4, overlap=True)
        engine.zero_grad()
        loss = crit(engine(x), y)
        loss.backward()          # buckets all-reduce as they fill
        engine.finish()          # wait + average
        optimizer.step()
    """

    def __init__(self, module: nn.Module, bucket_mb: Optional[float] = None,
                 overlap: bool = True, grad_dtype: torch.dtype = torch.float32,
                 broadcast_params: bool = True,
                 force_bucketed: bool = False,
                 reduce_dtype: Optional[torch.dtype] = None) -> None:
        """``bucket_mb=None`` sizes buckets for the 7 xGMI links
        (:func:`auto_bucket_mb`).  ``reduce_dtype=torch.bfloat16`` reduces
        on a bf16 wire buffer (half the xGMI bytes; fp32 local accumulation
        is unchanged) — off by default for bitwise-exact parity with the
        single-process run (SURVEY.md §5.8 "fp32 first")."""
        super().__init__()
        self.module = module
        self.world_size = get_world_size()
        self.overlap = overlap and self.world_size > 1
        self.grad_dtype = grad_dtype
        self.reduce_dtype = reduce_dtype
        if broadcast_params and self.world_size > 1:
            broadcast_module(module)
        # single-process: skip flat buckets entirely — preset grad views
        # force an extra AccumulateGrad add per param per step (~200 tiny
        # kernels); plain assigned grads are free
        self._bucketed = self.world_size > 1 or force_bucketed
        params = [p for p in module.parameters() if p.requires_grad]
        if not self._bucketed:
            self.buckets = []
            self._param_bucket = {}
            self._hooks = []
            self._launched = []
            self._params = params
            self._no_sync = False
            return
        device = params[0].device if params else torch.device("cpu")
        # reverse registration order ≈ grad production order during backward
        params = list(reversed(params))
        if bucket_mb is None:
            total_bytes = sum(p.numel() for p in params) * 4
            bucket_mb = auto_bucket_mb(total_bytes)
        self.bucket_mb = bucket_mb
        bucket_elems = max(int(bucket_mb * 1024 * 1024 / 4), 1)
        self.buckets: List[_Bucket] = []
        cur: List[nn.Parameter] = []
        cur_n = 0
        for p in params:
            cur.append(p)
            cur_n += p.numel()
            if cur_n >= bucket_elems:
                self.buckets.append(_Bucket(cur, device, grad_dtype,
                                            reduce_dtype))
                cur, cur_n = [], 0
        if cur:
            self.buckets.append(_Bucket(cur, device, grad_dtype,
                                        reduce_dtype))
        self._param_bucket: Dict[nn.Parameter, _Bucket] = {}
        for b in self.buckets:
            for p in b.params:
                self._param_bucket[p] = b
        self._hooks = []
        if self.overlap:
            for p in params:
                h = p.register_post_accumulate_grad_hook(self._grad_hook)
                self._hooks.append(h)
        self._launched: List[_Bucket] = []
        self._no_sync = False

    # -- training-step API -------------------------------------------------

    def set_no_sync(self, flag: bool) -> None:
        """Gradient accumulation: while True, backward accumulates into the
        flat buckets but no all-reduce is launched (DDP.no_sync semantics).
        Call with False before the boundary micro-step so its backward
        triggers the (overlapped) reduce of the accumulated sums."""
        if self._no_sync and not flag and self._bucketed:
            for b in self.buckets:   # re-arm fill counters for the boundary
                b.pending = len(b.params)
                b.handle = None
            self._launched.clear()
        self._no_sync = flag

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def zero_grad(self, set_to_none: bool = False) -> None:  # noqa: ARG002
        """Zero the flat bucket buffers (grads are views; never set to None).
        Single-process mode: plain set-to-none."""
        if not self._bucketed:
            for p in self._params:
                p.grad = None
            return
        for b in self.buckets:
            b.flat.zero_()
            b.pending = len(b.params)
            b.handle = None
        self._launched.clear()

    def _grad_hook(self, p: nn.Parameter) -> None:
        if self._no_sync:
            return
        b = self._param_bucket[p]
        b.pending -= 1
        if b.pending == 0 and self.world_size > 1:
            b.launch()
            self._launched.append(b)

    def finish(self) -> None:
        """Wait for in-flight reduces (or run them now) and average."""
        if self.world_size <= 1 or self._no_sync:
            return
        if self.overlap:
            for b in self.buckets:
                if b.handle is None:
                    # a bucket whose params got no grads this step: reduce
                    # anyway so ranks stay collectively consistent
                    b.launch()
            for b in self.buckets:
                b.wait()
        else:
            for b in self.buckets:
                b.launch()
            for b in self.buckets:
                b.wait()
        inv = 1.0 / self.world_size
        torch._foreach_mul_([b.flat for b in self.buckets], inv)

    # -- passthrough -------------------------------------------------------

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)

    def train(self, mode: bool = True):
        self.module.train(mode)
        return self

    def eval(self):
        return self.train(False)
