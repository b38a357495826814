"""Guard-band (canary) allocation checking for the hand-written kernels.

SURVEY.md §5.2(d): a debug allocator that places canary bytes around
tensors to catch out-of-bounds writes from hand-written HIP kernels.

Inside :func:`guard_allocations`, the Python-level allocation entry
points (``torch.empty``, ``torch.zeros``, ``torch.empty_like``,
``torch.zeros_like``) are patched so every buffer the op layer hands to
a kernel is embedded in a slightly larger slab with ``0xAB`` canary
bytes on both sides; :func:`check` (run automatically on context exit)
synchronizes the device, re-reads every canary and raises
:class:`GuardViolation` on corruption.

Only Python-level allocations are intercepted — ATen-internal
allocations (autograd buffers, library ops) never route through these
symbols — so the guard covers exactly the tensors our HIP kernels write
into, at zero cost outside the context.  This is a debug/test tool (see
``tests/test_guard.py`` and the GPU OOB sweep in
``tests/test_kernels_gpu.py``), not a hot-path feature.
"""

from __future__ import annotations

import contextlib
from typing import Dict, List, Optional, Tuple

import torch

#: canary elements on each side of every guarded allocation.  64 elements
#: is far beyond any kernel's vector width (widest access is 16 bytes), so
#: any overrun that misses the pad entirely would also miss a neighboring
#: tensor in a non-guarded run.
PAD_ELEMS = 64
_CANARY = 0xAB

_ORIG: Dict[str, object] = {}
_live: List["_Guard"] = []


class GuardViolation(RuntimeError):
    """A kernel wrote outside one of its output buffers."""


class _Guard:
    __slots__ = ("slab", "pad_bytes", "label")

    def __init__(self, slab: torch.Tensor, pad_bytes: int, label: str):
        self.slab = slab
        self.pad_bytes = pad_bytes
        self.label = label

    def ok(self) -> bool:
        u8 = self.slab.view(torch.uint8)
        return bool((u8[: self.pad_bytes] == _CANARY).all()) and \
            bool((u8[u8.numel() - self.pad_bytes:] == _CANARY).all())


def _norm_shape(args) -> Optional[Tuple[int, ...]]:
    if len(args) == 1 and isinstance(args[0], (tuple, list, torch.Size)):
        return tuple(int(s) for s in args[0])
    if args and all(isinstance(a, int) for a in args):
        return tuple(args)
    return None


def _guarded(shape: Tuple[int, ...], dtype: torch.dtype, device,
             memory_format, zero: bool) -> Optional[torch.Tensor]:
    numel = 1
    for s in shape:
        numel *= s
    if numel <= 0:
        return None
    if memory_format not in (None, torch.contiguous_format,
                             torch.channels_last):
        return None
    if memory_format == torch.channels_last and len(shape) != 4:
        return None
    slab = _ORIG["empty"](numel + 2 * PAD_ELEMS, dtype=dtype, device=device)
    esz = slab.element_size()
    u8 = slab.view(torch.uint8)
    u8[: PAD_ELEMS * esz].fill_(_CANARY)
    u8[(PAD_ELEMS + numel) * esz:].fill_(_CANARY)
    mid = slab[PAD_ELEMS: PAD_ELEMS + numel]
    if zero:
        mid.zero_()
    if memory_format == torch.channels_last:
        n, c, h, w = shape
        out = mid.view(n, h, w, c).permute(0, 3, 1, 2)
    else:
        out = mid.view(shape)
    _live.append(_Guard(slab, PAD_ELEMS * esz,
                        f"{tuple(shape)}/{str(dtype).replace('torch.', '')}"))
    return out


def _passthrough(kw) -> bool:
    return (kw.get("out") is not None or kw.get("pin_memory")
            or kw.get("requires_grad")
            or kw.get("layout") not in (None, torch.strided))


def _make_new(zero: bool):
    orig_key = "zeros" if zero else "empty"

    def _new(*args, **kw):
        shape = _norm_shape(args)
        if shape is None or _passthrough(kw):
            return _ORIG[orig_key](*args, **kw)
        dtype = kw.get("dtype") or torch.get_default_dtype()
        out = _guarded(shape, dtype, kw.get("device"),
                       kw.get("memory_format"), zero)
        return out if out is not None else _ORIG[orig_key](*args, **kw)

    return _new


def _make_like(zero: bool):
    orig_key = "zeros_like" if zero else "empty_like"

    def _like(t, **kw):
        if _passthrough(kw) or not isinstance(t, torch.Tensor):
            return _ORIG[orig_key](t, **kw)
        mf = kw.get("memory_format")
        if mf in (None, torch.preserve_format):
            if t.dim() == 4 and t.is_contiguous(
                    memory_format=torch.channels_last):
                mf = torch.channels_last
            elif t.is_contiguous():
                mf = torch.contiguous_format
            else:
                return _ORIG[orig_key](t, **kw)
        dtype = kw.get("dtype") or t.dtype
        device = kw.get("device") if kw.get("device") is not None else t.device
        out = _guarded(tuple(t.shape), dtype, device, mf, zero)
        return out if out is not None else _ORIG[orig_key](t, **kw)

    return _like


def check() -> None:
    """Sync the device and verify every live canary; raises GuardViolation."""
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    bad = [g.label for g in _live if not g.ok()]
    if bad:
        raise GuardViolation(
            f"guard-band corrupted on {len(bad)} of {len(_live)} "
            f"allocation(s): {', '.join(bad[:8])}")


@contextlib.contextmanager
def guard_allocations(check_on_exit: bool = True):
    """Patch torch allocation entry points; yields :func:`check`.

    Not reentrant.  On exit the patches are removed; if ``check_on_exit``
    the canaries are verified (raising :class:`GuardViolation` on any
    out-of-bounds write) and the guard registry is cleared either way.
    """
    if _ORIG:
        raise RuntimeError("guard_allocations is not reentrant")
    _ORIG.update(empty=torch.empty, zeros=torch.zeros,
                 empty_like=torch.empty_like, zeros_like=torch.zeros_like)
    torch.empty = _make_new(zero=False)
    torch.zeros = _make_new(zero=True)
    torch.empty_like = _make_like(zero=False)
    torch.zeros_like = _make_like(zero=True)
    try:
        yield check
    finally:
        torch.empty = _ORIG["empty"]
        torch.zeros = _ORIG["zeros"]
        torch.empty_like = _ORIG["empty_like"]
        torch.zeros_like = _ORIG["zeros_like"]
        _ORIG.clear()
        try:
            if check_on_exit:
                check()
        finally:
            _live.clear()
