"""HIP extension loader and engine-mode switch.

The compute engine has exactly two paths (no multi-backend dispatch):

- ``hip``: the hand-written CDNA4 (gfx950) kernels in ``ops/csrc``, used for
  every CUDA(=ROCm) tensor.  If the extension is not built, GPU execution
  **fails loudly** — there is no silent eager fallback on a GPU box.
- ``reference``: the pure-PyTorch fp32 reference implementation, used for all
  CPU tensors (and, only when explicitly requested via
  ``set_engine_mode("reference")``, on GPU for parity tests / baseline A/B).
"""

from __future__ import annotations

import importlib
import os
from typing import Optional

_EXT = None
_EXT_TRIED = False
_ENGINE_MODE = "hip"  # "hip" | "reference"

EXT_NAME = "_yams_hip"


def load_extension(required: bool = False):
    """Import the in-tree built HIP extension (ops/csrc → _yams_hip*.so)."""
    global _EXT, _EXT_TRIED
    if _EXT is not None:
        return _EXT
    if _EXT_TRIED and not required:
        return None
    _EXT_TRIED = True
    try:
        _EXT = importlib.import_module(
            f"yet_another_mobilenet_series_amd.ops.{EXT_NAME}")
    except ImportError:
        try:
            _EXT = importlib.import_module(EXT_NAME)
        except ImportError as e:
            if required:
                raise RuntimeError(
                    "yet_another_mobilenet_series_amd HIP extension "
                    f"({EXT_NAME}) is not built; run "
                    "`python setup.py build_ext --inplace` "
                    "(PYTORCH_ROCM_ARCH=gfx950). GPU execution refuses to "
                    "fall back to eager PyTorch."
                ) from e
            _EXT = None
    return _EXT


def extension_available() -> bool:
    return load_extension(required=False) is not None


def set_engine_mode(mode: str) -> None:
    global _ENGINE_MODE
    if mode not in ("hip", "reference"):
        raise ValueError(f"engine mode must be 'hip' or 'reference', got {mode!r}")
    _ENGINE_MODE = mode


def engine_mode() -> str:
    return _ENGINE_MODE


def use_hip(x) -> bool:
    """True if tensor `x` should run on the HIP kernel path."""
    return x.is_cuda and _ENGINE_MODE == "hip"


_FUSED_BLOCKS = os.environ.get("YAMS_FUSED", "1") != "0"


def use_fused_blocks() -> bool:
    """Block-level fusion toggle (YAMS_FUSED=0 falls back to per-layer
    HIP ops — used for A/B benchmarking, not a different backend)."""
    return _FUSED_BLOCKS


def set_fused_blocks(on: bool) -> None:
    global _FUSED_BLOCKS
    _FUSED_BLOCKS = on


def ext():
    """The extension module; raises if missing (GPU path must not fall back)."""
    return load_extension(required=True)


_SYNCBN = False


def set_syncbn(on: bool) -> None:
    """Cross-rank BN statistics (SURVEY.md §2.4 optional SyncBN): when on
    and a process group is initialised, training-mode BN computes batch
    statistics over the GLOBAL batch by all-reducing per-rank
    {sum, sumsq} before finalising.  Off by default (at 256/GPU the local
    statistics are the standard recipe)."""
    global _SYNCBN
    _SYNCBN = bool(on)


def use_syncbn() -> bool:
    if not _SYNCBN:
        return False
    import torch.distributed as dist
    return dist.is_initialized() and dist.get_world_size() > 1
