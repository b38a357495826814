"""Checkpoint save/load.

On-disk format is the reference's convention: a ``torch.save`` dict
``{"model": state_dict, "ema": shadow, "optimizer": state, "epoch": int,
"lr_scheduler": state, "rng": ..., "meta": ...}`` with fp32 OIHW weights
(our modules already store weights that way, so no layout transform is
needed at the boundary).

State-dict key names go through a remap table so that, once the reference
repo's real key names are readable, loading its released checkpoints is a
data change (extend ``KEY_REMAP``), not a code change.
Shape adapters handle [out,in] Linear vs [out,in,1,1] 1x1-conv heads.
"""

from __future__ import annotations

import os
import random
import tempfile
from typing import Dict, Optional

import numpy as np
import torch

# our-name <- reference-name aliases (filled in when the reference mount
# becomes readable; identity for now).
KEY_REMAP: Dict[str, str] = {}


def _rng_state() -> Dict:
    # stored as primitives + tensors only, so the whole checkpoint loads
    # under torch.load(weights_only=True)
    np_name, np_key, np_pos, np_gauss, np_cached = np.random.get_state()
    state = {
        "python": list(random.getstate()[1]),
        "numpy": {"name": str(np_name),
                  "key": torch.from_numpy(np_key.astype(np.int64)),
                  "pos": int(np_pos), "has_gauss": int(np_gauss),
                  "cached": float(np_cached)},
        "torch": torch.get_rng_state(),
    }
    if torch.cuda.is_available():
        state["cuda"] = torch.cuda.get_rng_state_all()
    return state


def _load_rng_state(state: Dict) -> None:
    py = state["python"]
    if isinstance(py, list):  # current format
        random.setstate((3, tuple(py), None))
    else:  # legacy tuple format
        random.setstate(py)
    nps = state["numpy"]
    if isinstance(nps, dict):
        np.random.set_state((nps["name"],
                             nps["key"].cpu().numpy().astype(np.uint32),
                             nps["pos"], nps["has_gauss"], nps["cached"]))
    else:
        np.random.set_state(nps)
    # map_location may have moved the ByteTensors to GPU: bring them back
    torch.set_rng_state(state["torch"].cpu())
    if torch.cuda.is_available() and "cuda" in state:
        try:
            torch.cuda.set_rng_state_all([t.cpu() for t in state["cuda"]])
        except RuntimeError:
            pass  # different device count than at save time


def save_checkpoint(path: str, model, optimizer=None, ema=None,
                    lr_scheduler=None, epoch: int = 0, global_step: int = 0,
                    meta: Optional[Dict] = None) -> None:
    """Atomic (tmp+rename) checkpoint write; master-only is the caller's job."""
    state = {
        "model": model.state_dict(),
        "epoch": epoch,
        "global_step": global_step,
        "rng": _rng_state(),
        "meta": meta or {},
    }
    if optimizer is not None:
        state["optimizer"] = optimizer.state_dict()
    if ema is not None:
        state["ema"] = ema.state_dict()
    if lr_scheduler is not None:
        state["lr_scheduler"] = lr_scheduler.state_dict()
    d = os.path.dirname(path) or "."
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    os.close(fd)
    try:
        torch.save(state, tmp)
        os.replace(tmp, path)
    finally:
        if os.path.exists(tmp):
            os.unlink(tmp)


def remap_state_dict(sd: Dict[str, torch.Tensor], model) -> Dict[str, torch.Tensor]:
    """Apply KEY_REMAP aliases and shape adapters to a foreign state dict."""
    out = {}
    model_sd = model.state_dict()
    for k, v in sd.items():
        k = KEY_REMAP.get(k, k)
        if k in model_sd:
            tgt = model_sd[k]
            if v.shape != tgt.shape and v.numel() == tgt.numel():
                # e.g. reference stores the head as a 1x1 conv [out,in,1,1],
                # we store Linear [out,in] (or vice versa)
                v = v.reshape(tgt.shape)
        out[k] = v
    return out


def load_checkpoint(path: str, model, optimizer=None, ema=None,
                    lr_scheduler=None, strict: bool = True,
                    restore_rng: bool = True, map_location="cpu") -> Dict:
    # our own resume checkpoints carry RNG state (numpy tuples etc.) that
    # weights_only rejects; try the safe path first and fall back for the
    # trusted-resume case only (ADVICE.md round-1 low finding)
    try:
        state = torch.load(path, map_location=map_location, weights_only=True)
    except Exception:
        state = torch.load(path, map_location=map_location, weights_only=False)
    sd = state["model"] if "model" in state else state
    model.load_state_dict(remap_state_dict(sd, model), strict=strict)
    if optimizer is not None and "optimizer" in state:
        optimizer.load_state_dict(state["optimizer"])
    if ema is not None and "ema" in state:
        ema.load_state_dict(state["ema"])
    if lr_scheduler is not None and "lr_scheduler" in state:
        lr_scheduler.load_state_dict(state["lr_scheduler"])
    if restore_rng and "rng" in state:
        _load_rng_state(state["rng"])
    return state


def load_pretrained(path: str, model, strict: bool = False) -> None:
    """Load weights only (the reference's ``pretrained`` config key).

    Pretrained files are potentially FOREIGN (e.g. the reference repo's
    released checkpoints), so this path never executes pickled code:
    ``weights_only=True``, no fallback.
    """
    state = torch.load(path, map_location="cpu", weights_only=True)
    sd = state.get("model", state) if isinstance(state, dict) else state
    model.load_state_dict(remap_state_dict(sd, model), strict=strict)
