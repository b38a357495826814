"""Distributed utilities (SURVEY.md §2.2 "Distributed", §5.8).

One process per GPU over ``torch.distributed``; backend ``"nccl"`` is RCCL
on ROCm (xGMI collectives), ``"gloo"`` is the CPU fake-world backend used
by the multi-process tests on GPU-less boxes.
"""

from __future__ import annotations

import datetime
import functools
import os
import threading
import time
from typing import Callable, Optional

import torch
import torch.distributed as dist


def init_dist(backend: str = "auto", timeout_sec: int = 300) -> int:
    """Initialise the process group from torchrun env vars; returns rank.

    No-op (rank 0, world 1) when launched without torchrun.
    """
    if dist.is_initialized():
        return dist.get_rank()
    if "WORLD_SIZE" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0
    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_sec))
    return rank


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", get_rank()))


def is_master() -> bool:
    return get_rank() == 0


def master_only(fn: Callable) -> Callable:
    """Decorator: run `fn` only on rank 0."""

    @functools.wraps(fn)
    def wrapper(*args, **kwargs):
        if is_master():
            return fn(*args, **kwargs)
        return None

    return wrapper


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def dist_all_reduce_tensor(t: torch.Tensor, op: str = "mean") -> torch.Tensor:
    """All-reduce a tensor across ranks (metric reduction, off the hot path)."""
    if not dist.is_initialized():
        return t
    t = t.clone()
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    if op == "mean":
        t /= dist.get_world_size()
    return t


def broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    """Broadcast parameters AND buffers from `src` (param sync at init)."""
    if not dist.is_initialized():
        return
    for t in list(module.parameters()) + list(module.buffers()):
        if t.dtype.is_floating_point or t.dtype in (torch.int64, torch.int32):
            dist.broadcast(t.data, src=src)


def destroy() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()


class Watchdog:
    """Abort the process (exit code 124) if the training step stalls.

    The process-group timeout covers collectives that RETURN an error, but
    a wedged GPU kernel or a hung RCCL channel can stall a rank without any
    API ever returning; every other rank then blocks in its next collective
    forever.  The watchdog is a daemon thread: the step
    loop calls :meth:`kick` once per step, and if no kick arrives within
    ``timeout_sec`` the watchdog logs and hard-exits non-zero so the
    launcher (torchrun) tears the job down and the checkpoint/resume path
    takes over.

    Use ``with Watchdog(120):`` around the step loop, or ``start()`` /
    ``stop()`` explicitly.  ``timeout_sec <= 0`` disables it (no thread).
    """

    EXIT_CODE = 124  # same convention as coreutils `timeout`

    def __init__(self, timeout_sec: float, name: str = "step",
                 logger=None) -> None:
        self.timeout_sec = float(timeout_sec)
        self.name = name
        self.logger = logger
        self._lock = threading.Lock()
        self._last = None
        self._stop = threading.Event()
        self._thread = None

    def start(self) -> "Watchdog":
        if self.timeout_sec <= 0 or self._thread is not None:
            return self
        self._last = time.monotonic()
        self._stop.clear()

        def _watch():
            while not self._stop.wait(min(self.timeout_sec / 4, 5.0)):
                with self._lock:
                    last = self._last
                idle = time.monotonic() - last
                if idle > self.timeout_sec:
                    msg = (f"watchdog: no {self.name} progress for "
                           f"{idle:.0f}s (> {self.timeout_sec:.0f}s) on "
                           f"rank {get_rank()}; aborting")
                    if self.logger is not None:
                        self.logger.error(msg)
                    print(msg, flush=True)
                    os._exit(self.EXIT_CODE)

        self._thread = threading.Thread(
            target=_watch, name="yams-watchdog", daemon=True)
        self._thread.start()
        return self

    def kick(self) -> None:
        if self._thread is None:
            return
        with self._lock:
            self._last = time.monotonic()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=1.0)
            self._thread = None

    def __enter__(self) -> "Watchdog":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
