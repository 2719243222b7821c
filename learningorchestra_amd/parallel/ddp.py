"""Data parallelism: RCCL all-reduce over xGMI (one process per GPU).

Replaces the reference's Spark worker fan-out. On ROCm the
torch.distributed "nccl" backend IS RCCL; on CPU test hosts "gloo" runs the
same code path (multi-process CPU tests, world_size 2).

The gradient "bucket" is the ParamArena's single flat fp32 grad tensor, so
one step needs exactly ONE all-reduce — sized in the hundreds of MB for the
big models, which is the right granularity for the 7-link point-to-point xGMI
fabric (per-link-bound rings want few, large transfers — SURVEY §5.8).
"""
from __future__ import annotations

import os
from datetime import timedelta
from typing import Optional

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def init_distributed(backend: Optional[str] = None, timeout_s: int = 300) -> int:
    """Initialize from torchrun env (RANK/WORLD_SIZE/LOCAL_RANK). Returns
    local rank. No-op (rank 0) when not launched distributed."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if "RANK" not in os.environ or world == 1:
        return 0
    if backend is None:
        # RCCL (like NCCL) refuses two ranks on one device ("Duplicate GPU
        # detected", verified on hardware) — oversubscribed runs (e.g. a
        # 2-rank test on a 1-GPU lease) fall back to gloo, which supports
        # CUDA tensors by host staging. One-rank-per-GPU runs use RCCL.
        n_dev = torch.cuda.device_count() if torch.cuda.is_available() else 0
        backend = "nccl" if 0 < world <= n_dev else "gloo"
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if torch.cuda.is_available():
        local_rank %= max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(local_rank)
    if not dist.is_initialized():
        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=timeout_s))
    return local_rank


def oversubscribed() -> bool:
    """More ranks than visible GPUs (a multi-rank test on a 1-GPU lease).
    This is the only configuration where two of our processes time-share one
    device; production runs are one process per GPU over RCCL."""
    if not (is_distributed() and torch.cuda.is_available()):
        return False
    return get_world_size() > torch.cuda.device_count()


class device_step_lock:
    """Cross-process exclusive section for GPU work in oversubscribed runs.

    Measured on MI355X (tools/nanrepro7.py): when two ranks time-share one
    GPU, mid-kernel preemption of our long-running MFMA/LDS kernels while the
    peer's kernels or copy tails are in flight corrupts wavefront state —
    grad buffers pick up pointer-looking garbage (0x73xx_xxxx-range words).
    Pure-PyTorch kernels (short blocks) are immune; serializing whole steps
    (lock + synchronize before release, 0/15 vs 8/15 bad iterations) is
    throughput-neutral on a shared device and makes the run correct.
    No-op unless oversubscribed. NEVER hold this across a collective — the
    peer needs the device to reach its own collective call (deadlock)."""

    def __init__(self, device_index: Optional[int] = None):
        self._active = oversubscribed()
        self._fh = None
        self._path = None
        if self._active:
            import tempfile
            idx = (torch.cuda.current_device()
                   if device_index is None else device_index)
            self._path = os.path.join(tempfile.gettempdir(),
                                      f"lo_dev{idx}.lock")

    def __enter__(self):
        if self._active:
            import fcntl
            self._fh = open(self._path, "w")
            fcntl.flock(self._fh, fcntl.LOCK_EX)
        return self

    def __exit__(self, *exc):
        if self._fh is not None:
            try:
                torch.cuda.synchronize()   # drain before the peer runs
            finally:
                self._fh.close()           # closing the fd releases the flock
                self._fh = None
        return False

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None


def all_reduce_grads(grad_flat: torch.Tensor, async_op: bool = False):
    """SUM all-reduce of the flat grad arena (grads carry 1/(B*world) so sum
    = global-batch mean). The gloo+CUDA combination (oversubscribed test
    leases) stages through host memory explicitly, sync-bracketed: an
    un-drained copy tail overlapping the peer's kernels corrupts under
    mid-kernel preemption (see device_step_lock; measured 15/15 bad
    iterations without the trailing sync, 0/15 with)."""
    if not is_distributed():
        return None
    if grad_flat.is_cuda and dist.get_backend() == "gloo":
        torch.cuda.synchronize()
        host = grad_flat.detach().to("cpu")
        dist.all_reduce(host, op=dist.ReduceOp.SUM)
        grad_flat.copy_(host)
        torch.cuda.synchronize()
        return None
    return dist.all_reduce(grad_flat, op=dist.ReduceOp.SUM, async_op=async_op)


def barrier() -> None:
    if is_distributed():
        dist.barrier()
