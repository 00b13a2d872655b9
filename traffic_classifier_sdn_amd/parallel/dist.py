"""Distributed layer: RCCL (torch.distributed "nccl" backend on ROCm) over
the xGMI links of one 8x MI355X node; gloo for CPU-only tests.

The reference has no distributed backend; this module is the
framework's only comm layer.  Design points for MI355X:

* one process per GPU, ranks map to devices by LOCAL_RANK;
* gradients / sufficient statistics here are KB-scale, so collectives are
  latency-bound, not bandwidth-bound (7 p2p xGMI links x ~153 GB/s):
  every per-step reduction is FUSED into one flat buffer before the
  all-reduce (``allreduce_flat``) instead of one call per tensor;
* row-sharding helpers keep every rank's shard resident in its 288 GB HBM.
"""

from __future__ import annotations

import os
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as td


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def is_initialized() -> bool:
    return td.is_available() and td.is_initialized()


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialise the process group from torchrun env vars.

    backend defaults to "nccl" (== RCCL on ROCm) when a GPU is visible,
    else "gloo".  Returns (rank, world_size); world_size 1 with no env
    initialisation is a valid single-process configuration.
    """
    world = env_world_size()
    if world <= 1:
        return 0, 1
    if is_initialized():
        return td.get_rank(), td.get_world_size()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    td.init_process_group(backend=backend)
    if backend == "nccl":
        torch.cuda.set_device(env_local_rank())
    return td.get_rank(), td.get_world_size()


def world_size() -> int:
    return td.get_world_size() if is_initialized() else 1


def rank() -> int:
    return td.get_rank() if is_initialized() else 0


def barrier() -> None:
    if is_initialized():
        td.barrier()


def allreduce_(t: torch.Tensor, op=None) -> torch.Tensor:
    """In-place sum all-reduce; no-op when single-process."""
    if is_initialized():
        td.all_reduce(t, op=op or td.ReduceOp.SUM)
    return t


def allreduce_flat(tensors: Sequence[torch.Tensor]) -> None:
    """Fuse many small tensors into ONE all-reduce (latency-bound regime on
    xGMI: one collective beats per-tensor calls), then scatter back."""
    if not is_initialized() or not tensors:
        return
    flat = torch.cat([t.reshape(-1) for t in tensors])
    td.all_reduce(flat, op=td.ReduceOp.SUM)
    off = 0
    for t in tensors:
        n = t.numel()
        t.copy_(flat[off : off + n].view_as(t))
        off += n


def allgather(t: torch.Tensor) -> List[torch.Tensor]:
    if not is_initialized():
        return [t]
    out = [torch.empty_like(t) for _ in range(world_size())]
    td.all_gather(out, t.contiguous())
    return out


def broadcast_(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if is_initialized():
        td.broadcast(t, src=src)
    return t


def shard_range(n: int, r: Optional[int] = None, w: Optional[int] = None) -> Tuple[int, int]:
    """Contiguous row shard [lo, hi) of n rows for this rank (balanced)."""
    r = rank() if r is None else r
    w = world_size() if w is None else w
    base, extra = divmod(n, w)
    lo = r * base + min(r, extra)
    hi = lo + base + (1 if r < extra else 0)
    return lo, hi


def shard_rows(X: torch.Tensor) -> torch.Tensor:
    lo, hi = shard_range(X.shape[0])
    return X[lo:hi]
