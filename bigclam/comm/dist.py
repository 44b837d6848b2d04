"""Distributed communication: RCCL over xGMI (GPU) / gloo (CPU tests).

Replaces the reference's Spark driver-centric movement (broadcast + shuffle +
driver collect — full inventory SURVEY.md §5.8).  Per sweep the only traffic
is:

- C8  halo exchange: boundary F rows via ``all_to_all_single`` (RCCL
  all-to-all is point-to-point over the 7 xGMI links — exactly the p2p
  send/recv-list pattern, scheduled by RCCL);
- C12 ``all_reduce`` of the 1xK column-sum delta;
- C14 ``all_reduce`` of the scalar LLH.

``backend="nccl"`` IS RCCL on ROCm.  All wrappers degrade to no-ops at
world_size 1 so the single-GPU and test paths share code.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def init_distributed(backend: str = None, timeout_s: int = 600) -> int:
    """Initialize from torchrun env vars; returns rank.  Safe to call when
    WORLD_SIZE is absent or 1 (no-op)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1 or is_distributed():
        return get_rank()
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    if "MASTER_PORT" not in os.environ:
        # no rendezvous info (manual launch without torchrun): pick a port
        # deterministically from job-identifying state every rank shares —
        # cwd + world size — so concurrent jobs in different directories
        # don't collide on a fixed hardcoded port.
        import hashlib

        h = hashlib.sha1(
            f"{os.getcwd()}:{world}".encode()
        ).digest()
        os.environ["MASTER_PORT"] = str(20000 + int.from_bytes(h[:2], "big") % 20000)
    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
    )
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return dist.get_rank()


def backend() -> str:
    """Active process-group backend name ('' when not distributed)."""
    return str(dist.get_backend()) if is_distributed() else ""


def all_reduce_(t: torch.Tensor) -> torch.Tensor:
    """In-place SUM all-reduce (no-op at world_size 1)."""
    if is_distributed():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def all_to_all(
    out: torch.Tensor,
    inp: torch.Tensor,
    out_splits,
    in_splits,
    async_op: bool = False,
):
    """Row-wise all_to_all_single on 2-D tensors (no-op at world_size 1).

    With ``async_op`` returns the Work handle (or None when not
    distributed) so the halo exchange can overlap with interior-node
    compute; the caller must ``wait()`` before reading ``out``.
    """
    if not is_distributed():
        return None if async_op else out
    work = dist.all_to_all_single(
        out,
        inp,
        output_split_sizes=list(out_splits),
        input_split_sizes=list(in_splits),
        async_op=async_op,
    )
    return work if async_op else out


def barrier():
    if is_distributed():
        dist.barrier()
