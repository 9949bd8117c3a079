"""RCCL batch fan-out over xGMI.

Single-node data-parallel inference: one process per GPU over
``torch.distributed`` (backend "nccl" IS RCCL on ROCm; "gloo" for CPU tests).
Collective pattern per inference batch:

  1. broadcast model bytes root -> all (once at init);
  2. batch fan-out: broadcast the input batch + local row slice
     (xGMI is 7 p2p links x ~153 GB/s per GPU; for the small latency-critical
     messages here, broadcast+slice beats a scatter chain and keeps messages
     few and fat — SURVEY.md §5 'distributed communication backend');
  3. all_gather of per-rank prediction shards.

No TP/PP/SP/EP: the models are single-GPU-sized (288 GB HBM per GPU), the
only parallel axis is samples (SURVEY.md §2.2 'Parallelism strategies').
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist


@dataclass
class FanoutContext:
    rank: int
    world_size: int
    device: torch.device
    backend: str

    @property
    def is_root(self) -> bool:
        return self.rank == 0


def init_fanout(backend: Optional[str] = None, timeout_s: float = 300.0) -> FanoutContext:
    """Initialize from torchrun env (RANK/WORLD_SIZE/MASTER_*). Falls back to
    a single-process context when no env is present."""
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_gpu = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_gpu else "gloo"
    if use_gpu:
        torch.cuda.set_device(local_rank % max(torch.cuda.device_count(), 1))
        device = torch.device("cuda", local_rank % max(torch.cuda.device_count(), 1))
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return FanoutContext(rank=rank, world_size=world, device=device, backend=backend)


def broadcast_tensors(ctx: FanoutContext, tensors: Sequence[torch.Tensor]) -> None:
    """Broadcast model tensors root -> all (in place, device-resident).

    Few, fat messages: tensors are broadcast one by one (they are already
    large contiguous buffers — the forest node table is a single tensor)."""
    if ctx.world_size == 1:
        return
    for t in tensors:
        dist.broadcast(t, src=0)


def shard_rows(ctx: FanoutContext, X: torch.Tensor) -> Tuple[torch.Tensor, List[int]]:
    """Fan a [N, ...] batch out: broadcast + local row slice.

    Returns (local shard, per-rank row counts). Rows split as evenly as
    possible; rank r takes rows [offsets[r], offsets[r+1]).
    """
    counts = [X.shape[0] // ctx.world_size] * ctx.world_size
    for i in range(X.shape[0] % ctx.world_size):
        counts[i] += 1
    if ctx.world_size == 1:
        return X, counts
    dist.broadcast(X, src=0)
    off = sum(counts[: ctx.rank])
    return X[off : off + counts[ctx.rank]].contiguous(), counts


def scatter_rows(ctx: FanoutContext, X_root: Optional[torch.Tensor],
                 counts: List[int], row_shape: Tuple[int, ...],
                 dtype: torch.dtype) -> torch.Tensor:
    """Scatter row shards root -> ranks (RCCL send/recv under the hood).

    xGMI-aware alternative to broadcast+slice: every GPU pair has a direct
    link, so the root pushes each rank ONLY its shard (~constant time in
    world size instead of linear). Falls back to broadcast+slice if the
    backend lacks scatter. X_root is required on rank 0 ([sum(counts), ...]);
    other ranks pass None."""
    maxc = max(counts)
    if ctx.world_size == 1:
        assert X_root is not None
        return X_root
    # The path choice must be DETERMINISTIC and identical on every rank: a
    # per-call try/except here could diverge across ranks (root fails, peers
    # wait in scatter) and deadlock the job — so the decision is made from
    # the backend name alone, and a real scatter failure propagates loudly.
    if dist.get_backend() not in ("nccl", "gloo"):
        full = X_root if ctx.is_root else torch.empty(
            (sum(counts),) + row_shape, dtype=dtype, device=ctx.device)
        dist.broadcast(full, src=0)
        off = sum(counts[: ctx.rank])
        return full[off : off + counts[ctx.rank]].contiguous()
    recv = torch.empty((maxc,) + row_shape, dtype=dtype, device=ctx.device)
    scatter_list = None
    if ctx.is_root:
        scatter_list = []
        off = 0
        for c in counts:
            t = torch.zeros((maxc,) + row_shape, dtype=dtype, device=ctx.device)
            t[:c] = X_root[off : off + c]
            scatter_list.append(t)
            off += c
    dist.scatter(recv, scatter_list, src=0)
    return recv[: counts[ctx.rank]]


def scatter_rows_async(ctx: FanoutContext, X_root: Optional[torch.Tensor],
                       counts: List[int], row_shape: Tuple[int, ...],
                       dtype: torch.dtype):
    """Async scatter: returns (work, padded_recv). The caller overlaps the
    NEXT batch's scatter with the CURRENT batch's compute (serving
    pipeline): xGMI scatter is per-peer point-to-point, so the transfer
    runs on RCCL's comm stream concurrently with compute and the exposed
    per-step cost is max(compute, scatter) instead of the sum. Call
    work.wait() then slice recv[:counts[rank]].

    world 1: returns (None, X_root)."""
    if ctx.world_size == 1:
        assert X_root is not None
        return None, X_root
    maxc = max(counts)
    recv = torch.empty((maxc,) + row_shape, dtype=dtype, device=ctx.device)
    scatter_list = None
    if ctx.is_root:
        scatter_list = []
        off = 0
        for c in counts:
            t = torch.zeros((maxc,) + row_shape, dtype=dtype, device=ctx.device)
            t[:c] = X_root[off : off + c]
            scatter_list.append(t)
            off += c
    work = dist.scatter(recv, scatter_list, src=0, async_op=True)
    return work, recv


def all_gather_rows(ctx: FanoutContext, local: torch.Tensor,
                    counts: List[int]) -> torch.Tensor:
    """Gather per-rank row shards back into the full [N, ...] tensor."""
    if ctx.world_size == 1:
        return local
    maxc = max(counts)
    pad_shape = (maxc,) + tuple(local.shape[1:])
    send = local
    if local.shape[0] < maxc:
        send = torch.zeros(pad_shape, dtype=local.dtype, device=local.device)
        send[: local.shape[0]] = local
    elif not local.is_contiguous():
        send = local.contiguous()
    outs = [torch.empty(pad_shape, dtype=local.dtype, device=local.device)
            for _ in range(ctx.world_size)]
    dist.all_gather(outs, send)
    return torch.cat([o[: counts[r]] for r, o in enumerate(outs)], dim=0)


def destroy() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
