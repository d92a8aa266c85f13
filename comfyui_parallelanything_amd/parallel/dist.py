"""Process-per-GPU transport: RCCL over xGMI via torch.distributed.

The reference moved every cross-device byte with host-visible ``.to()``
copies from a thread pool (census in SURVEY.md §2c). The MI355X-native scale
path is one process per GPU, ``torch.distributed`` with backend "nccl"
(which IS RCCL on ROCm), and stream-ordered collectives over xGMI:

- replicate(): flat dtype-bucketed broadcast (replicate.broadcast_module).
- per-step scatterv/gatherv: direct point-to-point sends — each MI355X has 7
  single-hop xGMI links (~153 GB/s each) to its peers, so a variable-size
  batch scatter is N-1 concurrent single-hop transfers, not a ring. Per-step
  payloads (a [8,16,128,128] bf16 latent is ~4 MB) are latency-bound;
  batch_isend_irecv posts them all in one group.

CPU tests run the same code over the gloo backend (world_size 2); the
collective call pattern is identical.
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Dict, List, Optional, Sequence

import torch
import torch.distributed as dist


class CommStats:
    """Per-process byte counters for the xGMI transport (SURVEY.md §5
    metrics: 'per-GPU busy time, xGMI bytes')."""

    def __init__(self):
        self.sent_bytes = 0
        self.recv_bytes = 0
        self.ops = 0

    def sent(self, t: torch.Tensor):
        self.sent_bytes += t.numel() * t.element_size()
        self.ops += 1

    def recvd(self, t: torch.Tensor):
        self.recv_bytes += t.numel() * t.element_size()
        self.ops += 1

    def summary(self) -> dict:
        return {"sent_bytes": self.sent_bytes, "recv_bytes": self.recv_bytes,
                "p2p_ops": self.ops}


COMM_STATS = CommStats()


@dataclass
class DistInfo:
    rank: int
    world_size: int
    local_rank: int
    device: torch.device
    backend: str

    @property
    def is_lead(self) -> bool:
        return self.rank == 0


def init_distributed(backend: Optional[str] = None, timeout_s: int = 600) -> DistInfo:
    """Initialize from torchrun env vars; single-process fallback when absent.

    Backend defaults to nccl (RCCL) when HIP devices are visible, gloo
    otherwise. Rendezvous must use 127.0.0.1 in this environment.
    """
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device(f"cuda:{local_rank % max(1, torch.cuda.device_count())}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    return DistInfo(rank, world, local_rank, device, backend)


def _offsets(sizes: Sequence[int]) -> List[int]:
    offs = [0]
    for s in sizes:
        offs.append(offs[-1] + s)
    return offs


def scatterv(
    full: Optional[torch.Tensor],
    sizes: Sequence[int],
    info: DistInfo,
    src: int = 0,
    template: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Scatter dim-0 chunks of ``full`` (valid on rank ``src``) by ``sizes``.

    Point-to-point: the source posts one isend per peer (each a single xGMI
    hop), peers post one irecv; the source's own chunk is a local slice.
    ``template`` supplies dtype/trailing-shape on non-source ranks (when
    None, full must be valid everywhere and is used as the template —
    useful when every rank already generated identical synthetic input).
    """
    sizes = list(sizes)
    assert len(sizes) == info.world_size
    if info.world_size == 1:
        assert full is not None
        return full
    if info.rank == src:
        assert full is not None and full.shape[0] == sum(sizes)
        offs = _offsets(sizes)
        chunks = [full[offs[i] : offs[i + 1]].contiguous() for i in range(len(sizes))]
        ops = []
        for r in range(info.world_size):
            if r != src and sizes[r] > 0:
                ops.append(dist.P2POp(dist.isend, chunks[r], peer=r))
                COMM_STATS.sent(chunks[r])
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        return chunks[src]
    ref = template if template is not None else full
    assert ref is not None, "non-source ranks need a template tensor"
    my = torch.empty(
        (sizes[info.rank], *ref.shape[1:]), dtype=ref.dtype, device=info.device
    )
    if sizes[info.rank] > 0:
        COMM_STATS.recvd(my)
        for w in dist.batch_isend_irecv([dist.P2POp(dist.irecv, my, peer=src)]):
            w.wait()
    return my


def gatherv(
    chunk: torch.Tensor,
    sizes: Sequence[int],
    info: DistInfo,
    dst: int = 0,
) -> Optional[torch.Tensor]:
    """Gather dim-0 chunks to rank ``dst``; returns the full tensor there,
    None elsewhere. Point-to-point mirror of scatterv."""
    sizes = list(sizes)
    assert len(sizes) == info.world_size
    if info.world_size == 1:
        return chunk
    if info.rank == dst:
        out = torch.empty(
            (sum(sizes), *chunk.shape[1:]), dtype=chunk.dtype, device=info.device
        )
        offs = _offsets(sizes)
        ops = []
        views = []
        for r in range(info.world_size):
            if sizes[r] == 0:
                continue
            view = out[offs[r] : offs[r + 1]]
            if r == dst:
                view.copy_(chunk)
            else:
                buf = torch.empty_like(view)
                views.append((view, buf))
                ops.append(dist.P2POp(dist.irecv, buf, peer=r))
                COMM_STATS.recvd(buf)
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        for view, buf in views:
            view.copy_(buf)
        return out
    if sizes[info.rank] > 0:
        c = chunk.contiguous()
        COMM_STATS.sent(c)
        for w in dist.batch_isend_irecv([dist.P2POp(dist.isend, c, peer=dst)]):
            w.wait()
    return None


def barrier(info: DistInfo) -> None:
    if info.world_size > 1 and dist.is_initialized():
        if info.backend == "nccl":
            dist.barrier(device_ids=[info.device.index])
        else:
            dist.barrier()


def all_max(value: float, info: DistInfo) -> float:
    """MAX over ranks of a scalar (bench timing contract)."""
    if info.world_size <= 1 or not dist.is_initialized():
        return value
    t = torch.tensor(
        [value],
        dtype=torch.float64,
        device=info.device if info.backend == "nccl" else "cpu",
    )
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())
