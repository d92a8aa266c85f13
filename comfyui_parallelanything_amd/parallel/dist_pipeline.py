"""Process-per-GPU pipeline mode: batch==1 layer sharding over RCCL p2p.

The in-process pipeline (parallel/pipeline.py) hands activations across
devices with peer copies; this is the process-group equivalent from
SURVEY.md §2c ("pipeline activation send/recv — RCCL p2p over xGMI"):
contiguous block ranges are assigned to RANKS by weight (same policy as the
in-process mode, reference any_device_parallel.py:1168-1178), every rank
replays the full forward, and each wrapped block either

- executes on its OWNER rank (receiving the true hidden state from the
  previous owner when the range changes hands, single-hop xGMI p2p), or
- passes its input through unchanged on non-owner ranks (blocks are
  shape-preserving, so downstream local code keeps consistent shapes).

After the last block, its owner sends the hidden state to rank 0, whose
final layers produce the real output; other ranks' outputs are dummies.
Works over gloo for the CPU tests — the call pattern is identical to RCCL.

Micro-batching note: the in-process pipeline supports GPipe-style
micro-batching (pipeline.py, batches too small for DP). Here it is
deliberately absent: in process-group mode any batch > 1 routes to DP,
which beats pipelining outright (no bubble), and batch == 1 has nothing
to split. Overlapping micro-batches across ranks would also require
tagged p2p matching, which RCCL does not support (NCCL/RCCL send/recv
ignores tags) — concurrent per-thread send/recv streams would mismatch.
"""
from __future__ import annotations

import logging
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist
from torch import nn

from .dist import DistInfo
from .pipeline import BLOCK_LIST_NAMES, assign_block_ranges

log = logging.getLogger("parallelanything")


def _send_tensors(tensors, dst: int) -> None:
    for t in tensors:
        dist.send(t.contiguous(), dst=dst)


def _recv_like(tensors, src: int):
    out = []
    for t in tensors:
        buf = torch.empty_like(t)
        dist.recv(buf, src=src)
        out.append(buf)
    return out


class DistPipelineBlock(nn.Module):
    """Rank-owned block: run + hand off on the owner, identity elsewhere."""

    def __init__(self, block: nn.Module, index: int, owner: int,
                 prev_owner: Optional[int], next_owner: Optional[int],
                 is_last: bool, info: DistInfo):
        super().__init__()
        self.block = block
        self.index = index
        self.owner = owner
        self.prev_owner = prev_owner
        self.next_owner = next_owner
        self.is_last = is_last
        self.info = info

    def forward(self, *args, **kwargs):
        rank = self.info.rank
        # how many leading args are flowing hidden state: every tensor arg up
        # to the first non-tensor is part of the residual stream IF the block
        # returns the same arity; conservative: flow = what the block returns.
        n_flow = getattr(self.block, "_flow_arity", 1)
        flow = list(args[:n_flow])
        rest = args[n_flow:]

        if self.index == 0 and self.owner != 0:
            # a list's input is only guaranteed valid on rank 0 (inter-list
            # glue like FLUX's txt/img concat runs there on real data):
            # rank 0 feeds the first block's owner.
            if rank == 0:
                _send_tensors(flow, dst=self.owner)
            elif rank == self.owner:
                flow = _recv_like(flow, src=0)

        if rank == self.owner:
            if self.prev_owner is not None and self.prev_owner != rank:
                flow = _recv_like(flow, src=self.prev_owner)
            out = self.block(*flow, *rest, **kwargs)
            out_t = list(out) if isinstance(out, tuple) else [out]
            if self.is_last:
                if rank != 0:
                    _send_tensors(out_t, dst=0)
            elif self.next_owner is not None and self.next_owner != rank:
                _send_tensors(out_t, dst=self.next_owner)
            return out
        if self.is_last and rank == 0:
            # rank 0 needs the real final hidden state for the output layers
            flow = _recv_like(flow, src=self.owner)
            return tuple(flow) if n_flow > 1 else flow[0]
        # non-owner: identity pass-through keeps shapes for local replay
        return tuple(flow) if n_flow > 1 else flow[0]


def install_dist_pipeline(model: nn.Module, info: DistInfo,
                          weights: Optional[Sequence[float]] = None) -> int:
    """Wrap every recognized block list for rank-sharded execution.

    Returns the number of wrapped blocks. weights default to an even split
    over ranks. Call on EVERY rank with identical weights.
    """
    if weights is None:
        weights = [1.0 / info.world_size] * info.world_size
    wrapped = 0
    for list_name in BLOCK_LIST_NAMES:
        blocks = getattr(model, list_name, None)
        if not isinstance(blocks, nn.ModuleList) or len(blocks) == 0:
            continue
        owners = assign_block_ranges(len(blocks), weights)
        # probe flow arity from the block class (DoubleStreamBlock returns
        # (img, txt); single-stream/UNet-style blocks return one tensor)
        for idx in range(len(blocks)):
            blk = blocks[idx]
            n_flow = 2 if type(blk).__name__ == "DoubleStreamBlock" else 1
            blk._flow_arity = n_flow
            blocks[idx] = DistPipelineBlock(
                block=blk,
                index=idx,
                owner=owners[idx],
                prev_owner=owners[idx - 1] if idx > 0 else None,
                next_owner=owners[idx + 1] if idx + 1 < len(blocks) else None,
                is_last=(idx == len(blocks) - 1),
                info=info,
            )
            wrapped += 1
    if wrapped:
        log.info("dist pipeline: %d blocks sharded over %d ranks",
                 wrapped, info.world_size)
    return wrapped


def uninstall_dist_pipeline(model: nn.Module) -> int:
    n = 0
    for list_name in BLOCK_LIST_NAMES:
        blocks = getattr(model, list_name, None)
        if not isinstance(blocks, nn.ModuleList):
            continue
        for idx in range(len(blocks)):
            if isinstance(blocks[idx], DistPipelineBlock):
                blocks[idx] = blocks[idx].block
                n += 1
    return n
