"""Batch==1 pipeline (layer-sharded) mode.

Reference behavior being reproduced (SURVEY.md §2 component #12):

- thread-local activation flag (any_device_parallel.py:16-22) — the lead
  replica's own forward drives control flow; only wrapped block bodies
  execute on peer devices (:1295-1305).
- block lists scanned on the lead replica: double_blocks, single_blocks,
  transformer_blocks, layers (:1156); contiguous ranges sized
  ``round(weight * num_blocks)`` with the remainder on the last device
  (:1168-1178).
- at forward with the flag set, ALL args/kwargs are moved to the owner
  device unconditionally — auxiliary tensors (timesteps, modulation vectors)
  may lag on the lead device even when the hidden states already arrived
  from the previous stage (the reference's "CRITICAL FIX" invariant,
  :72-78). The last block returns to the lead device (:83-86).

MI355X notes: stage hand-offs are direct peer copies over xGMI links
(single-hop, ~153 GB/s/link) — a [1, seq, hidden] activation is latency-, not
bandwidth-bound. In process-group mode the equivalent hand-off is an RCCL
p2p send/recv (see dist.py); in-process mode (this file) uses stream-ordered
peer DMA via tensor.to(non_blocking=True) on the owner's stream.
"""
from __future__ import annotations

import copy
import threading
from dataclasses import fields, is_dataclass
from typing import Dict, List

import torch
from torch import nn

_pipeline_state = threading.local()

BLOCK_LIST_NAMES = ("double_blocks", "single_blocks", "transformer_blocks", "layers")


def pipeline_mode_active() -> bool:
    return getattr(_pipeline_state, "active", False)


def set_pipeline_mode(active: bool) -> None:
    _pipeline_state.active = bool(active)


def _move(x, device: torch.device):
    """Recursive move incl. dataclasses (reference _move_tensor, :40-57)."""
    if isinstance(x, torch.Tensor):
        return x.to(device, non_blocking=True) if x.device != device else x
    if isinstance(x, (list, tuple)):
        return type(x)(_move(v, device) for v in x)
    if isinstance(x, dict):
        return {k: _move(v, device) for k, v in x.items()}
    if is_dataclass(x) and not isinstance(x, type):
        new = copy.copy(x)
        for f in fields(x):
            setattr(new, f.name, _move(getattr(x, f.name), device))
        return new
    return x


class ParallelBlock(nn.Module):
    """Dual-mode block wrapper: local in DP mode, owner-device in pipeline
    mode (reference ParallelBlock, any_device_parallel.py:24-87)."""

    def __init__(
        self,
        local_block: nn.Module,
        block_idx: int,
        owner_device: torch.device,
        peers: Dict[str, nn.Module],
        is_last_block: bool,
        lead_device: torch.device,
    ):
        super().__init__()
        self.local_block = local_block
        self.block_idx = block_idx
        self.owner_device = owner_device
        self.peers = peers
        self.is_last_block = is_last_block
        self.lead_device = lead_device
        self.owner_block = peers.get(str(owner_device), local_block)

    def forward(self, *args, **kwargs):
        if not pipeline_mode_active():
            return self.local_block(*args, **kwargs)
        dev = self.owner_device
        # ALL inputs move unconditionally: hidden states may already sit on
        # `dev` while aux inputs lag on the lead device (reference :72-78).
        args = tuple(_move(a, dev) for a in args)
        kwargs = {k: _move(v, dev) for k, v in kwargs.items()}
        out = self.owner_block(*args, **kwargs)
        if self.is_last_block:
            out = _move(out, self.lead_device)
        return out


def assign_block_ranges(num_blocks: int, weights) -> List[int]:
    """Contiguous owner index per block (reference :1168-1178):
    round(weight*n) blocks per device in chain order, remainder to last."""
    owners: List[int] = []
    current = 0
    n_dev = len(weights)
    for i, w in enumerate(weights):
        count = int(round(w * num_blocks))
        if i == n_dev - 1:
            count = num_blocks - current
        for _ in range(count):
            if current < num_blocks:
                owners.append(i)
                current += 1
    while len(owners) < num_blocks:  # all-zero rounding edge
        owners.append(n_dev - 1)
    return owners


class PipelineConfig:
    """Wires ParallelBlock wrappers into the lead replica and exposes the
    batch==1 forward (engine routes here; reference :1295-1305).

    ``microbatches`` > 1 enables GPipe-style micro-batching (NOT in the
    reference — SURVEY.md §6 flags it as an optional extension): a batch
    B > 1 that is too small for DP is split into ``min(microbatches, B)``
    micro-batches, each driven through the block-sharded pipeline by its
    own host thread. Per device, micro-batch work serializes on that
    device's current stream (correct stage ordering for free); ACROSS
    devices the stages overlap, so stage 1 processes micro-batch m while
    stage 0 runs m+1 — classic fill/drain. Inference-only (no_grad),
    stateless block bodies, so concurrent threads are safe."""

    def __init__(self, engine, microbatches: int = 1):
        self.engine = engine
        self.microbatches = max(1, int(microbatches))
        self.configured = False
        self._wire()

    def _wire(self) -> None:
        eng = self.engine
        lead_name = eng.chain.lead
        lead_replica = eng.replicas[lead_name]
        lead_dev = torch.device(lead_name)
        for list_name in BLOCK_LIST_NAMES:
            blocks = getattr(lead_replica, list_name, None)
            if not isinstance(blocks, nn.ModuleList) or len(blocks) == 0:
                continue
            owners = assign_block_ranges(len(blocks), eng.chain.weights)
            for idx in range(len(blocks)):
                owner_name = eng.chain.devices[owners[idx]]
                peers = {
                    str(torch.device(d)): getattr(r, list_name)[idx]
                    for d, r in eng.replicas.items()
                    if hasattr(r, list_name)
                }
                blocks[idx] = ParallelBlock(
                    local_block=blocks[idx],
                    block_idx=idx,
                    owner_device=torch.device(owner_name),
                    peers=peers,
                    is_last_block=(idx == len(blocks) - 1),
                    lead_device=lead_dev,
                )
            self.configured = True

    @torch.no_grad()
    def forward(self, x, timesteps, context=None, **kwargs):
        from .split import (
            concatenate_results,
            get_batch_size,
            split_batch,
            split_kwargs,
        )

        eng = self.engine
        batch = get_batch_size(x)
        n_mb = min(self.microbatches, batch)
        if n_mb <= 1:
            set_pipeline_mode(True)
            try:
                return eng._lead_only(x, timesteps, context, **kwargs)
            finally:
                set_pipeline_mode(False)

        sizes = [batch // n_mb + (1 if i < batch % n_mb else 0)
                 for i in range(n_mb)]
        x_mb = split_batch(x, sizes)
        t_mb = split_batch(timesteps, sizes)
        c_mb = split_batch(context, sizes) if context is not None else None
        kw_mb = split_kwargs(kwargs, sizes, batch)

        results: List = [None] * n_mb
        errors: List = []

        def run(i: int) -> None:
            set_pipeline_mode(True)  # flag is thread-local: set per thread
            try:
                with torch.no_grad():
                    results[i] = eng._lead_only(
                        x_mb[i], t_mb[i],
                        c_mb[i] if c_mb is not None else None,
                        **kw_mb[i],
                    )
            except BaseException as err:  # noqa: BLE001
                errors.append(err)
            finally:
                set_pipeline_mode(False)

        threads = [threading.Thread(target=run, args=(i,), daemon=True)
                   for i in range(n_mb)]
        for t in threads:  # launch in order: micro-batch m fills stage 0 first
            t.start()
        for t in threads:
            t.join()
        if errors:
            raise errors[0]
        return concatenate_results(results, dim=0)


def configure_pipeline(engine, microbatches: int = 1) -> None:
    """Attach pipeline mode to an engine when >1 device and the model
    exposes a recognized block list; silently a no-op otherwise."""
    if len(engine.chain.devices) < 2:
        return
    cfg = PipelineConfig(engine, microbatches=microbatches)
    if cfg.configured:
        engine.pipeline = cfg
