"""Replication engine: full-model replicas per device.

Two paths, both replacing the reference's host-staged per-tensor clone
machinery (safe_model_clone / clone_module_simple,
any_device_parallel.py:390-722 — Path A re-instantiates the class and streams
the state dict key by key through host memory; Path B recursively clones
modules):

- In-process (ComfyUI node / single process, N devices): structural deepcopy
  with every parameter/buffer redirected to a direct device-to-device copy on
  the owning device — over xGMI for GPU->GPU, never staged through host.
  Same-device setups return the source module as the replica unless a copy is
  forced (skip-clone rule, any_device_parallel.py:594-597; LoRA forces the
  copy even on the original device, :1073-1081).

- Process-per-GPU (RCCL): every rank materializes the module locally and
  rank 0 broadcasts the weights as a handful of flat dtype-bucketed RCCL
  broadcasts over xGMI (SURVEY.md §2 component #3 disposition) — see
  broadcast_module below. Bucketing keeps launch count O(#dtypes * #buckets)
  instead of O(#params) (FLUX-class models have ~2000 parameter tensors).
"""
from __future__ import annotations

import copy
from typing import Iterable, List, Optional

import torch
import torch.distributed as dist
from torch import nn

from .fp8 import sanitize_param_dtype


def _iter_tensors(module: nn.Module):
    for p in module.parameters():
        yield p
    for b in module.buffers():
        yield b


# Device-bound runtime caches that must never be shared across replicas —
# the reference scrubbed an attr list on every clone (clear_flux_caches,
# any_device_parallel.py:166-195); our models keep per-device caches in
# dicts with these names, which are simply reset on the new replica so it
# repopulates on its own GPU.
CACHE_ATTRS = ("_pe_cache", "_mod_cache")


def clear_replica_caches(module: nn.Module) -> int:
    n = 0
    for sub in module.modules():
        for attr in CACHE_ATTRS:
            cache = getattr(sub, attr, None)
            if isinstance(cache, dict) and cache:
                cache.clear()
                n += 1
    return n


def replicate_module(
    src: nn.Module,
    device,
    force_copy: bool = False,
    non_blocking: bool = True,
) -> nn.Module:
    """Clone ``src`` onto ``device`` with direct (peer) copies.

    Parameters and buffers are copied straight from wherever they live to the
    target device; module structure and plain attributes are deep-copied.
    Replicas are inference-only: eval() + requires_grad_(False)
    (reference parity: any_device_parallel.py:710-712).
    """
    target = torch.device(device)
    src_dev = next(iter(_iter_tensors(src)), torch.empty(0)).device

    if src_dev == target and not force_copy:
        src.eval()
        return src

    memo: dict = {}
    for p in src.parameters():
        data = sanitize_param_dtype(p.data, target)
        new_p = nn.Parameter(
            data.to(target, non_blocking=non_blocking), requires_grad=False
        )
        memo[id(p)] = new_p
    for b in src.buffers():
        data = sanitize_param_dtype(b, target)
        memo[id(b)] = data.to(target, non_blocking=non_blocking)

    replica = copy.deepcopy(src, memo)
    clear_replica_caches(replica)
    replica.eval()
    for p in replica.parameters():
        p.requires_grad_(False)
    return replica


# ---------------------------------------------------------------------------
# Process-group path: flat dtype-bucketed broadcast (RCCL over xGMI).
# ---------------------------------------------------------------------------

def _bucketize(tensors: List[torch.Tensor], bucket_bytes: int):
    """Group same-dtype tensors into <= bucket_bytes chunks, order-stable."""
    buckets: List[List[torch.Tensor]] = []
    cur: List[torch.Tensor] = []
    cur_bytes = 0
    cur_dtype: Optional[torch.dtype] = None
    for t in tensors:
        nbytes = t.numel() * t.element_size()
        if cur and (t.dtype != cur_dtype or cur_bytes + nbytes > bucket_bytes):
            buckets.append(cur)
            cur, cur_bytes = [], 0
        cur.append(t)
        cur_dtype = t.dtype
        cur_bytes += nbytes
    if cur:
        buckets.append(cur)
    return buckets


@torch.no_grad()
def broadcast_module(
    module: nn.Module,
    src_rank: int = 0,
    group=None,
    bucket_bytes: int = 512 * 1024 * 1024,
) -> None:
    """Broadcast ``module``'s params+buffers from ``src_rank`` to all ranks.

    Flat dtype-bucketed broadcasts: each bucket is one contiguous tensor and
    one collective — sized so a FLUX-class bf16 model (~24 GB) moves in ~50
    broadcasts instead of ~2000 per-tensor ones. The 288 GB of HBM3E per GPU
    makes large staging buckets free; xGMI tree broadcast moves each at
    link rate. No-op when torch.distributed isn't initialized (single
    process) or world_size == 1.
    """
    if not dist.is_available() or not dist.is_initialized():
        return
    if dist.get_world_size(group) <= 1:
        return
    tensors = [t for t in _iter_tensors(module)]
    if not tensors:
        return
    device = tensors[0].device
    for bucket in _bucketize(tensors, bucket_bytes):
        flat = torch.empty(
            sum(t.numel() for t in bucket), dtype=bucket[0].dtype, device=device
        )
        my_rank = dist.get_rank(group)
        if my_rank == src_rank:
            offset = 0
            for t in bucket:
                flat[offset : offset + t.numel()].copy_(t.reshape(-1))
                offset += t.numel()
        dist.broadcast(flat, src=src_rank, group=group)
        if my_rank != src_rank:
            offset = 0
            for t in bucket:
                t.copy_(flat[offset : offset + t.numel()].view_as(t.reshape(-1)).reshape(t.shape))
                offset += t.numel()
