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
import logging
from typing import List, Optional

import torch
import torch.distributed as dist
from torch import nn

from .fp8 import sanitize_param_dtype

log = logging.getLogger("parallelanything")


def _iter_tensors(module: nn.Module):
    for p in module.parameters():
        yield p
    for b in module.buffers():
        yield b


# Device-bound runtime caches that must never be shared across replicas.
# In-tree models keep per-device caches in dicts with these names; they are
# reset on the new replica so it repopulates on its own GPU.
CACHE_ATTRS = ("_pe_cache", "_mod_cache")

# Foreign-model cache attr names, matching the reference's scrub list
# (clear_flux_caches, any_device_parallel.py:166-195, list at :167-174):
# RoPE freq tables, FLUX img/txt position ids, positional embeddings,
# kv caches, attention bias, video/temporal ids. Unlike the reference —
# which nulls ANY attr with these names, including registered nn.Parameters
# (a learned ViT-style ``pos_embed`` would be destroyed) — only plain
# ``__dict__`` tensor attrs are scrubbed: registered params/buffers are
# real weights and are device-copied by the clone itself.
FOREIGN_CACHE_ATTRS = (
    "img_ids", "txt_ids", "_img_ids", "_txt_ids",
    "cached_img_ids", "cached_txt_ids",
    "pos_emb", "_pos_emb", "pos_embed", "_pos_embed", "cached_pos_emb",
    "rope", "_rope", "freqs_cis", "_freqs_cis", "freqs", "_freqs",
    "cache", "_cache", "kv_cache", "_kv_cache", "attn_bias", "_attn_bias",
    "rope_cache", "_rope_cache", "freqs_cis_cache", "_freqs_cis_cache",
    "temporal_ids", "frame_ids", "video_ids", "temp_pos_emb",
)


def _is_tensorish(v) -> bool:
    if isinstance(v, torch.Tensor):
        return True
    if isinstance(v, (list, tuple)) and v:
        return all(isinstance(t, torch.Tensor) for t in v)
    return False


def clear_replica_caches(module: nn.Module) -> int:
    """Reset device-bound runtime caches on a fresh replica.

    - in-tree dict caches (CACHE_ATTRS) are cleared in place;
    - foreign-model tensor caches (FOREIGN_CACHE_ATTRS) held as plain
      instance attributes are set to None so the replica recomputes them
      on its own device — never a registered Parameter or buffer.
    """
    n = 0
    for sub in module.modules():
        for attr in CACHE_ATTRS:
            cache = getattr(sub, attr, None)
            if isinstance(cache, dict) and cache:
                cache.clear()
                n += 1
        d = sub.__dict__
        for attr in FOREIGN_CACHE_ATTRS:
            if attr in d and attr not in sub._parameters and \
                    attr not in sub._buffers:
                v = d[attr]
                if _is_tensorish(v):
                    d[attr] = None
                    n += 1
                elif isinstance(v, dict) and v and all(
                    _is_tensorish(t) for t in v.values()
                ):
                    v.clear()
                    n += 1
    return n


def replicate_module(
    src: nn.Module,
    device,
    force_copy: bool = False,
    non_blocking: bool = True,
) -> nn.Module:
    """Clone ``src`` onto ``device`` with direct (peer) copies.

    Two-strategy ladder matching the reference's safe_model_clone
    (any_device_parallel.py:586-722):

    1. structural deepcopy with every parameter/buffer redirected (via the
       deepcopy memo) to a direct device-to-device copy — fast, preserves
       arbitrary Python attributes;
    2. on any deepcopy failure (foreign modules holding non-picklable
       attrs: locks, file handles, C handles…), a recursive structural
       clone (reference clone_module_simple, :390-584): params/buffers
       device-copied, submodules rebuilt, plain attrs best-effort copied
       and shared when uncopyable.

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
            data.to(target, non_blocking=non_blocking, copy=True),
            requires_grad=False
        )
        memo[id(p)] = new_p
    for b in src.buffers():
        data = sanitize_param_dtype(b, target)
        memo[id(b)] = data.to(target, non_blocking=non_blocking, copy=True)

    try:
        # deepcopy gets its OWN memo copy: a failed deepcopy leaves
        # partially-constructed objects in its memo, which must never leak
        # into the structural fallback below.
        replica = copy.deepcopy(src, dict(memo))
    except Exception as err:  # noqa: BLE001
        log.warning(
            "deepcopy clone of %s failed (%r); falling back to structural "
            "clone", type(src).__name__, err,
        )
        replica = _structural_clone(src, target, dict(memo), non_blocking)
    clear_replica_caches(replica)
    replica.eval()
    for p in replica.parameters():
        p.requires_grad_(False)
    return replica


# nn.Module machinery attrs that get FRESH instances from nn.Module.__init__
# in the structural clone — sharing hook dicts / state across replicas would
# couple them (reference clone_module_simple rebuilds these too, :506-584).
_MODULE_MACHINERY = frozenset(nn.Module().__dict__.keys())


def _clone_attr(v, target: torch.device, memo: dict, non_blocking: bool):
    """Best-effort attribute clone for the structural fallback path."""
    if v is None or isinstance(v, (bool, int, float, str, bytes)):
        return v
    if id(v) in memo:
        return memo[id(v)]
    if isinstance(v, torch.Tensor):
        # copy=True: .to() is a no-op for same-device force-copies (the
        # LoRA clone-even-lead rule) and the replica must never alias
        # source storage
        out = v.to(target, non_blocking=non_blocking, copy=True)
        memo[id(v)] = out
        return out
    if isinstance(v, nn.Module):
        return _structural_clone(v, target, memo, non_blocking)
    try:
        return copy.deepcopy(v, memo)
    except Exception:  # noqa: BLE001
        # uncopyable (lock, handle, closure…): SHARE the object — same
        # best-effort the reference takes in its generic path (:559-560);
        # device-bound tensors never reach here (handled above).
        return v


def _structural_clone(
    src: nn.Module, target: torch.device, memo: dict, non_blocking: bool
) -> nn.Module:
    """Recursive structural clone (reference clone_module_simple contract,
    any_device_parallel.py:390-584): new instance via __new__, fresh
    nn.Module machinery, params/buffers from the device-copy memo,
    submodules recursed, remaining ``__dict__`` attrs best-effort."""
    if id(src) in memo:
        return memo[id(src)]
    new = src.__class__.__new__(src.__class__)
    nn.Module.__init__(new)
    memo[id(src)] = new

    for name, p in src._parameters.items():
        if p is None:
            new._parameters[name] = None
        elif id(p) in memo:
            new._parameters[name] = memo[id(p)]
        else:
            data = sanitize_param_dtype(p.data, target)
            np_ = nn.Parameter(
                data.to(target, non_blocking=non_blocking, copy=True),
                requires_grad=False,
            )
            memo[id(p)] = np_
            new._parameters[name] = np_
    for name, b in src._buffers.items():
        if b is None:
            new._buffers[name] = None
        elif id(b) in memo:
            new._buffers[name] = memo[id(b)]
        else:
            data = sanitize_param_dtype(b, target)
            nb = data.to(target, non_blocking=non_blocking, copy=True)
            memo[id(b)] = nb
            new._buffers[name] = nb
    new._non_persistent_buffers_set = set(src._non_persistent_buffers_set)
    for name, m in src._modules.items():
        new._modules[name] = (
            None if m is None else _structural_clone(m, target, memo,
                                                     non_blocking)
        )
    new.training = src.training
    for k, v in src.__dict__.items():
        if k in _MODULE_MACHINERY or k in (
            "_parameters", "_buffers", "_modules",
            "_non_persistent_buffers_set", "training",
        ):
            continue
        object.__setattr__(new, k, _clone_attr(v, target, memo, non_blocking))
    return new


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
