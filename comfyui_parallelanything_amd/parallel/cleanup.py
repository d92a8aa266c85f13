"""Lifecycle teardown: finalizer-driven release of parallel state.

Reference behavior (SURVEY.md §2 component #11, any_device_parallel.py):
- ``weakref.finalize(model, cleanup_parallel_model, ref)`` registered at
  setup (:1459); teardown restores ``_original_forward`` (:224-229), frees
  replicas (:231-248), deletes state attrs (:250-258), optionally unloads
  models and purges per-device caches (:260-282); ``aggressive_cleanup`` is
  gc + sync + empty_cache on every device (:197-209).

MI355X version: replicas are dropped (HBM returned to the caching
allocator), pipeline block wrappers are unwrapped from the lead replica,
streams are released; there is no CPU eviction step — the reference CPUs
replicas to make room on small-VRAM cards, which 288 GB of HBM3E does not
need and which would stage the model through host memory for nothing.
"""
from __future__ import annotations

import gc
import logging
import weakref

import torch
from torch import nn

from .engine import uninstall_parallel_forward
from .pipeline import BLOCK_LIST_NAMES, ParallelBlock

log = logging.getLogger("parallelanything")


def aggressive_cleanup() -> None:
    """gc + sync + empty_cache on every HIP device
    (reference aggressive_cleanup, :197-209)."""
    gc.collect()
    if torch.cuda.is_available():
        for i in range(torch.cuda.device_count()):
            with torch.cuda.device(i):
                torch.cuda.synchronize()
                torch.cuda.empty_cache()


def unwrap_pipeline_blocks(module: nn.Module) -> int:
    """Restore original blocks where ParallelBlock wrappers were installed
    (the lead replica may alias the user's model)."""
    n = 0
    for list_name in BLOCK_LIST_NAMES:
        blocks = getattr(module, list_name, None)
        if not isinstance(blocks, nn.ModuleList):
            continue
        for idx in range(len(blocks)):
            if isinstance(blocks[idx], ParallelBlock):
                blocks[idx] = blocks[idx].local_block
                n += 1
    return n


def cleanup_parallel_model(model_ref) -> None:
    """Teardown entry; accepts a weakref or the model itself
    (reference cleanup_parallel_model, :211-282)."""
    model = model_ref() if isinstance(model_ref, weakref.ref) else model_ref
    if model is None:
        return
    if not getattr(model, "_true_parallel_active", False):
        return
    purge_cache = getattr(model, "_parallel_purge_cache", True)
    engine = getattr(model, "_parallel_engine", None)
    try:
        if engine is not None:
            for replica in list(engine.replicas.values()):
                unwrap_pipeline_blocks(replica)
            engine.release()
        unwrap_pipeline_blocks(model)
    finally:
        uninstall_parallel_forward(model)
        for attr in ("_parallel_purge_cache", "_parallel_purge_models"):
            if hasattr(model, attr):
                try:
                    delattr(model, attr)
                except AttributeError:
                    pass
    if purge_cache:
        aggressive_cleanup()
    log.info("parallel state released")


def register_finalizer(owner, model) -> weakref.finalize:
    """GC-driven cleanup like the reference's weakref.finalize (:1459)."""
    return weakref.finalize(owner, cleanup_parallel_model, weakref.ref(model))
