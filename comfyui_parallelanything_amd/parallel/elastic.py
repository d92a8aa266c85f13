"""Failure detection / degradation for process-group (RCCL) mode.

The reference's "elasticity" is inference-grade degradation inside one
process: drop an OOM device and renormalize, fall back to lead-only on a
runtime failure (SURVEY.md §5 Failure detection). The process-per-GPU
equivalent here:

- step_with_fallback: runs a distributed step; on a collective/peer failure
  it tears the process group down (RCCL comms destroyed) and re-runs the
  step locally on this rank's full batch — the lead rank keeps serving.
- A watchdog timeout on collectives comes from init_distributed(timeout_s):
  RCCL aborts the hanging collective after the timeout and raises, which
  routes into the same fallback.
"""
from __future__ import annotations

import logging
from typing import Callable

import torch.distributed as dist

from .dist import DistInfo

log = logging.getLogger("parallelanything")


def abort_to_local(info: DistInfo) -> DistInfo:
    """Destroy the process group and continue single-rank.

    After a peer failure the surviving rank cannot rebuild an N-way RCCL
    comm without a rendezvous with the dead peer; the inference-serving
    answer (mirroring the reference's lead-only fallback,
    any_device_parallel.py:1435-1446) is to degrade to local execution.
    """
    if dist.is_initialized():
        try:
            dist.destroy_process_group()
        except Exception:  # noqa: BLE001
            log.exception("destroy_process_group failed during degradation")
    return DistInfo(
        rank=0,
        world_size=1,
        local_rank=info.local_rank,
        device=info.device,
        backend=info.backend,
    )


def step_with_fallback(
    dist_step: Callable[[DistInfo], object],
    local_step: Callable[[], object],
    info: DistInfo,
) -> tuple:
    """Run one distributed step; degrade to local on collective failure.

    Returns (result, info) — info becomes single-rank after a degradation,
    and subsequent calls run local_step directly.
    """
    if info.world_size <= 1:
        return local_step(), info
    try:
        return dist_step(info), info
    except Exception as err:  # noqa: BLE001
        log.error("distributed step failed (%r); degrading to local", err)
        info = abort_to_local(info)
        return local_step(), info
