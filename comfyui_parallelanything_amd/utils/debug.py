"""Debug / race-hunting modes (SURVEY.md §5 race detection disposition).

The engine's thread-safety is by construction (one launcher thread, one
stream per GPU, results slotted by index); the sanitizer-equivalent on ROCm
is kernel serialization plus deterministic-split golden tests:

- PA_DEBUG_SERIALIZE=1 (call apply_debug_env() early, before the first HIP
  call): AMD_SERIALIZE_KERNEL=3 serializes every kernel launch and copy,
  HIP_LAUNCH_BLOCKING=1 makes launches synchronous — any async-ordering bug
  becomes deterministic and attributable to the faulting kernel.
- PA_DEBUG_DETERMINISTIC=1: torch deterministic algorithms + disabled TF32
  (no-op on gfx950 — no TF32 — but keeps CPU comparisons strict).
"""
from __future__ import annotations

import logging
import os

log = logging.getLogger("parallelanything")


def apply_debug_env() -> dict:
    applied = {}
    if os.environ.get("PA_DEBUG_SERIALIZE", "0") == "1":
        os.environ["AMD_SERIALIZE_KERNEL"] = "3"
        os.environ["AMD_SERIALIZE_COPY"] = "3"
        os.environ["HIP_LAUNCH_BLOCKING"] = "1"
        applied["serialize"] = True
        log.warning("PA_DEBUG_SERIALIZE: all HIP launches serialized")
    if os.environ.get("PA_DEBUG_DETERMINISTIC", "0") == "1":
        import torch

        torch.use_deterministic_algorithms(True, warn_only=True)
        applied["deterministic"] = True
    return applied
