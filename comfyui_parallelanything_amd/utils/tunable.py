"""hipBLASLt algorithm selection via PyTorch TunableOp.

GEMMs are 61% of the flagship step (profiles/r02_final_kernel_stats.csv).
A one-time offline tuning pass (scripts: PYTORCH_TUNABLEOP_TUNING=1 over
bench.py) picks the best hipblaslt algorithm per GEMM shape; loading the
committed results is a measured -1.0% on the flagship step (537.8/536.1
baseline bracket vs 531.8 tuned, same box, interleaved A/B/A).

TunableOp validates PT/HIP/hipBLASLt/arch versions inside the CSV and
silently ignores stale entries, so loading is always safe. Disable with
PA_NO_TUNABLEOP=1. New shapes not in the file run the default algorithm
(tuning stays OFF at run time — no warmup cost, deterministic).
"""
from __future__ import annotations

import logging
import os

import torch

log = logging.getLogger("parallelanything")

_DATA = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                     "data", "tunableop_gfx950.csv")
_loaded = False


def enable_tuned_gemms(path: str | None = None) -> bool:
    """Load pre-tuned hipBLASLt algorithm selections (idempotent)."""
    global _loaded
    if _loaded:
        return True
    if os.environ.get("PA_NO_TUNABLEOP") == "1":
        return False
    p = path or _DATA
    if not torch.cuda.is_available() or not os.path.exists(p):
        return False
    try:
        torch.cuda.tunable.enable(True)
        torch.cuda.tunable.tuning_enable(False)  # select only, never search
        torch.cuda.tunable.read_file(p)
        _loaded = True
        log.info("TunableOp: loaded tuned GEMM algorithms from %s", p)
        return True
    except Exception as err:  # noqa: BLE001
        log.warning("TunableOp load failed (%r); using defaults", err)
        return False
