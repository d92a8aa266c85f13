"""Dynamic load balancing: VRAM-blended split weights.

Policy from the reference (any_device_parallel.py:737-766): when any chain
device is a GPU, blend 0.7 * user_weight + 0.3 * free_vram_share and
renormalize before computing split sizes; chains with no GPU (or no
readable VRAM) use the static weights unchanged.

MI355X notes: free HBM is read with torch.cuda.mem_get_info (hipMemGetInfo
underneath) instead of total_memory - memory_allocated — the reference's
formula (any_device_parallel.py:728-732) ignores other processes and the
caching allocator; mem_get_info is the truthful number on ROCm.
"""
from __future__ import annotations

from typing import List, Sequence

import torch

from .split import compute_split_sizes


def get_free_vram_mb(device_name: str) -> float:
    """Free HBM in MiB for a cuda device string; 0 for cpu/unreadable."""
    try:
        if str(device_name).startswith("cuda"):
            idx = torch.device(device_name).index or 0
            free_b, _total_b = torch.cuda.mem_get_info(idx)
            return free_b / (1024.0**2)
    except Exception:
        pass
    return 0.0


def vram_blended_weights(
    devices: Sequence[str], weights: Sequence[float]
) -> List[float]:
    """0.7*user + 0.3*vram_share blend, renormalized
    (any_device_parallel.py:751-763)."""
    vram = [get_free_vram_mb(d) if str(d).startswith("cuda") else 0.0 for d in devices]
    total_vram = sum(vram)
    if total_vram <= 0:
        return list(weights)
    adjusted = [
        0.7 * w + 0.3 * (v / total_vram) if v > 0 else w
        for w, v in zip(weights, vram)
    ]
    total = sum(adjusted)
    return [a / total for a in adjusted]


def auto_split_batch(
    batch_size: int, devices: Sequence[str], weights: Sequence[float]
) -> List[int]:
    """VRAM-aware split sizes (any_device_parallel.py:737-766)."""
    if not any(str(d).startswith("cuda") for d in devices):
        return compute_split_sizes(batch_size, weights)
    return compute_split_sizes(batch_size, vram_blended_weights(devices, weights))


class AdaptiveBalancer:
    """Per-step timing feedback on top of the static weights.

    The reference's own limitation list calls out "Static load balancing -
    Percentages fixed per run" (reference README.md); this closes it: after
    each DP step, per-device busy times update an EMA of throughput
    (samples/sec), and the effective weights blend user weights with the
    measured throughput share. A device that keeps finishing late sheds load.
    """

    def __init__(self, devices: Sequence[str], user_weights: Sequence[float],
                 blend: float = 0.5, ema: float = 0.5):
        self.devices = list(devices)
        self.user_weights = list(user_weights)
        self.blend = blend
        self.ema = ema
        self.throughput = {d: None for d in self.devices}

    def record(self, device: str, chunk_size: int, seconds: float) -> None:
        if seconds <= 0 or chunk_size <= 0:
            return
        tput = chunk_size / seconds
        prev = self.throughput.get(device)
        self.throughput[device] = (
            tput if prev is None else self.ema * tput + (1 - self.ema) * prev
        )

    def weights(self) -> List[float]:
        tputs = [self.throughput.get(d) for d in self.devices]
        if any(t is None for t in tputs):
            return list(self.user_weights)
        total_t = sum(tputs)
        if total_t <= 0:
            return list(self.user_weights)
        blended = [
            (1 - self.blend) * w + self.blend * (t / total_t)
            for w, t in zip(self.user_weights, tputs)
        ]
        total = sum(blended)
        return [b / total for b in blended]

    def split(self, batch_size: int) -> List[int]:
        return compute_split_sizes(batch_size, self.weights())
