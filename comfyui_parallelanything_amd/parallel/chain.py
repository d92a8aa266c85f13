"""DEVICE_CHAIN schema, device enumeration and weight normalization.

Behavioral contract from the reference (cited so parity can be checked):
- DEVICE_CHAIN is a list[dict] with keys "device", "percentage", "weight"
  (any_device_parallel.py:823-832, :872-882).
- Chain nodes append configs; the list node drops zero-percent slots
  (any_device_parallel.py:875-876).
- setup normalizes weights so percentages need not sum to 100
  (any_device_parallel.py:1019-1027).

On ROCm, HIP devices enumerate through torch.cuda.* (device strings "cuda:N"
map to MI355X GPUs); "cpu" is a first-class chain device and is the
no-GPU test backend, as in the reference (any_device_parallel.py:771).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Sequence

import torch


def available_devices() -> List[str]:
    """Enumerate chain-eligible devices: cpu first, then each HIP GPU.

    Mirrors ParallelDevice.get_available_devices (any_device_parallel.py:769-786)
    minus the NVIDIA-era mps/xpu/directml branches: on an MI355X node the only
    compute devices are "cpu" and "cuda:N" (HIP).
    """
    devices = ["cpu"]
    if torch.cuda.is_available():
        devices += [f"cuda:{i}" for i in range(torch.cuda.device_count())]
    return devices


def make_entry(device: str, percentage: float) -> Dict[str, float]:
    """One DEVICE_CHAIN entry with the reference's exact dict schema."""
    return {
        "device": str(device),
        "percentage": float(percentage),
        "weight": float(percentage) / 100.0,
    }


def chain_append(prev: Sequence[dict] | None, device: str, percentage: float) -> List[dict]:
    """Chainable add (ParallelDevice.add_device, any_device_parallel.py:819-832)."""
    new_chain = list(prev) if prev else []
    new_chain.append(make_entry(device, percentage))
    return new_chain


def chain_from_slots(slots: Sequence[tuple]) -> List[dict]:
    """4-slot list-node build; zero-percent slots dropped
    (ParallelDeviceList.create_list, any_device_parallel.py:872-882)."""
    return [make_entry(dev, pct) for dev, pct in slots if pct and pct > 0]


def normalize_weights(chain: Sequence[dict]) -> List[float]:
    """Renormalize chain percentages to weights summing to 1.

    Reference semantics (any_device_parallel.py:1019-1027): percentages need
    not sum to 100; weights are pct_i / sum(pct). A chain whose percentages
    sum to zero falls back to an even split.
    """
    pcts = [float(e.get("percentage", 0.0)) for e in chain]
    total = sum(pcts)
    if total <= 0:
        n = max(1, len(chain))
        return [1.0 / n] * n
    return [p / total for p in pcts]


@dataclass(frozen=True)
class DeviceChain:
    """Validated, normalized view of a DEVICE_CHAIN list."""

    devices: tuple  # device strings in chain order; index 0 is the lead
    weights: tuple  # normalized, sum == 1.0

    @classmethod
    def from_list(cls, chain: Sequence[dict]) -> "DeviceChain":
        if not chain:
            raise ValueError("empty DEVICE_CHAIN")
        devices = tuple(str(e["device"]) for e in chain)
        for d in devices:
            torch.device(d)  # raises on malformed device strings
        return cls(devices=devices, weights=tuple(normalize_weights(chain)))

    @property
    def lead(self) -> str:
        return self.devices[0]

    def __len__(self) -> int:
        return len(self.devices)

    def drop(self, index: int) -> "DeviceChain":
        """Remove one device and renormalize over the survivors — the OOM
        degradation step (any_device_parallel.py:1114-1128)."""
        if len(self.devices) <= 1:
            raise ValueError("cannot drop the last device in the chain")
        devices = tuple(d for i, d in enumerate(self.devices) if i != index)
        kept = [w for i, w in enumerate(self.weights) if i != index]
        total = sum(kept)
        if total <= 0:
            weights = tuple(1.0 / len(devices) for _ in devices)
        else:
            weights = tuple(w / total for w in kept)
        return DeviceChain(devices=devices, weights=weights)
