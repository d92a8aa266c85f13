"""Tracing / metrics: roctx ranges and per-step timing.

The reference's only instrumentation is [ParallelAnything]-prefixed prints
(SURVEY.md §5 Tracing). Here:
- roctx range annotations around scatter/forward/gather so rocprofv3
  --sys-trace shows the engine phases next to the kernel trace (torch's
  nvtx API emits roctx markers on ROCm builds);
- StepTimer: wall sec/it + images/sec, the headline metric;
- per-GPU busy time via HIP events when requested.
"""
from __future__ import annotations

import contextlib
import json
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

_HAVE_NVTX = hasattr(torch.cuda, "nvtx")


@contextlib.contextmanager
def trace_range(name: str):
    """roctx range (no-op off-GPU)."""
    if _HAVE_NVTX and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


@dataclass
class StepStats:
    wall_s: float
    images: int

    @property
    def images_per_s(self) -> float:
        return self.images / self.wall_s if self.wall_s > 0 else 0.0


@dataclass
class StepTimer:
    """Collects per-step wall times; optionally per-device busy time."""

    devices: List[str] = field(default_factory=list)
    steps: List[StepStats] = field(default_factory=list)
    _t0: Optional[float] = None
    _events: Dict[str, tuple] = field(default_factory=dict)

    def start(self, measure_gpu: bool = False):
        if measure_gpu:
            for d in self.devices:
                if torch.device(d).type == "cuda":
                    e0 = torch.cuda.Event(enable_timing=True)
                    e0.record(torch.cuda.current_stream(torch.device(d)))
                    self._events[d] = (e0, None)
        self._t0 = time.perf_counter()

    def stop(self, images: int) -> StepStats:
        for d, (e0, _) in list(self._events.items()):
            e1 = torch.cuda.Event(enable_timing=True)
            e1.record(torch.cuda.current_stream(torch.device(d)))
            self._events[d] = (e0, e1)
        s = StepStats(time.perf_counter() - (self._t0 or 0.0), images)
        self.steps.append(s)
        return s

    def gpu_busy_ms(self) -> Dict[str, float]:
        out = {}
        for d, (e0, e1) in self._events.items():
            if e1 is not None:
                e1.synchronize()
                out[d] = e0.elapsed_time(e1)
        return out

    def summary(self) -> dict:
        if not self.steps:
            return {}
        walls = [s.wall_s for s in self.steps]
        total_imgs = sum(s.images for s in self.steps)
        total_wall = sum(walls)
        out = {
            "steps": len(self.steps),
            "sec_per_it": total_wall / len(self.steps),
            "images_per_s": total_imgs / total_wall if total_wall else 0.0,
            "min_step_s": min(walls),
            "max_step_s": max(walls),
        }
        if len(self.steps) > 1:
            # steady state excludes step 0 (cold caches: MIOpen algo
            # search, RoPE tables, allocator) — the headline-comparable
            # number for a denoise loop
            sw = walls[1:]
            si = sum(s.images for s in self.steps[1:])
            out["steady_sec_per_it"] = sum(sw) / len(sw)
            out["steady_images_per_s"] = si / sum(sw) if sum(sw) else 0.0
        return out

    def dump(self, path: str) -> None:
        with open(path, "w") as f:
            json.dump(self.summary(), f, indent=2)
