"""ComfyUI-ParallelAnything for AMD Instinct MI355X (gfx950, CDNA4).

A brand-new MI355X-native multi-GPU diffusion batch-parallel engine with the
capabilities of the reference ComfyUI-ParallelAnything custom node
(reference: /root/reference/any_device_parallel.py), re-designed MI355X-first:

- PyTorch-ROCm host framework; hand-written HIP/CDNA4 (gfx950) kernels for the
  per-step model math (attention, fused norms, RoPE, timestep embedding).
- RCCL over xGMI (torch.distributed, backend "nccl" == RCCL on ROCm) for
  process-per-GPU scale-out; in-process multi-device mode with per-GPU HIP
  streams + events for the ComfyUI node path.
- Single launcher thread, no GIL-bound thread pool, events instead of
  full-device synchronize (reference serialized with torch.cuda.synchronize
  around every per-device forward: any_device_parallel.py:1385-1397).

Public ComfyUI surface matches the reference (any_device_parallel.py:1473-1483):
NODE_CLASS_MAPPINGS / NODE_DISPLAY_NAME_MAPPINGS with ParallelDevice,
ParallelDeviceList, ParallelAnything, and the DEVICE_CHAIN link type
(list[{"device", "percentage", "weight"}]).
"""

__version__ = "1.0.0"

from .nodes import (  # noqa: F401
    NODE_CLASS_MAPPINGS,
    NODE_DISPLAY_NAME_MAPPINGS,
    ParallelAnything,
    ParallelDevice,
    ParallelDeviceList,
)

__all__ = [
    "NODE_CLASS_MAPPINGS",
    "NODE_DISPLAY_NAME_MAPPINGS",
    "ParallelAnything",
    "ParallelDevice",
    "ParallelDeviceList",
    "__version__",
]
