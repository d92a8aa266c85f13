"""FP8 policy for gfx950 (CDNA4).

The reference downcasts fp8 weights/activations to fp16 on any device whose
compute capability is below (9,0) (any_device_parallel.py:93-124, :654-655,
:688-699, :1243-1244). On MI355X the policy inverts: gfx950 has native fp8
MFMA at ~2x the bf16 rate, so fp8 tensors are KEPT fp8 and fed to the fp8
MFMA paths. The one real hazard on CDNA4 is the FORMAT: gfx950 consumes OCP
e4m3fn / e5m2; MI300X-era fnuz encodings are incompatible and must be
re-encoded before use (MI355X_MICROARCH.md §Matrix cores; guide §4).
"""
from __future__ import annotations

import torch

_FP8_DTYPES = set()
_FP8_OCP = set()
_FP8_FNUZ = set()
for _name in ("float8_e4m3fn", "float8_e5m2"):
    if hasattr(torch, _name):
        _FP8_DTYPES.add(getattr(torch, _name))
        _FP8_OCP.add(getattr(torch, _name))
for _name in ("float8_e4m3fnuz", "float8_e5m2fnuz"):
    if hasattr(torch, _name):
        _FP8_DTYPES.add(getattr(torch, _name))
        _FP8_FNUZ.add(getattr(torch, _name))


def is_float8_dtype(dtype: torch.dtype) -> bool:
    """True for any fp8 dtype, fnuz variants included
    (reference: is_float8_dtype, any_device_parallel.py:93-98)."""
    return dtype in _FP8_DTYPES or "float8" in str(dtype)


def is_ocp_fp8(dtype: torch.dtype) -> bool:
    return dtype in _FP8_OCP


def is_fnuz_fp8(dtype: torch.dtype) -> bool:
    return dtype in _FP8_FNUZ or "fnuz" in str(dtype)


def device_supports_float8(device) -> bool:
    """gfx950 supports fp8 natively; cpu does not run fp8 matmul kernels.

    Replaces the reference's SM >= (9,0) gate (any_device_parallel.py:112-124)
    with the MI355X truth: every HIP device in this framework's scope is a
    gfx9xx CDNA part with fp8 MFMA.
    """
    return torch.device(device).type == "cuda"


def to_ocp_fp8(t: torch.Tensor) -> torch.Tensor:
    """Re-encode a legacy fnuz fp8 tensor to the OCP format gfx950 consumes.

    fnuz and OCP e4m3 have different exponent bias (fnuz has no inf and a
    single NaN); a bit-cast is WRONG — go through a wider dtype.
    """
    if not is_fnuz_fp8(t.dtype):
        return t
    target = torch.float8_e4m3fn if "e4m3" in str(t.dtype) else torch.float8_e5m2
    return t.to(torch.float32).to(target)


def sanitize_param_dtype(t: torch.Tensor, device) -> torch.Tensor:
    """Policy applied when weights land on a replica device:

    - fp8 on a gfx950 GPU: keep fp8, but normalize fnuz -> OCP.
    - fp8 on cpu (test backend): upcast to fp16 so CPU matmuls run —
      the reference's own fallback (any_device_parallel.py:688-699).
    """
    if not is_float8_dtype(t.dtype):
        return t
    if device_supports_float8(device):
        return to_ocp_fp8(t)
    return t.to(torch.float16)
