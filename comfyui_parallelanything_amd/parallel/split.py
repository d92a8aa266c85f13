"""Batch scatter / kwargs-split / gather-concat math.

Pure-Python/tensor math shared by every transport (in-process streams and
RCCL process groups). Behavioral contract from the reference:

- split sizes: ``max(1, int(B*w))`` per device, remainder to the LAST device
  (any_device_parallel.py:1321-1322). Devices whose computed size is <= 0 are
  dropped from the active set (any_device_parallel.py:1324-1337).
- batch detection: tensors report shape[0]; lists report their first tensor's
  shape[0], else len (any_device_parallel.py:1210-1220).
- kwargs: a tensor kwarg with shape[0]==B splits on dim 0; a list/tuple kwarg
  whose elements ALL are tensors with shape[0]==B splits element-wise;
  everything else is broadcast to every chunk (any_device_parallel.py:1252-1267).
- gather: tensors cat on dim 0; list/tuple outputs cat element-wise with
  non-tensor slots taken from the first result (any_device_parallel.py:1269-1285).
"""
from __future__ import annotations

from typing import Any, Dict, List, Sequence, Tuple

import torch


def get_batch_size(x: Any) -> int:
    """Reference: any_device_parallel.py:1210-1220."""
    if isinstance(x, torch.Tensor):
        return int(x.shape[0])
    if isinstance(x, (list, tuple)) and len(x) > 0:
        sizes = [int(t.shape[0]) for t in x if isinstance(t, torch.Tensor)]
        if sizes:
            return sizes[0]
        return len(x)
    return 1


def compute_split_sizes(batch_size: int, weights: Sequence[float]) -> List[int]:
    """Per-device chunk sizes from normalized weights.

    Reference semantics (any_device_parallel.py:1321-1322): floor with a
    min-1 floor per device, remainder (positive or negative) absorbed by the
    last entry. When the min-1 floors over-commit (batch < #devices after
    flooring), the reference's negative tail would crash torch.split; we
    repair by stealing from the largest earlier entries so sizes are >= 0 and
    sum to batch_size. Size-0 entries are dropped by the caller via
    active_split (reference: skip-zero loop :1324-1337).
    """
    if batch_size < 0:
        raise ValueError(f"negative batch size {batch_size}")
    n = len(weights)
    if n == 0:
        raise ValueError("no weights")
    sizes = [max(1, int(batch_size * w)) for w in weights]
    sizes[-1] = batch_size - sum(sizes[:-1])
    # Repair a negative tail (more devices than samples): steal from the
    # largest earlier chunks until the tail is exactly >= 0.
    i_order = sorted(range(n - 1), key=lambda i: -sizes[i])
    k = 0
    while sizes[-1] < 0 and k < 10_000:
        i = i_order[k % max(1, len(i_order))]
        if sizes[i] > 0:
            sizes[i] -= 1
            sizes[-1] += 1
        k += 1
    if sizes[-1] < 0:  # pragma: no cover - unreachable for batch_size >= 0
        raise RuntimeError("split repair failed")
    return sizes


def active_split(
    devices: Sequence[str], weights: Sequence[float], sizes: Sequence[int]
) -> Tuple[List[str], List[float], List[int]]:
    """Drop size-0 devices (any_device_parallel.py:1324-1337)."""
    act = [(d, w, s) for d, w, s in zip(devices, weights, sizes) if s > 0]
    if not act:
        raise ValueError("no active devices after split")
    devs, ws, ss = zip(*act)
    return list(devs), list(ws), list(ss)


def split_batch(x: Any, split_sizes: Sequence[int]) -> List[Any]:
    """Split positional inputs on dim 0 (any_device_parallel.py:1222-1237)."""
    sizes = list(split_sizes)
    if isinstance(x, torch.Tensor):
        return list(torch.split(x, sizes, dim=0))
    if isinstance(x, (list, tuple)):
        per_elem = []
        for t in x:
            if isinstance(t, torch.Tensor):
                per_elem.append(torch.split(t, sizes, dim=0))
            else:
                per_elem.append([t] * len(sizes))
        return [type(x)(col[i] for col in per_elem) for i in range(len(sizes))]
    return [x] * len(sizes)


def split_kwargs(
    kwargs: Dict[str, Any], split_sizes: Sequence[int], total_batch: int
) -> List[Dict[str, Any]]:
    """Split batch-shaped kwargs, broadcast the rest
    (any_device_parallel.py:1252-1267)."""
    sizes = list(split_sizes)
    out: List[Dict[str, Any]] = [{} for _ in sizes]
    for key, value in kwargs.items():
        if isinstance(value, torch.Tensor) and value.shape[0] == total_batch:
            for i, chunk in enumerate(torch.split(value, sizes, dim=0)):
                out[i][key] = chunk
        elif (
            isinstance(value, (list, tuple))
            and len(value) > 0
            and all(
                isinstance(t, torch.Tensor) and t.shape[0] == total_batch
                for t in value
            )
        ):
            cols = [torch.split(t, sizes, dim=0) for t in value]
            for i in range(len(sizes)):
                out[i][key] = type(value)(c[i] for c in cols)
        else:
            for i in range(len(sizes)):
                out[i][key] = value
    return out


def concatenate_results(results: Sequence[Any], dim: int = 0) -> Any:
    """Gather-side concat incl. nested list/tuple outputs
    (any_device_parallel.py:1269-1285)."""
    if len(results) == 0:
        return results
    first = results[0]
    if isinstance(first, torch.Tensor):
        return torch.cat(list(results), dim=dim)
    if isinstance(first, (list, tuple)):
        merged = []
        for i in range(len(first)):
            if isinstance(first[i], torch.Tensor):
                merged.append(torch.cat([r[i] for r in results], dim=dim))
            else:
                merged.append(first[i])
        return type(first)(merged)
    return results


def move_to_device(x: Any, device, non_blocking: bool = False) -> Any:
    """Recursive tensor move (any_device_parallel.py:1239-1250).

    gfx950 supports fp8 MFMA natively (OCP e4m3fn/e5m2), so unlike the
    reference there is no fp8->fp16 downcast on the move path — fp8 tensors
    stay fp8 (SURVEY.md component #8 disposition).
    """
    dev = torch.device(device)
    if isinstance(x, torch.Tensor):
        if x.device != dev:
            x = x.to(dev, non_blocking=non_blocking)
        return x
    if isinstance(x, (list, tuple)):
        return type(x)(move_to_device(t, dev, non_blocking) for t in x)
    if isinstance(x, dict):
        return {k: move_to_device(v, dev, non_blocking) for k, v in x.items()}
    return x
