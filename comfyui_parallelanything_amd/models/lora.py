"""Headless LoRA loading + weight merging.

Inside ComfyUI the reference relies on the host's LoRA loaders and only
bakes live patches before replication (reference any_device_parallel.py
:971-1004 — our utils/comfy_shim.apply_lora_patches mirrors that). This
module provides the equivalent capability for headless serving (bench/CLI
/API): load a LoRA .safetensors and merge ``W += scale * (alpha/r) B @ A``
directly into the model's Linear weights BEFORE replication — replicas
then carry the patched weights exactly like the reference's
clone-after-patch path.

Supported key conventions:
- PEFT / diffusers: ``{module.path}.lora_A.weight`` / ``.lora_B.weight``
- kohya: ``lora_unet_{module_path_with_underscores}.lora_down.weight`` /
  ``.lora_up.weight`` with an optional scalar ``.alpha`` tensor

Merging is exact and invertible: ``merge_lora(model, sd, scale=-s)``
undoes a ``scale=s`` merge (up to fp accumulation order).
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import torch
from torch import nn

log = logging.getLogger("parallelanything")

_PREFIXES = ("lora_unet_", "lora_te_", "lora_")


def load_lora(path: str) -> Dict[str, torch.Tensor]:
    """Read a LoRA .safetensors into a flat state dict (CPU tensors)."""
    from safetensors.torch import load_file

    return load_file(path)


def _pairs(sd: Dict[str, torch.Tensor]):
    """Group flat keys into (module_name, down/A, up/B, alpha) tuples."""
    out: Dict[str, Dict[str, torch.Tensor]] = {}
    for key, t in sd.items():
        for marker, slot in (
            (".lora_A.weight", "down"), (".lora_B.weight", "up"),
            (".lora_down.weight", "down"), (".lora_up.weight", "up"),
            (".alpha", "alpha"),
        ):
            if key.endswith(marker):
                out.setdefault(key[: -len(marker)], {})[slot] = t
                break
    for name, slots in out.items():
        if "down" in slots and "up" in slots:
            yield name, slots["down"], slots["up"], slots.get("alpha")


def _resolve(model: nn.Module, name: str) -> Optional[nn.Module]:
    """Resolve a dotted or kohya-underscored module path.

    kohya flattens dots to underscores, which is ambiguous against
    attribute names that themselves contain underscores (double_blocks,
    img_attn, ...): resolved by greedy longest-prefix match against the
    actual child names at each level.
    """
    for prefix in _PREFIXES:
        if name.startswith(prefix):
            name = name[len(prefix):]
            break
    cur: Optional[nn.Module] = model
    try:  # dotted path (PEFT)
        for part in name.split("."):
            cur = cur[int(part)] if part.isdigit() else getattr(cur, part)
        if isinstance(cur, nn.Module):
            return cur
    except (AttributeError, IndexError, KeyError, TypeError, ValueError):
        pass
    tokens = name.split("_")
    cur = model
    i = 0
    while i < len(tokens):
        children = dict(cur.named_children())
        nxt: Optional[Tuple[nn.Module, int]] = None
        for j in range(len(tokens), i, -1):
            cand = "_".join(tokens[i:j])
            if cand in children:
                nxt = (children[cand], j)
                break
        if nxt is None:
            return None
        cur, i = nxt
    return cur


def _target_linear(mod: Optional[nn.Module]) -> Optional[nn.Linear]:
    if isinstance(mod, nn.Linear):
        return mod
    # fused wrappers (models/layers.py GELULinear) keep the Linear at .lin
    lin = getattr(mod, "lin", None)
    if isinstance(lin, nn.Linear):
        return lin
    return None


@torch.no_grad()
def merge_lora(
    model: nn.Module, sd: Dict[str, torch.Tensor], scale: float = 1.0
) -> int:
    """Merge ``W += scale * (alpha/r) * up @ down`` into matching Linears.

    Returns the number of modules patched. Call BEFORE fp8 quantization
    and BEFORE engine.setup() so every replica carries the merged weights
    (reference clone-after-patch invariant, :1073-1081). Unknown keys are
    skipped with a warning; a quantized (non-floating) weight is skipped.
    """
    merged = 0
    for name, down, up, alpha in _pairs(sd):
        lin = _target_linear(_resolve(model, name))
        if lin is None:
            log.warning("lora: no Linear for key %r; skipped", name)
            continue
        w = lin.weight
        if not w.is_floating_point():
            log.warning("lora: %r weight is quantized; merge before "
                        "quantize_fp8 — skipped", name)
            continue
        r = down.shape[0]
        a = float(alpha.item()) if alpha is not None else float(r)
        if up.shape[1] != r or w.shape != (up.shape[0], down.shape[1]):
            log.warning("lora: shape mismatch for %r (W %s, up %s, down "
                        "%s); skipped", name, tuple(w.shape),
                        tuple(up.shape), tuple(down.shape))
            continue
        delta = (up.to(torch.float32) @ down.to(torch.float32))
        delta = delta * (scale * a / r)
        w += delta.to(device=w.device, dtype=w.dtype)
        merged += 1
    return merged


@torch.no_grad()
def merge_lora_file(model: nn.Module, path: str, scale: float = 1.0) -> int:
    return merge_lora(model, load_lora(path), scale)


def lora_target_names(model: nn.Module,
                      sd: Dict[str, torch.Tensor]) -> List[str]:
    """Which LoRA modules resolve against this model (for diagnostics)."""
    return [name for name, *_ in _pairs(sd)
            if _target_linear(_resolve(model, name)) is not None]
