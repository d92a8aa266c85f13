"""ComfyUI integration shim.

The reference touches exactly these comfy.model_management functions
(SURVEY.md §2 component #13): soft_empty_cache (:209,273,605,1013),
unload_all_models (:263,1016), get_torch_device (:952,997,1000). When
ComfyUI is importable we delegate; headless (benchmarks, tests, serving)
we provide behavior-equivalent local implementations so the node code has
one call surface either way.
"""
from __future__ import annotations

import gc
import logging

import torch

log = logging.getLogger("parallelanything")

try:  # pragma: no cover - only inside a live ComfyUI process
    import comfy.model_management as _mm  # type: ignore

    HAVE_COMFY = True
except Exception:  # noqa: BLE001
    _mm = None
    HAVE_COMFY = False


def soft_empty_cache() -> None:
    if _mm is not None:
        _mm.soft_empty_cache()
        return
    gc.collect()
    if torch.cuda.is_available():
        torch.cuda.empty_cache()


def unload_all_models() -> None:
    if _mm is not None:
        _mm.unload_all_models()
        return
    # headless: nothing is registered with a model manager; cache purge only
    soft_empty_cache()


def get_torch_device() -> torch.device:
    if _mm is not None:
        return _mm.get_torch_device()
    if torch.cuda.is_available():
        return torch.device("cuda:0")
    return torch.device("cpu")


def unwrap_model(model):
    """MODEL unwrap precedence: model.model.diffusion_model ->
    model.diffusion_model -> raw module (reference :921-930). Returns
    (diffusion_module, model_wrapper_or_None)."""
    inner = getattr(model, "model", None)
    if inner is not None and hasattr(inner, "diffusion_model"):
        return inner.diffusion_model, model
    if hasattr(model, "diffusion_model"):
        return model.diffusion_model, model
    return model, None


def detect_lora_patches(model_wrapper) -> bool:
    """LoRA patch detection at the reference's three lookup sites
    (:971-1004): wrapper.patches, wrapper.model.patches, object patches."""
    if model_wrapper is None:
        return False
    for holder in (model_wrapper, getattr(model_wrapper, "model", None)):
        if holder is None:
            continue
        patches = getattr(holder, "patches", None)
        if patches:
            return True
        obj_patches = getattr(holder, "object_patches", None)
        if obj_patches:
            return True
    return False


def apply_lora_patches(model_wrapper, device) -> bool:
    """Bake live LoRA patches into the weights before replication
    (reference patch_model(device_to=...), :993-1004)."""
    if model_wrapper is None or not hasattr(model_wrapper, "patch_model"):
        return False
    try:
        model_wrapper.patch_model(device_to=torch.device(device))
        return True
    except TypeError:
        try:
            model_wrapper.patch_model()
            return True
        except Exception:  # noqa: BLE001
            log.warning("patch_model failed; replicating unpatched weights")
    except Exception:  # noqa: BLE001
        log.warning("patch_model failed; replicating unpatched weights")
    return False
