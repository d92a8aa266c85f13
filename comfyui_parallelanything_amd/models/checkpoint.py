"""Checkpoint I/O: safetensors save/load for every model family.

The reference consumes ComfyUI's in-memory MODEL and leaves checkpoint
loading upstream (SURVEY.md §5 Checkpoint/resume). Headless deployments of
this framework need their own load path: state dicts in safetensors with a
small JSON config header, loadable directly onto a target HIP device
(assign=True — no host-side double buffering of a 24 GB model).
"""
from __future__ import annotations

import dataclasses
import json
import os
from typing import Optional

import torch
from torch import nn


def save_checkpoint(model: nn.Module, path: str) -> None:
    """Write <path>.safetensors + <path>.json (config)."""
    from safetensors.torch import save_file

    cfg = getattr(model, "cfg", None)
    meta = {
        "class": type(model).__name__,
        "config": dataclasses.asdict(cfg) if cfg is not None else {},
    }
    state = {k: v.contiguous() for k, v in model.state_dict().items()}
    save_file(state, path + ".safetensors")
    with open(path + ".json", "w") as f:
        json.dump(meta, f, indent=2, default=str)


def load_checkpoint(path: str, device="cpu", dtype: Optional[torch.dtype] = None):
    """Instantiate the saved class from its config and load weights
    directly onto ``device``."""
    from safetensors.torch import load_file

    with open(path + ".json") as f:
        meta = json.load(f)
    model = _construct(meta["class"], meta["config"], device)
    state = load_file(path + ".safetensors", device=str(device))
    if dtype is not None:
        state = {k: v.to(dtype) for k, v in state.items()}
    model.load_state_dict(state, assign=True)
    return model.eval()


def _construct(cls_name: str, config: dict, device):
    from .mmdit import Flux, FluxConfig, ZImage, ZImageConfig
    from .sd_unet import SDUNet, UNetConfig
    from .wan import WanConfig, WanDiT

    registry = {
        "Flux": (Flux, FluxConfig),
        "ZImage": (ZImage, ZImageConfig),
        "SDUNet": (SDUNet, UNetConfig),
        "WanDiT": (WanDiT, WanConfig),
    }
    if cls_name not in registry:
        raise ValueError(f"unknown checkpoint class {cls_name!r}")
    model_cls, cfg_cls = registry[cls_name]
    cfg_fields = {f.name for f in dataclasses.fields(cfg_cls)}
    kwargs = {}
    for k, v in config.items():
        if k in cfg_fields:
            kwargs[k] = tuple(v) if isinstance(v, list) else v
    with torch.device(device):
        return model_cls(cfg_cls(**kwargs))
