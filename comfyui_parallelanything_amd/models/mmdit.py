"""MMDiT model family: FLUX.1-class dual/single-stream DiT and the
Z-Image-class single-stream DiT.

Random-init, synthetic-input implementations at the public architectures'
shapes (there is no network for checkpoints in this environment). The
forward signature is the engine's contract: forward(x, timesteps,
context=None, **kwargs) — x is the latent [B, C, H, W], context the text
embedding [B, T, ctx_dim], y an optional pooled vector (reference intercept
signature: any_device_parallel.py:1287).

Block-list attribute names (double_blocks / single_blocks / layers) match
what the reference's pipeline mode scans (any_device_parallel.py:1156), so
batch==1 layer-sharding works on these models unchanged.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional, Tuple

import torch
from torch import nn

from .. import ops
from .layers import (
    DoubleStreamBlock,
    LastLayer,
    MLPEmbedder,
    ModulationBank,
    SingleStreamBlock,
    rope_2d_table,
)


def _base_block(blk):
    """Reach through pipeline wrappers (ParallelBlock.local_block /
    DistPipelineBlock.block) to the underlying block."""
    return getattr(blk, "local_block", None) or getattr(blk, "block", None) or blk


@dataclass
class FluxConfig:
    in_channels: int = 16
    patch_size: int = 2
    hidden: int = 3072
    num_heads: int = 24
    depth_double: int = 19
    depth_single: int = 38
    mlp_ratio: float = 4.0
    context_dim: int = 4096   # T5-XXL features
    vec_dim: int = 768        # CLIP pooled
    axes_dim: Tuple[int, ...] = (16, 56, 56)
    theta: float = 10000.0
    guidance_embed: bool = True
    time_embed_dim: int = 256

    @classmethod
    def flux1_dev(cls) -> "FluxConfig":
        return cls()

    @classmethod
    def sd35_large(cls) -> "FluxConfig":
        """SD3.5-Large-class MMDiT: joint (dual-stream) trunk only — no
        single-stream tail — with 64-dim heads (exercises the D=64
        attention path at MMDiT scale)."""
        return cls(hidden=2432, num_heads=38, depth_double=38,
                   depth_single=0, context_dim=4096, vec_dim=2048,
                   axes_dim=(16, 24, 24), guidance_embed=False)

    @classmethod
    def sd35_tiny(cls) -> "FluxConfig":
        return cls(in_channels=4, hidden=64, num_heads=4, depth_double=3,
                   depth_single=0, context_dim=32, vec_dim=16,
                   axes_dim=(4, 6, 6), time_embed_dim=32,
                   guidance_embed=False)

    @classmethod
    def tiny(cls) -> "FluxConfig":
        """CPU-test scale."""
        return cls(in_channels=4, hidden=64, num_heads=4, depth_double=2,
                   depth_single=2, context_dim=32, vec_dim=16,
                   axes_dim=(4, 6, 6), time_embed_dim=32)


class Flux(nn.Module):
    """FLUX.1-class MMDiT (dual-stream + single-stream trunk)."""

    def __init__(self, cfg: Optional[FluxConfig] = None):
        super().__init__()
        cfg = cfg or FluxConfig()
        self.cfg = cfg
        p = cfg.patch_size
        self.patch_dim = cfg.in_channels * p * p
        self.img_in = nn.Linear(self.patch_dim, cfg.hidden)
        self.txt_in = nn.Linear(cfg.context_dim, cfg.hidden)
        self.time_in = MLPEmbedder(cfg.time_embed_dim, cfg.hidden)
        self.vector_in = MLPEmbedder(cfg.vec_dim, cfg.hidden)
        self.guidance_in = (
            MLPEmbedder(cfg.time_embed_dim, cfg.hidden) if cfg.guidance_embed else None
        )
        self.double_blocks = nn.ModuleList(
            DoubleStreamBlock(cfg.hidden, cfg.num_heads, cfg.mlp_ratio)
            for _ in range(cfg.depth_double)
        )
        self.single_blocks = nn.ModuleList(
            SingleStreamBlock(cfg.hidden, cfg.num_heads, cfg.mlp_ratio)
            for _ in range(cfg.depth_single)
        )
        self.final_layer = LastLayer(cfg.hidden, self.patch_dim)
        # per-replica RoPE cache, keyed by (h, w, txt_len); lives on the
        # owning device by construction (vs reference clear_flux_caches).
        self._pe_cache: dict = {}
        self._mod_cache: dict = {}  # ModulationBank per replica (CACHE_ATTRS)

    def _patchify(self, x: torch.Tensor):
        B, C, H, W = x.shape
        p = self.cfg.patch_size
        h, w = H // p, W // p
        x = x.view(B, C, h, p, w, p).permute(0, 2, 4, 1, 3, 5).reshape(
            B, h * w, C * p * p
        )
        return x, h, w

    def _unpatchify(self, x: torch.Tensor, h: int, w: int):
        B = x.shape[0]
        p = self.cfg.patch_size
        C = self.cfg.in_channels
        return (
            x.view(B, h, w, C, p, p)
            .permute(0, 3, 1, 4, 2, 5)
            .reshape(B, C, h * p, w * p)
        )

    def _pe(self, h: int, w: int, txt_len: int, device, dtype):
        key = (h, w, txt_len, str(device))
        pe = self._pe_cache.get(key)
        if pe is None:
            pe = rope_2d_table(
                h, w, self.cfg.axes_dim, self.cfg.theta, device, txt_len
            )
            self._pe_cache[key] = pe
        return pe

    @torch.no_grad()
    def forward(self, x, timesteps, context=None, y=None, guidance=None, **kwargs):
        cfg = self.cfg
        B = x.shape[0]
        img, h, w = self._patchify(x)
        img = self.img_in(img)
        if context is None:
            context = torch.zeros(B, 1, cfg.context_dim, device=x.device, dtype=x.dtype)
        txt = self.txt_in(context)
        vec = self.time_in.forward_timestep(timesteps)
        if y is None:
            y = torch.zeros(B, cfg.vec_dim, device=x.device, dtype=x.dtype)
        vec = vec + self.vector_in(y)
        if self.guidance_in is not None:
            if guidance is None:
                guidance = torch.full_like(timesteps, 4.0)
            vec = vec + self.guidance_in.forward_timestep(guidance)
        pe = self._pe(h, w, txt.shape[1], x.device, x.dtype)

        bank = self._mod_cache.get("bank")
        if bank is None:
            try:
                mods = []
                for blk in self.double_blocks:
                    b = _base_block(blk)
                    mods += [b.img_mod, b.txt_mod]
                mods += [
                    _base_block(blk).modulation for blk in self.single_blocks
                ]
                bank = ModulationBank(mods)
            except TypeError:
                bank = False  # quantized mods: per-block path
            self._mod_cache["bank"] = bank
        banked = bank(vec) if bank else None

        i = 0
        for block in self.double_blocks:
            img, txt = block(
                img, txt, vec, pe,
                mods=(banked[i], banked[i + 1]) if banked else None,
            )
            i += 2
        xcat = torch.cat([txt, img], dim=1)
        for block in self.single_blocks:
            xcat = block(xcat, vec, pe,
                         mods=(banked[i][0],) if banked else None)
            i += 1
        img = xcat[:, txt.shape[1]:]
        out = self.final_layer(img, vec)
        return self._unpatchify(out, h, w)


@dataclass
class ZImageConfig:
    """Z-Image-class ~6B single-stream DiT."""

    in_channels: int = 16
    patch_size: int = 2
    hidden: int = 3584
    num_heads: int = 28
    depth: int = 36
    mlp_ratio: float = 4.0
    context_dim: int = 2560
    axes_dim: Tuple[int, ...] = (32, 48, 48)
    theta: float = 10000.0
    time_embed_dim: int = 256

    @classmethod
    def z_image_turbo(cls) -> "ZImageConfig":
        return cls()

    @classmethod
    def tiny(cls) -> "ZImageConfig":
        return cls(in_channels=4, hidden=64, num_heads=4, depth=3,
                   context_dim=32, axes_dim=(4, 6, 6), time_embed_dim=32)


class ZImage(nn.Module):
    """Z-Image-class single-stream DiT: text tokens prepended to the image
    sequence, trunk of single-stream blocks (block list attr: ``layers``)."""

    def __init__(self, cfg: Optional[ZImageConfig] = None):
        super().__init__()
        cfg = cfg or ZImageConfig()
        self.cfg = cfg
        p = cfg.patch_size
        self.patch_dim = cfg.in_channels * p * p
        self.img_in = nn.Linear(self.patch_dim, cfg.hidden)
        self.txt_in = nn.Linear(cfg.context_dim, cfg.hidden)
        self.time_in = MLPEmbedder(cfg.time_embed_dim, cfg.hidden)
        self.layers = nn.ModuleList(
            SingleStreamBlock(cfg.hidden, cfg.num_heads, cfg.mlp_ratio)
            for _ in range(cfg.depth)
        )
        self.final_layer = LastLayer(cfg.hidden, self.patch_dim)
        self._pe_cache: dict = {}
        self._mod_cache: dict = {}

    def _pe(self, h, w, txt_len, device):
        key = (h, w, txt_len, str(device))
        pe = self._pe_cache.get(key)
        if pe is None:
            pe = rope_2d_table(h, w, self.cfg.axes_dim, self.cfg.theta, device, txt_len)
            self._pe_cache[key] = pe
        return pe

    @torch.no_grad()
    def forward(self, x, timesteps, context=None, **kwargs):
        cfg = self.cfg
        B, C, H, W = x.shape
        p = cfg.patch_size
        h, w = H // p, W // p
        img = x.view(B, C, h, p, w, p).permute(0, 2, 4, 1, 3, 5).reshape(
            B, h * w, self.patch_dim
        )
        img = self.img_in(img)
        if context is None:
            context = torch.zeros(B, 1, cfg.context_dim, device=x.device, dtype=x.dtype)
        txt = self.txt_in(context)
        vec = self.time_in.forward_timestep(timesteps)
        seq = torch.cat([txt, img], dim=1)
        pe = self._pe(h, w, txt.shape[1], x.device)
        bank = self._mod_cache.get("bank")
        if bank is None:
            try:
                bank = ModulationBank(
                    [_base_block(blk).modulation for blk in self.layers]
                )
            except TypeError:
                bank = False
            self._mod_cache["bank"] = bank
        banked = bank(vec) if bank else None
        for i, block in enumerate(self.layers):
            seq = block(seq, vec, pe,
                        mods=(banked[i][0],) if banked else None)
        out = self.final_layer(seq[:, txt.shape[1]:], vec)
        return (
            out.view(B, h, w, C, p, p)
            .permute(0, 3, 1, 4, 2, 5)
            .reshape(B, C, H, W)
        )
