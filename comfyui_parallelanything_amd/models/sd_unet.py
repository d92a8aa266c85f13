"""SD1.5 / SDXL-class UNet family.

Classic latent-diffusion UNet: ResBlocks (GroupNorm+SiLU+Conv with timestep
injection) and SpatialTransformer blocks (self-attn + cross-attn + GEGLU FF)
at the configured levels. GroupNorm+SiLU and attention route through the
ops dispatch (gfx950 HIP kernels on GPU); convs go through MIOpen via torch.

Forward contract: forward(x, timesteps, context=None, y=None) — matches the
engine intercept signature (reference any_device_parallel.py:1287).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
from torch import nn

from .. import ops


class GNSiLU(nn.Module):
    def __init__(self, channels: int, groups: int = 32):
        super().__init__()
        self.groups = min(groups, channels)
        self.weight = nn.Parameter(torch.ones(channels))
        self.bias = nn.Parameter(torch.zeros(channels))

    def forward(self, x):
        return ops.group_norm_silu(x, self.groups, self.weight, self.bias)


class ResBlock(nn.Module):
    def __init__(self, in_ch: int, out_ch: int, emb_dim: int):
        super().__init__()
        self.in_norm = GNSiLU(in_ch)
        self.conv1 = nn.Conv2d(in_ch, out_ch, 3, padding=1)
        self.emb_proj = nn.Linear(emb_dim, out_ch)
        self.out_norm = GNSiLU(out_ch)
        self.conv2 = nn.Conv2d(out_ch, out_ch, 3, padding=1)
        self.skip = nn.Conv2d(in_ch, out_ch, 1) if in_ch != out_ch else nn.Identity()

    def forward(self, x, emb):
        h = self.conv1(self.in_norm(x))
        h = h + self.emb_proj(torch.nn.functional.silu(emb))[:, :, None, None]
        h = self.conv2(self.out_norm(h))
        return self.skip(x) + h


class CrossAttention(nn.Module):
    def __init__(self, dim: int, ctx_dim: int, num_heads: int):
        super().__init__()
        self.num_heads = num_heads
        self.to_q = nn.Linear(dim, dim, bias=False)
        self.to_k = nn.Linear(ctx_dim, dim, bias=False)
        self.to_v = nn.Linear(ctx_dim, dim, bias=False)
        self.to_out = nn.Linear(dim, dim)
        self.scale = 1.0 / math.sqrt(dim // num_heads)

    def forward(self, x, context=None):
        ctx = x if context is None else context
        H = self.num_heads
        q = self.to_q(x).unflatten(-1, (H, -1))
        k = self.to_k(ctx).unflatten(-1, (H, -1))
        v = self.to_v(ctx).unflatten(-1, (H, -1))
        return self.to_out(ops.attention_bshd(q, k, v, self.scale).flatten(2))


class GEGLU(nn.Module):
    def __init__(self, dim: int, inner: int):
        super().__init__()
        self.proj = nn.Linear(dim, inner * 2)

    def forward(self, x):
        a, b = self.proj(x).chunk(2, dim=-1)
        return a * torch.nn.functional.gelu(b)


class TransformerBlock(nn.Module):
    def __init__(self, dim: int, ctx_dim: int, num_heads: int):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim)
        self.attn1 = CrossAttention(dim, dim, num_heads)  # self
        self.norm2 = nn.LayerNorm(dim)
        self.attn2 = CrossAttention(dim, ctx_dim, num_heads)  # cross
        self.norm3 = nn.LayerNorm(dim)
        self.ff = nn.Sequential(GEGLU(dim, dim * 4), nn.Linear(dim * 4, dim))

    def forward(self, x, context=None):
        x = x + self.attn1(self.norm1(x))
        x = x + self.attn2(self.norm2(x), context)
        x = x + self.ff(self.norm3(x))
        return x


class SpatialTransformer(nn.Module):
    def __init__(self, channels: int, ctx_dim: int, num_heads: int, depth: int):
        super().__init__()
        self.norm = nn.GroupNorm(min(32, channels), channels, eps=1e-6)
        self.proj_in = nn.Linear(channels, channels)
        self.blocks = nn.ModuleList(
            TransformerBlock(channels, ctx_dim, num_heads) for _ in range(depth)
        )
        self.proj_out = nn.Linear(channels, channels)

    def forward(self, x, context=None):
        B, C, H, W = x.shape
        h = self.norm(x).permute(0, 2, 3, 1).reshape(B, H * W, C)
        h = self.proj_in(h)
        for blk in self.blocks:
            h = blk(h, context)
        h = self.proj_out(h)
        return x + h.reshape(B, H, W, C).permute(0, 3, 1, 2)


class Downsample(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.op = nn.Conv2d(ch, ch, 3, stride=2, padding=1)

    def forward(self, x):
        return self.op(x)


class Upsample(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.conv = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        return self.conv(torch.nn.functional.interpolate(x, scale_factor=2, mode="nearest"))


@dataclass
class UNetConfig:
    in_channels: int = 4
    model_channels: int = 320
    out_channels: int = 4
    channel_mult: Tuple[int, ...] = (1, 2, 4, 4)
    num_res_blocks: int = 2
    transformer_depth: Tuple[int, ...] = (1, 1, 1, 0)  # per level
    context_dim: int = 768
    head_dim: int = 64
    adm_in_channels: int = 0  # SDXL pooled-conditioning dim (0 = off)

    @classmethod
    def sd15(cls) -> "UNetConfig":
        return cls()

    @classmethod
    def sdxl(cls) -> "UNetConfig":
        return cls(channel_mult=(1, 2, 4), transformer_depth=(0, 2, 10),
                   context_dim=2048, adm_in_channels=2816)

    @classmethod
    def tiny(cls) -> "UNetConfig":
        return cls(model_channels=32, channel_mult=(1, 2), num_res_blocks=1,
                   transformer_depth=(1, 1), context_dim=32, head_dim=8)


class _Seq(nn.Module):
    """Timestep/context-aware sequential container."""

    def __init__(self, *mods):
        super().__init__()
        self.mods = nn.ModuleList(mods)

    def forward(self, x, emb, context):
        for m in self.mods:
            if isinstance(m, ResBlock):
                x = m(x, emb)
            elif isinstance(m, SpatialTransformer):
                x = m(x, context)
            else:
                x = m(x)
        return x


class SDUNet(nn.Module):
    """SD1.5/SDXL-class UNet."""

    def __init__(self, cfg: Optional[UNetConfig] = None):
        super().__init__()
        cfg = cfg or UNetConfig.sd15()
        self.cfg = cfg
        mc = cfg.model_channels
        emb_dim = mc * 4
        self.time_embed = nn.Sequential(
            nn.Linear(mc, emb_dim), nn.SiLU(), nn.Linear(emb_dim, emb_dim)
        )
        self.label_emb = (
            nn.Sequential(nn.Linear(cfg.adm_in_channels, emb_dim), nn.SiLU(),
                          nn.Linear(emb_dim, emb_dim))
            if cfg.adm_in_channels else None
        )
        self.input_conv = nn.Conv2d(cfg.in_channels, mc, 3, padding=1)

        def heads(ch):
            return max(1, ch // cfg.head_dim)

        self.down = nn.ModuleList()
        ch = mc
        input_chs = [mc]
        for level, mult in enumerate(cfg.channel_mult):
            out_ch = mc * mult
            for _ in range(cfg.num_res_blocks):
                mods = [ResBlock(ch, out_ch, emb_dim)]
                ch = out_ch
                if cfg.transformer_depth[level] > 0:
                    mods.append(SpatialTransformer(
                        ch, cfg.context_dim, heads(ch), cfg.transformer_depth[level]))
                self.down.append(_Seq(*mods))
                input_chs.append(ch)
            if level != len(cfg.channel_mult) - 1:
                self.down.append(_Seq(Downsample(ch)))
                input_chs.append(ch)

        mid_depth = cfg.transformer_depth[-1] or 1
        self.mid = _Seq(
            ResBlock(ch, ch, emb_dim),
            SpatialTransformer(ch, cfg.context_dim, heads(ch), mid_depth),
            ResBlock(ch, ch, emb_dim),
        )

        self.up = nn.ModuleList()
        for level, mult in reversed(list(enumerate(cfg.channel_mult))):
            out_ch = mc * mult
            for i in range(cfg.num_res_blocks + 1):
                skip_ch = input_chs.pop()
                mods = [ResBlock(ch + skip_ch, out_ch, emb_dim)]
                ch = out_ch
                if cfg.transformer_depth[level] > 0:
                    mods.append(SpatialTransformer(
                        ch, cfg.context_dim, heads(ch), cfg.transformer_depth[level]))
                if level != 0 and i == cfg.num_res_blocks:
                    mods.append(Upsample(ch))
                self.up.append(_Seq(*mods))

        self.out_norm = GNSiLU(ch)
        self.out_conv = nn.Conv2d(ch, cfg.out_channels, 3, padding=1)

    @torch.no_grad()
    def forward(self, x, timesteps, context=None, y=None, **kwargs):
        cfg = self.cfg
        emb = self.time_embed(
            ops.timestep_embedding(timesteps, cfg.model_channels).to(x.dtype)
        )
        if self.label_emb is not None:
            if y is None:
                y = torch.zeros(x.shape[0], cfg.adm_in_channels,
                                device=x.device, dtype=x.dtype)
            emb = emb + self.label_emb(y)
        if context is None:
            context = torch.zeros(x.shape[0], 1, cfg.context_dim,
                                  device=x.device, dtype=x.dtype)
        h = self.input_conv(x)
        skips = [h]
        for mod in self.down:
            h = mod(h, emb, context)
            skips.append(h)
        h = self.mid(h, emb, context)
        for mod in self.up:
            h = torch.cat([h, skips.pop()], dim=1)
            h = mod(h, emb, context)
        return self.out_conv(self.out_norm(h))
