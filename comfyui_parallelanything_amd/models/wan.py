"""WAN2.2-class video DiT (image/text-to-video, temporal attention path).

Single-stream video transformer: 3D-patchified video latents, blocks of
{self-attention over the space-time token grid with 3D RoPE, cross-attention
to text context, FFN}, AdaLN modulation from the timestep embedding. Block
list attr is ``transformer_blocks`` so the reference-compatible pipeline
mode can shard it (any_device_parallel.py:1156).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional, Tuple

import torch
from torch import nn

from .. import ops
from .layers import FusedMLP, MLPEmbedder, QKNorm, ln_mod_into


def rope_3d_table(f: int, h: int, w: int, axes_dim: Tuple[int, ...],
                  theta: float, device) -> torch.Tensor:
    """3D (frame, row, col) RoPE table -> [f*h*w, D/2, 2] fp32."""
    fs = torch.arange(f, device=device)
    ys = torch.arange(h, device=device)
    xs = torch.arange(w, device=device)
    gf, gy, gx = torch.meshgrid(fs, ys, xs, indexing="ij")
    ids = torch.stack([gf, gy, gx], dim=-1).reshape(-1, 3).float()
    tables = [ops.rope_freqs(ids[:, ax], axes_dim[ax], theta) for ax in range(3)]
    return torch.cat(tables, dim=1)


class WanBlock(nn.Module):
    def __init__(self, dim: int, ffn_dim: int, num_heads: int, ctx_dim: int):
        super().__init__()
        head_dim = dim // num_heads
        self.num_heads = num_heads
        self.scale = 1.0 / math.sqrt(head_dim)
        # modulation: 6 AdaLN params from time embedding (+ learned bias)
        self.mod = nn.Parameter(torch.randn(1, 6, dim) / dim**0.5)
        self.self_qkv = nn.Linear(dim, dim * 3)
        self.self_norm = QKNorm(head_dim)
        self.self_proj = nn.Linear(dim, dim)
        self.norm_cross = nn.LayerNorm(dim)
        self.cross_q = nn.Linear(dim, dim)
        self.cross_k = nn.Linear(ctx_dim, dim)
        self.cross_v = nn.Linear(ctx_dim, dim)
        self.cross_proj = nn.Linear(dim, dim)
        self.ffn = FusedMLP(dim, ffn_dim, dim)

    def forward(self, x, e, context, pe):
        # e: [B, 6, dim] time-modulation; learned bias added per block
        m = (e + self.mod).unbind(dim=1)  # 6 x [B, dim]
        shift1, scale1, gate1, shift2, scale2, gate2 = m

        qkv = ln_mod_into(self.self_qkv, x, scale1, shift1).unflatten(
            -1, (3, self.num_heads, -1)
        )
        q, k, v = qkv.unbind(2)  # [B,S,H,D] views
        ops.qk_norm_rope_(
            q, k, self.self_norm.query_norm.scale,
            self.self_norm.key_norm.scale, pe,
        )
        attn = ops.attention_bshd(q, k, v, self.scale).flatten(2)
        x = ops.gate_residual(x, gate1, self.self_proj(attn))

        h = self.norm_cross(x)
        H = self.num_heads
        q = self.cross_q(h).unflatten(-1, (H, -1))
        k = self.cross_k(context).unflatten(-1, (H, -1))
        v = self.cross_v(context).unflatten(-1, (H, -1))
        x = x + self.cross_proj(
            ops.attention_bshd(q, k, v, self.scale).flatten(2)
        )

        return ops.gate_residual(
            x, gate2, self.ffn.forward_ln(x, scale2, shift2)
        )


@dataclass
class WanConfig:
    in_channels: int = 16
    # I2V conditioning channels concatenated on the channel axis BEFORE
    # patchify: a binary first-frame mask (4ch) + the reference image's VAE
    # latent (16ch), zero elsewhere in time — the WAN2.2 I2V input layout
    # (in_dim 36 = 16 noise + 20 cond). 0 = T2V.
    cond_channels: int = 0
    patch_size: Tuple[int, int, int] = (1, 2, 2)  # (frame, h, w)
    dim: int = 5120
    ffn_dim: int = 13824
    num_heads: int = 40
    depth: int = 40
    ctx_dim: int = 4096  # umT5 features
    axes_dim: Tuple[int, ...] = (44, 42, 42)
    theta: float = 10000.0
    time_embed_dim: int = 256

    @classmethod
    def wan22_a14b(cls) -> "WanConfig":
        return cls()

    @classmethod
    def wan22_a14b_i2v(cls) -> "WanConfig":
        return cls(cond_channels=20)

    @classmethod
    def wan22_5b(cls) -> "WanConfig":
        return cls(dim=3072, ffn_dim=14336, num_heads=24, depth=30,
                   patch_size=(1, 2, 2), axes_dim=(44, 42, 42))

    @classmethod
    def tiny(cls) -> "WanConfig":
        return cls(in_channels=4, dim=64, ffn_dim=128, num_heads=4, depth=2,
                   ctx_dim=32, axes_dim=(8, 4, 4), time_embed_dim=32)

    @classmethod
    def tiny_i2v(cls) -> "WanConfig":
        return cls(in_channels=4, cond_channels=5, dim=64, ffn_dim=128,
                   num_heads=4, depth=2, ctx_dim=32, axes_dim=(8, 4, 4),
                   time_embed_dim=32)


class WanDiT(nn.Module):
    """WAN2.2-class video DiT. Input latent: [B, C, F, H, W]."""

    def __init__(self, cfg: Optional[WanConfig] = None):
        super().__init__()
        cfg = cfg or WanConfig()
        self.cfg = cfg
        pf, ph, pw = cfg.patch_size
        # output predicts noise for the latent channels only; the input
        # embedding additionally sees the I2V conditioning channels
        self.patch_dim = cfg.in_channels * pf * ph * pw
        self.patch_dim_in = (cfg.in_channels + cfg.cond_channels) * pf * ph * pw
        self.patch_in = nn.Linear(self.patch_dim_in, cfg.dim)
        self.txt_in = FusedMLP(cfg.ctx_dim, cfg.dim, cfg.dim)
        self.time_in = MLPEmbedder(cfg.time_embed_dim, cfg.dim)
        self.time_proj = nn.Linear(cfg.dim, cfg.dim * 6)
        self.transformer_blocks = nn.ModuleList(
            WanBlock(cfg.dim, cfg.ffn_dim, cfg.num_heads, cfg.dim)
            for _ in range(cfg.depth)
        )
        self.head_mod = nn.Linear(cfg.dim, cfg.dim * 2)
        self.head = nn.Linear(cfg.dim, self.patch_dim)
        self._pe_cache: dict = {}

    def _pe(self, f, h, w, device):
        key = (f, h, w, str(device))
        pe = self._pe_cache.get(key)
        if pe is None:
            pe = rope_3d_table(f, h, w, self.cfg.axes_dim, self.cfg.theta, device)
            self._pe_cache[key] = pe
        return pe

    @torch.no_grad()
    def forward(self, x, timesteps, context=None, image_cond=None, **kwargs):
        cfg = self.cfg
        B, C, F, H, W = x.shape
        pf, ph, pw = cfg.patch_size
        f, h, w = F // pf, H // ph, W // pw
        if cfg.cond_channels:
            # I2V: concatenate mask + reference-image latent channels
            # (batch-shaped kwarg, so the engine's kwargs-split rules
            # scatter it with the latent — split.py split_kwargs)
            if image_cond is None:
                raise ValueError(
                    "I2V WanDiT requires image_cond "
                    f"[B,{cfg.cond_channels},F,H,W]"
                )
            if image_cond.shape != (B, cfg.cond_channels, F, H, W):
                raise ValueError(
                    f"image_cond shape {tuple(image_cond.shape)} != "
                    f"{(B, cfg.cond_channels, F, H, W)}"
                )
            x_in = torch.cat([x, image_cond.to(x.dtype)], dim=1)
        else:
            x_in = x
        Ct = x_in.shape[1]
        tokens = (
            x_in.view(B, Ct, f, pf, h, ph, w, pw)
            .permute(0, 2, 4, 6, 1, 3, 5, 7)
            .reshape(B, f * h * w, self.patch_dim_in)
        )
        seq = self.patch_in(tokens)
        if context is None:
            context = torch.zeros(B, 1, cfg.ctx_dim, device=x.device, dtype=x.dtype)
        ctx = self.txt_in(context)
        tvec = self.time_in.forward_timestep(timesteps)
        e = self.time_proj(torch.nn.functional.silu(tvec)).view(B, 6, cfg.dim)
        pe = self._pe(f, h, w, x.device)
        for block in self.transformer_blocks:
            seq = block(seq, e, ctx, pe)
        shift, scale = self.head_mod(torch.nn.functional.silu(tvec)).chunk(2, dim=-1)
        out = self.head(ops.layer_norm_mod(seq, scale, shift))
        return (
            out.view(B, f, h, w, C, pf, ph, pw)
            .permute(0, 4, 1, 5, 2, 6, 3, 7)
            .reshape(B, C, F, H, W)
        )
