"""Pure-PyTorch reference implementations of every custom op.

These are the numerics ground truth for the HIP kernels (tests compare the
gfx950 kernels against these run in fp32) and the CPU execution path for the
no-GPU test backend. They are NOT the GPU hot path — on HIP devices the
dispatch layer (ops/__init__.py) requires the native extension.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: Optional[torch.Tensor], eps: float = 1e-6) -> torch.Tensor:
    """RMSNorm over the last dim (FLUX qk-norm and single-stream norms)."""
    dtype = x.dtype
    xf = x.float()
    rrms = torch.rsqrt(xf.pow(2).mean(dim=-1, keepdim=True) + eps)
    out = xf * rrms
    if weight is not None:
        out = out * weight.float()
    return out.to(dtype)


def layer_norm_mod(
    x: torch.Tensor, scale: torch.Tensor, shift: torch.Tensor, eps: float = 1e-6
) -> torch.Tensor:
    """AdaLN-modulated LayerNorm: LN(x) * (1 + scale) + shift.

    x: [B, S, D]; scale/shift: [B, D] (broadcast over S) or [B, S, D].
    No learned affine — modulation plays that role (MMDiT convention).
    """
    dtype = x.dtype
    out = F.layer_norm(x.float(), (x.shape[-1],), eps=eps)
    if scale.dim() == x.dim() - 1:
        scale = scale.unsqueeze(1)
        shift = shift.unsqueeze(1)
    out = out * (1.0 + scale.float()) + shift.float()
    return out.to(dtype)


def gate_residual(residual: torch.Tensor, gate: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """residual + gate * x with gate [B, D] broadcast over sequence."""
    if gate.dim() == x.dim() - 1:
        gate = gate.unsqueeze(1)
    return residual + gate * x


def group_norm_silu(
    x: torch.Tensor,
    num_groups: int,
    weight: Optional[torch.Tensor],
    bias: Optional[torch.Tensor],
    eps: float = 1e-6,
) -> torch.Tensor:
    """GroupNorm + SiLU fused (UNet ResBlock prologue)."""
    dtype = x.dtype
    out = F.group_norm(x.float(), num_groups,
                       weight.float() if weight is not None else None,
                       bias.float() if bias is not None else None, eps)
    return F.silu(out).to(dtype)


def rope_freqs(
    positions: torch.Tensor, dim: int, theta: float = 10000.0
) -> torch.Tensor:
    """cos/sin table for rotary embedding.

    positions: [..., S] integer/float positions -> returns [..., S, dim/2, 2]
    stacked (cos, sin) in fp32. Axes composition (2D image RoPE) is done by
    the caller concatenating per-axis tables along the dim/2 axis.
    """
    half = dim // 2
    freqs = torch.arange(0, half, dtype=torch.float32, device=positions.device)
    inv = theta ** (-freqs / half)
    ang = positions.float().unsqueeze(-1) * inv  # [..., S, half]
    return torch.stack([torch.cos(ang), torch.sin(ang)], dim=-1)


def rope_apply(x: torch.Tensor, cs: torch.Tensor) -> torch.Tensor:
    """Apply rotary embedding.

    x: [B, H, S, D]; cs: [S, D/2, 2] or [B, S, D/2, 2] fp32 (cos,sin).
    Pairs are adjacent elements (x0,x1),(x2,x3)...
    """
    dtype = x.dtype
    B, H, S, D = x.shape
    xf = x.float().reshape(B, H, S, D // 2, 2)
    if cs.dim() == 3:
        cos = cs[None, None, :, :, 0]
        sin = cs[None, None, :, :, 1]
    else:
        cos = cs[:, None, :, :, 0]
        sin = cs[:, None, :, :, 1]
    x0, x1 = xf[..., 0], xf[..., 1]
    out = torch.stack([x0 * cos - x1 * sin, x0 * sin + x1 * cos], dim=-1)
    return out.reshape(B, H, S, D).to(dtype)


def attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: Optional[float] = None
) -> torch.Tensor:
    """Plain attention, [B, H, S, D] -> [B, H, S, D]. fp32 softmax."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    dtype = q.dtype
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, v.float()).to(dtype)


def timestep_embedding(
    t: torch.Tensor, dim: int, max_period: float = 10000.0, time_factor: float = 1000.0
) -> torch.Tensor:
    """Sinusoidal timestep embedding, fp32 in/out [B, dim]."""
    t = t.float() * time_factor
    half = dim // 2
    freqs = torch.exp(
        -math.log(max_period)
        * torch.arange(half, dtype=torch.float32, device=t.device)
        / half
    )
    args = t[:, None] * freqs[None]
    emb = torch.cat([torch.cos(args), torch.sin(args)], dim=-1)
    if dim % 2:
        emb = torch.cat([emb, torch.zeros_like(emb[:, :1])], dim=-1)
    return emb
