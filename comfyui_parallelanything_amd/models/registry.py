"""Model registry + synthetic input factories for the BASELINE configs.

Maps the five BASELINE.json configs to (model ctor, synthetic batch maker).
All weights are random-init and inputs synthetic — this environment has no
network for checkpoints; shapes follow each public architecture.
"""
from __future__ import annotations

from typing import Any, Callable, Dict, Tuple

import torch

from .mmdit import Flux, FluxConfig, ZImage, ZImageConfig
from .sd_unet import SDUNet, UNetConfig
from .wan import WanConfig, WanDiT


def _latent_hw(px: int) -> int:
    return px // 8  # SD-family VAE stride


def make_flux(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = FluxConfig.tiny() if tiny else FluxConfig.flux1_dev()
    torch.manual_seed(0)
    with torch.device(dev):
        m = Flux(cfg)
    return m.to(dtype=dtype).eval()


def flux_inputs(batch: int, px: int = 1024, dev="cpu", dtype=torch.bfloat16,
                tiny=False, seed: int = 1234):
    cfg = FluxConfig.tiny() if tiny else FluxConfig.flux1_dev()
    g = torch.Generator(device="cpu").manual_seed(seed)
    hw = _latent_hw(px) if not tiny else 16
    x = torch.randn(batch, cfg.in_channels, hw, hw, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 512 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.context_dim, generator=g).to(dev, dtype)
    y = torch.randn(batch, cfg.vec_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {"y": y}


def make_zimage(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = ZImageConfig.tiny() if tiny else ZImageConfig.z_image_turbo()
    torch.manual_seed(0)
    with torch.device(dev):
        m = ZImage(cfg)
    return m.to(dtype=dtype).eval()


def zimage_inputs(batch: int, px: int = 1024, dev="cpu", dtype=torch.bfloat16,
                  tiny=False, seed: int = 1234):
    cfg = ZImageConfig.tiny() if tiny else ZImageConfig.z_image_turbo()
    g = torch.Generator(device="cpu").manual_seed(seed)
    hw = _latent_hw(px) if not tiny else 16
    x = torch.randn(batch, cfg.in_channels, hw, hw, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 64 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.context_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {}


def make_sd15(dev="cpu", dtype=torch.float32, tiny=False):
    cfg = UNetConfig.tiny() if tiny else UNetConfig.sd15()
    torch.manual_seed(0)
    with torch.device(dev):
        m = SDUNet(cfg)
    return m.to(dtype=dtype).eval()


def sd15_inputs(batch: int, px: int = 256, dev="cpu", dtype=torch.float32,
                tiny=False, seed: int = 1234):
    cfg = UNetConfig.tiny() if tiny else UNetConfig.sd15()
    g = torch.Generator(device="cpu").manual_seed(seed)
    hw = _latent_hw(px) if not tiny else 16
    x = torch.randn(batch, cfg.in_channels, hw, hw, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 77 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.context_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {}


def make_sdxl(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = UNetConfig.tiny() if tiny else UNetConfig.sdxl()
    torch.manual_seed(0)
    with torch.device(dev):
        m = SDUNet(cfg)
    return m.to(dtype=dtype).eval()


def sdxl_inputs(batch: int, px: int = 1024, dev="cpu", dtype=torch.bfloat16,
                tiny=False, seed: int = 1234):
    cfg = UNetConfig.tiny() if tiny else UNetConfig.sdxl()
    g = torch.Generator(device="cpu").manual_seed(seed)
    hw = _latent_hw(px) if not tiny else 16
    x = torch.randn(batch, cfg.in_channels, hw, hw, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    ctx = torch.randn(batch, 77, cfg.context_dim, generator=g).to(dev, dtype)
    kw = {}
    if cfg.adm_in_channels:
        kw["y"] = torch.randn(batch, cfg.adm_in_channels, generator=g).to(dev, dtype)
    return x, t, ctx, kw


def make_sd3(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = FluxConfig.sd35_tiny() if tiny else FluxConfig.sd35_large()
    torch.manual_seed(0)
    with torch.device(dev):
        m = Flux(cfg)
    return m.to(dtype=dtype).eval()


def sd3_inputs(batch: int, px: int = 1024, dev="cpu", dtype=torch.bfloat16,
               tiny=False, seed: int = 1234):
    cfg = FluxConfig.sd35_tiny() if tiny else FluxConfig.sd35_large()
    g = torch.Generator(device="cpu").manual_seed(seed)
    hw = _latent_hw(px) if not tiny else 16
    x = torch.randn(batch, cfg.in_channels, hw, hw, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 154 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.context_dim, generator=g).to(dev, dtype)
    y = torch.randn(batch, cfg.vec_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {"y": y}


def make_wan(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = WanConfig.tiny() if tiny else WanConfig.wan22_a14b()
    torch.manual_seed(0)
    with torch.device(dev):
        m = WanDiT(cfg)
    return m.to(dtype=dtype).eval()


def wan_inputs(batch: int, frames: int = 21, h: int = 90, w: int = 160,
               dev="cpu", dtype=torch.bfloat16, tiny=False, seed: int = 1234):
    cfg = WanConfig.tiny() if tiny else WanConfig.wan22_a14b()
    g = torch.Generator(device="cpu").manual_seed(seed)
    if tiny:
        frames, h, w = 4, 8, 8
    x = torch.randn(batch, cfg.in_channels, frames, h, w, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 512 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.ctx_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {}


def make_wan5b(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = WanConfig.tiny() if tiny else WanConfig.wan22_5b()
    torch.manual_seed(0)
    with torch.device(dev):
        m = WanDiT(cfg)
    return m.to(dtype=dtype).eval()


def wan5b_inputs(batch: int, frames: int = 21, h: int = 90, w: int = 160,
                 dev="cpu", dtype=torch.bfloat16, tiny=False,
                 seed: int = 1234):
    cfg = WanConfig.tiny() if tiny else WanConfig.wan22_5b()
    g = torch.Generator(device="cpu").manual_seed(seed)
    if tiny:
        frames, h, w = 4, 8, 8
    x = torch.randn(batch, cfg.in_channels, frames, h, w, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 512 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.ctx_dim, generator=g).to(dev, dtype)
    return x, t, ctx, {}


def make_wan_i2v(dev="cpu", dtype=torch.bfloat16, tiny=False):
    cfg = WanConfig.tiny_i2v() if tiny else WanConfig.wan22_a14b_i2v()
    torch.manual_seed(0)
    with torch.device(dev):
        m = WanDiT(cfg)
    return m.to(dtype=dtype).eval()


def wan_i2v_inputs(batch: int, frames: int = 21, h: int = 90, w: int = 160,
                   dev="cpu", dtype=torch.bfloat16, tiny=False,
                   seed: int = 1234):
    """WAN2.2 I2V (BASELINE config 5: 720p batch 4): text-to-video inputs
    plus the channel-concatenated image conditioning — first-frame mask +
    reference-image VAE latent, zero on later frames."""
    cfg = WanConfig.tiny_i2v() if tiny else WanConfig.wan22_a14b_i2v()
    g = torch.Generator(device="cpu").manual_seed(seed)
    if tiny:
        frames, h, w = 4, 8, 8
    x = torch.randn(batch, cfg.in_channels, frames, h, w, generator=g).to(dev, dtype)
    t = torch.rand(batch, generator=g).to(dev, torch.float32)
    txt_len = 512 if not tiny else 8
    ctx = torch.randn(batch, txt_len, cfg.ctx_dim, generator=g).to(dev, dtype)
    mask_ch = cfg.cond_channels - cfg.in_channels
    cond = torch.zeros(batch, cfg.cond_channels, frames, h, w)
    cond[:, :mask_ch, 0] = 1.0  # first-frame mask
    cond[:, mask_ch:, 0] = torch.randn(
        batch, cfg.in_channels, h, w, generator=g
    )  # reference-image latent on frame 0
    return x, t, ctx, {"image_cond": cond.to(dev, dtype)}


MODELS: Dict[str, Tuple[Callable, Callable]] = {
    "flux": (make_flux, flux_inputs),
    "zimage": (make_zimage, zimage_inputs),
    "sd15": (make_sd15, sd15_inputs),
    "sdxl": (make_sdxl, sdxl_inputs),
    "sd3": (make_sd3, sd3_inputs),
    "wan": (make_wan, wan_inputs),
    "wan5b": (make_wan5b, wan5b_inputs),
    "wan_i2v": (make_wan_i2v, wan_i2v_inputs),
}
