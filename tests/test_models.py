"""Shape/finiteness checks for every model family at tiny scale (CPU)."""
import pytest
import torch

from comfyui_parallelanything_amd.models.registry import MODELS


@pytest.mark.parametrize("name", sorted(MODELS))
def test_tiny_forward_shapes(name):
    make, inputs = MODELS[name]
    m = make(tiny=True, dtype=torch.float32)
    x, t, c, kw = inputs(2, tiny=True, dtype=torch.float32)
    out = m(x, t, context=c, **kw)
    assert out.shape == x.shape
    assert torch.isfinite(out).all()


@pytest.mark.parametrize("name", sorted(MODELS))
def test_deterministic(name):
    make, inputs = MODELS[name]
    m = make(tiny=True, dtype=torch.float32)
    x, t, c, kw = inputs(2, tiny=True, dtype=torch.float32)
    o1 = m(x, t, context=c, **kw)
    o2 = m(x, t, context=c, **kw)
    assert torch.equal(o1, o2)


def test_flux_block_lists_present():
    from comfyui_parallelanything_amd.models.mmdit import Flux, FluxConfig

    m = Flux(FluxConfig.tiny())
    assert len(m.double_blocks) == 2 and len(m.single_blocks) == 2


def test_flux_full_config_shapes():
    from comfyui_parallelanything_amd.models.mmdit import FluxConfig

    cfg = FluxConfig.flux1_dev()
    assert cfg.hidden == 3072 and cfg.num_heads == 24
    assert cfg.depth_double == 19 and cfg.depth_single == 38
    assert sum(cfg.axes_dim) == cfg.hidden // cfg.num_heads  # rope covers head dim


def test_param_counts_in_range():
    """Full-size configs must be the real model class (guards against
    accidentally benchmarking a toy)."""
    from comfyui_parallelanything_amd.models.mmdit import Flux, FluxConfig

    with torch.device("meta"):
        m = Flux(FluxConfig.flux1_dev())
    n = sum(p.numel() for p in m.parameters())
    assert 10e9 < n < 14e9, f"FLUX-class param count off: {n/1e9:.2f}B"


def test_wan_param_count():
    from comfyui_parallelanything_amd.models.wan import WanConfig, WanDiT

    with torch.device("meta"):
        m = WanDiT(WanConfig.wan22_a14b())
    n = sum(p.numel() for p in m.parameters())
    assert 12e9 < n < 18e9, f"WAN-class param count off: {n/1e9:.2f}B"


def test_zimage_param_count():
    from comfyui_parallelanything_amd.models.mmdit import ZImage, ZImageConfig

    with torch.device("meta"):
        m = ZImage(ZImageConfig.z_image_turbo())
    n = sum(p.numel() for p in m.parameters())
    assert 4e9 < n < 8e9, f"Z-Image-class param count off: {n/1e9:.2f}B"


def test_sdxl_param_count():
    from comfyui_parallelanything_amd.models.sd_unet import SDUNet, UNetConfig

    with torch.device("meta"):
        m = SDUNet(UNetConfig.sdxl())
    n = sum(p.numel() for p in m.parameters())
    assert 2.0e9 < n < 3.5e9, f"SDXL-class param count off: {n/1e9:.2f}B"
