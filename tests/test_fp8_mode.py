"""FP8 serving-mode tests (gfx950 _scaled_mm path)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_fp8_linear_matches_bf16():
    from comfyui_parallelanything_amd.models.quant import FP8Linear, _supports_scaled_mm

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm on this build")
    torch.manual_seed(0)
    lin = torch.nn.Linear(1024, 2048).cuda().to(torch.bfloat16)
    q = FP8Linear.from_linear(lin)
    x = torch.randn(64, 1024, device="cuda", dtype=torch.bfloat16)
    ref = lin(x).float()
    out = q(x).float()
    # fp8 per-tensor quant: compare relative error magnitude
    rel = (out - ref).norm() / ref.norm()
    assert rel < 0.06, f"fp8 relative error too high: {rel:.4f}"


def test_quantize_fp8_model_runs():
    from comfyui_parallelanything_amd.models.quant import quantize_fp8, _supports_scaled_mm
    from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm on this build")
    m = make_flux(dev="cuda", dtype=torch.bfloat16, tiny=True)
    # tiny config has small dims; lower the threshold to exercise the swap
    n = quantize_fp8(m, min_features=32)
    assert n > 0
    x, t, c, kw = flux_inputs(2, dev="cuda", dtype=torch.bfloat16, tiny=True)
    with torch.no_grad():
        out = m(x, t, context=c, **kw)
    assert torch.isfinite(out.float()).all()
