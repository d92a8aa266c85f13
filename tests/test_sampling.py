import math

import pytest
import torch

from comfyui_parallelanything_amd.sampling import (
    SAMPLERS,
    flow_sigmas,
    karras_sigmas,
    sample_dpmpp_2m,
    sample_flow_euler,
    sample_flow_heun,
)


def test_flow_sigmas_endpoints():
    s = flow_sigmas(10)
    assert s[0] == 1.0 and s[-1] == 0.0 and len(s) == 11
    assert (s[:-1] > s[1:]).all()  # monotone decreasing


def test_flow_sigmas_shift():
    s = flow_sigmas(10, shift=3.0)
    assert s[0] == pytest.approx(1.0) and s[-1] == pytest.approx(0.0)
    # shift > 1 pushes mass toward high sigma
    assert s[5] > flow_sigmas(10)[5]


def test_karras_sigmas():
    s = karras_sigmas(20)
    assert len(s) == 21 and s[-1] == 0
    assert s[0] == pytest.approx(14.61, rel=1e-3)


def test_flow_euler_exact_for_linear_field():
    # dx/ds = v = c (constant): exact for Euler regardless of step count
    c = torch.randn(2, 3)

    def model(x, t, context=None):
        return c.expand_as(x)

    x0 = torch.randn(2, 3)
    out = sample_flow_euler(model, x0.clone(), flow_sigmas(7))
    # integral from s=1 to 0 of c ds = -c
    torch.testing.assert_close(out, x0 - c.expand_as(x0), rtol=1e-5, atol=1e-6)


def test_flow_heun_converges_faster_than_euler():
    # dx/ds = -x has exact solution x(0) = x(1) * e^{1}
    def model(x, t, context=None):
        return -x

    x0 = torch.ones(1, 1)
    exact = x0 * math.e
    err = {}
    for name, fn in (("euler", sample_flow_euler), ("heun", sample_flow_heun)):
        out = fn(model, x0.clone(), flow_sigmas(16))
        err[name] = (out - exact).abs().item()
    assert err["heun"] < err["euler"] * 0.2


def test_dpmpp_2m_recovers_clean_signal():
    # model that perfectly predicts the noise: eps = (x - x_clean)/sigma
    x_clean = torch.randn(2, 4)
    sigmas = karras_sigmas(12)

    def model(x, t, context=None):
        sig = t[0].item()
        return (x - x_clean) / sig

    x = x_clean + sigmas[0] * torch.randn(2, 4)
    out = sample_dpmpp_2m(model, x, sigmas)
    torch.testing.assert_close(out, x_clean, rtol=1e-3, atol=1e-3)


def test_samplers_with_tiny_model():
    from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux

    m = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(2, tiny=True, dtype=torch.float32)
    for name in ("euler", "heun"):
        out = SAMPLERS[name](m, x.clone(), flow_sigmas(3), context=c, **kw)
        assert out.shape == x.shape and torch.isfinite(out).all()


def test_end_to_end_denoise_with_engine():
    """Full sampler loop through the parallel-installed model on [cpu,cpu]
    matches the single-device loop exactly (BASELINE config 1, end to end)."""
    from comfyui_parallelanything_amd.models.registry import make_sd15, sd15_inputs
    from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
    from comfyui_parallelanything_amd.parallel.engine import (
        ParallelEngine,
        install_parallel_forward,
    )
    from comfyui_parallelanything_amd.sampling import karras_sigmas, sample_dpmpp_2m

    m = make_sd15(tiny=True)
    x, t, c, kw = sd15_inputs(2, tiny=True)
    sig = karras_sigmas(6)
    ref = sample_dpmpp_2m(m, x.clone(), sig, context=c, **kw)

    eng = ParallelEngine(
        DeviceChain.from_list([make_entry("cpu", 50), make_entry("cpu", 50)]),
        auto_vram_balance=False,
    )
    eng.setup(m)
    install_parallel_forward(m, eng)
    out = sample_dpmpp_2m(m, x.clone(), sig, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
