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


def test_quant_fp8_kernel():
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("quant_fp8"):
        pytest.skip("no quant_fp8 in extension")
    torch.manual_seed(0)
    x = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16) * 3
    scale = torch.tensor([x.abs().amax().item() / 448.0], device="cuda")
    amax = torch.zeros(1, device="cuda")
    x8 = ops.quant_fp8(x, scale, amax)
    assert x8.dtype == torch.float8_e4m3fn
    ref = (x.float() / scale).clamp(-448, 448).to(torch.float8_e4m3fn)
    # compare dequantized values
    torch.testing.assert_close(x8.float(), ref.float(), rtol=0, atol=0)
    # fused amax matches the true amax
    assert abs(amax.item() - x.abs().amax().item()) < 1e-3


def test_fp8_linear_delayed_scaling_stable():
    from comfyui_parallelanything_amd.models.quant import FP8Linear, _supports_scaled_mm

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    torch.manual_seed(1)
    lin = torch.nn.Linear(1024, 1024).cuda().to(torch.bfloat16)
    q = FP8Linear.from_linear(lin)
    x = torch.randn(64, 1024, device="cuda", dtype=torch.bfloat16)
    ref = lin(x).float()
    outs = [q(x).float() for _ in range(4)]  # delayed scale warm + steady
    for out in outs[1:]:
        rel = (out - ref).norm() / ref.norm()
        assert rel < 0.06, f"fp8 delayed-scale error {rel:.4f}"
    # the kernel-maintained scale tracks amax/448 (decayed) and the block
    # counter (slot 1) is reset after every call
    amax = q.a_amax[0].item()
    assert q.a_amax[1].item() == 0.0
    assert abs(q.x_scale.item() - max(amax / 448.0, 1e-12)) < 1e-6


def test_quant_fp8_fused_scale_epilogue():
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("quant_fp8"):
        pytest.skip("no quant_fp8 in extension")
    torch.manual_seed(2)
    x = torch.randn(512, 2048, device="cuda", dtype=torch.bfloat16) * 2
    true_amax = x.abs().amax().item()
    scale = torch.tensor([true_amax / 448.0], device="cuda")
    amax = torch.zeros(2, device="cuda")  # slot 1 = kernel block counter
    x8 = ops.quant_fp8(x, scale, amax)
    ref = (x.float() / (true_amax / 448.0)).clamp(-448, 448).to(
        torch.float8_e4m3fn
    )
    # kernel multiplies by the reciprocal; at fp8 rounding boundaries that
    # can land one quantum off the divide — compare within one ULP of fp8
    diff = (x8.float() - ref.float()).abs()
    ulp = torch.maximum(ref.float().abs() * 2 ** -3,
                        torch.full_like(diff, 2 ** -9))
    assert (diff <= ulp).all(), f"max diff {diff.max().item()}"
    assert (diff > 0).float().mean().item() < 0.001  # almost all exact
    # epilogue: amax decayed once, scale rewritten for the next call,
    # counter back to zero
    assert abs(amax[0].item() - true_amax * 0.999) < 1e-3
    assert amax[1].item() == 0.0
    assert abs(scale.item() - true_amax * 0.999 / 448.0) < 1e-6


def test_quant_fp8_scale_used_snapshot():
    """The epilogue must snapshot the scale it quantized WITH into
    scale_used before overwriting scale with the next call's value."""
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("quant_fp8"):
        pytest.skip("no quant_fp8 in extension")
    torch.manual_seed(3)
    x = torch.randn(128, 1024, device="cuda", dtype=torch.bfloat16)
    # deliberately NOT amax/448 (so next-scale differs from used) but big
    # enough that 448*s0 > randn's amax — no clamping distorts the deq check
    s0 = 0.012
    scale = torch.tensor([s0], device="cuda")
    s0_f32 = scale.item()  # the fp32 value the kernel actually reads
    amax = torch.zeros(2, device="cuda")
    used = torch.zeros(1, device="cuda")
    x8 = ops.quant_fp8(x, scale, amax, scale_used=used)
    assert used.item() == s0_f32, "scale_used must be the entry scale bits"
    true_amax = x.abs().amax().item()
    assert abs(scale.item() - true_amax * 0.999 / 448.0) < 1e-6
    # dequantizing with the USED scale reproduces x (modulo fp8 rounding)
    deq = x8.float() * used
    rel = (deq - x.float()).norm() / x.float().norm()
    assert rel < 0.05


def test_fp8_linear_amax_jump_uses_snapshot_scale():
    """Activation amplitude jumping between calls must not corrupt the
    output: _scaled_mm dequantizes with the scale actually used, not the
    next-call scale the kernel epilogue writes (the aliased-buffer bug
    multiplied the output by next_scale/used_scale)."""
    from comfyui_parallelanything_amd.models.quant import (
        FP8Linear, _supports_scaled_mm,
    )

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    torch.manual_seed(4)
    lin = torch.nn.Linear(1024, 1024).cuda().to(torch.bfloat16)
    q = FP8Linear.from_linear(lin)
    x1 = torch.randn(64, 1024, device="cuda", dtype=torch.bfloat16)
    q(x1)  # warm: sets the delayed scale from x1's amax
    # 1.5x amplitude: mild clamping only (inherent delayed-scaling cost is
    # small) but with the bug the output comes back ~1.5x too large.
    x2 = x1 * 1.5
    out = q(x2).float()
    ref = lin(x2).float()
    rel = (out - ref).norm() / ref.norm()
    assert rel < 0.10, f"fp8 amax-jump error {rel:.4f}"


def test_layer_norm_mod_fp8_fused():
    """Fused AdaLN+fp8 kernel == layer_norm_mod -> quantize, and maintains
    the same delayed-scaling state contract as quant_fp8."""
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("layer_norm_mod_fp8"):
        pytest.skip("no layer_norm_mod_fp8 in extension")
    torch.manual_seed(5)
    B, S, D = 2, 64, 1024
    x = torch.randn(B, S, D, device="cuda", dtype=torch.bfloat16)
    sc = torch.randn(B, D, device="cuda", dtype=torch.bfloat16) * 0.1
    sh = torch.randn(B, D, device="cuda", dtype=torch.bfloat16) * 0.1
    ln = ops.layer_norm_mod(x, sc, sh)
    true_amax = ln.float().abs().amax().item()
    s0 = true_amax / 448.0
    qscale = torch.tensor([s0], device="cuda")
    amax = torch.zeros(2, device="cuda")
    used = torch.zeros(1, device="cuda")
    x8 = ops.layer_norm_mod_fp8(x, sc, sh, qscale, amax, used)
    assert x8.dtype == torch.float8_e4m3fn and x8.shape == x.shape
    deq = x8.float() * used
    rel = (deq - ln.float()).norm() / ln.float().norm()
    assert rel < 0.05, f"fused LN+quant error {rel:.4f}"
    # delayed-scaling contract: amax tracked, counter reset, next scale
    assert abs(amax[0].item() - true_amax * 0.999) < 2e-2
    assert amax[1].item() == 0.0
    assert used.item() == qscale.new_tensor([s0]).item()


def test_fp8_linear_forward_ln_matches_composed():
    """FP8Linear.forward_ln (fused LN+quant) ~= LN then FP8Linear."""
    from comfyui_parallelanything_amd import ops
    from comfyui_parallelanything_amd.models.quant import (
        FP8Linear, _supports_scaled_mm,
    )

    if not _supports_scaled_mm() or not ops.hip_available("layer_norm_mod_fp8"):
        pytest.skip("no fp8 fused-LN path")
    torch.manual_seed(6)
    lin = torch.nn.Linear(1024, 1024).cuda().to(torch.bfloat16)
    q1 = FP8Linear.from_linear(lin)
    q2 = FP8Linear.from_linear(lin)
    x = torch.randn(2, 64, 1024, device="cuda", dtype=torch.bfloat16)
    sc = torch.randn(2, 1024, device="cuda", dtype=torch.bfloat16) * 0.1
    sh = torch.randn(2, 1024, device="cuda", dtype=torch.bfloat16) * 0.1
    fused = q1.forward_ln(x, sc, sh).float()
    composed = q2(ops.layer_norm_mod(x, sc, sh)).float()
    rel = (fused - composed).norm() / composed.norm()
    assert rel < 0.02, f"fused vs composed {rel:.4f}"


def test_fp8_model_uses_fused_ln(monkeypatch):
    """quantized flux-tiny routes through ln_quant (no standalone quant
    pass on the block LN sites) and stays finite."""
    from comfyui_parallelanything_amd import ops
    from comfyui_parallelanything_amd.models.quant import (
        FP8Linear, _supports_scaled_mm, quantize_fp8,
    )
    from comfyui_parallelanything_amd.models.registry import (
        flux_inputs, make_flux,
    )

    if not _supports_scaled_mm() or not ops.hip_available("layer_norm_mod_fp8"):
        pytest.skip("no fp8 fused-LN path")
    m = make_flux(dev="cuda", dtype=torch.bfloat16, tiny=True)
    quantize_fp8(m, min_features=32)
    calls = {"n": 0}
    orig = FP8Linear.ln_quant

    def spy(self, *a, **k):
        calls["n"] += 1
        return orig(self, *a, **k)

    monkeypatch.setattr(FP8Linear, "ln_quant", spy)
    x, t, c, kw = flux_inputs(2, dev="cuda", dtype=torch.bfloat16, tiny=True)
    with torch.no_grad():
        out = m(x, t, context=c, **kw)
    assert torch.isfinite(out.float()).all()
    assert calls["n"] > 0, "fused LN+quant path not taken"


def test_fnuz_to_ocp_reencode_on_gpu():
    """fnuz -> OCP re-encode on DEVICE tensors (round-1 gap: CPU-only
    coverage). Values must survive the format change (different exponent
    bias makes a bit-cast wrong)."""
    from comfyui_parallelanything_amd.parallel.fp8 import (
        sanitize_param_dtype, to_ocp_fp8,
    )

    if not hasattr(torch, "float8_e4m3fnuz"):
        pytest.skip("no fnuz dtype in this torch")
    torch.manual_seed(7)
    vals = torch.randn(256, device="cuda").clamp(-8, 8)
    fnuz = vals.to(torch.float8_e4m3fnuz)
    ocp = to_ocp_fp8(fnuz)
    assert ocp.dtype == torch.float8_e4m3fn and ocp.is_cuda
    # both encodings decode to (approximately) the same reals
    torch.testing.assert_close(
        ocp.float(), fnuz.float(), rtol=0.07, atol=0.02
    )
    # sanitize keeps fp8 on GPU (gfx950 native) and normalizes the format
    s = sanitize_param_dtype(fnuz, "cuda:0")
    assert s.dtype == torch.float8_e4m3fn
    # replication of a module carrying an fnuz buffer lands OCP on device
    from comfyui_parallelanything_amd.parallel.replicate import (
        replicate_module,
    )

    m = torch.nn.Linear(8, 8).cuda().to(torch.bfloat16)
    m.register_buffer("w8", fnuz.clone())
    rep = replicate_module(m, "cuda:0", force_copy=True)
    assert rep.w8.dtype == torch.float8_e4m3fn and rep.w8.is_cuda


def test_gelu_fp8_fused():
    """Fused tanh-GELU + e4m3 quant == gelu then quantize, same
    delayed-scaling contract."""
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("gelu_fp8"):
        pytest.skip("no gelu_fp8 in extension")
    torch.manual_seed(8)
    y = torch.randn(64, 2048, device="cuda", dtype=torch.bfloat16) * 2
    g = torch.nn.functional.gelu(y.float(), approximate="tanh")
    true_amax = g.abs().amax().item()
    scale = torch.tensor([true_amax / 448.0], device="cuda")
    amax = torch.zeros(2, device="cuda")
    used = torch.zeros(1, device="cuda")
    x8 = ops.gelu_fp8(y, scale, amax, used)
    deq = x8.float() * used
    rel = (deq - g).norm() / g.norm()
    assert rel < 0.05, f"fused gelu+quant error {rel:.4f}"
    assert abs(amax[0].item() - true_amax * 0.999) < 2e-2
    assert amax[1].item() == 0.0


def test_fp8_fusedmlp_uses_gelu_fp8(monkeypatch):
    """Quantized FusedMLP routes GELU through the fused gelu+quant path
    and matches the composed fp8 computation."""
    from comfyui_parallelanything_amd import ops
    from comfyui_parallelanything_amd.models.layers import FusedMLP
    from comfyui_parallelanything_amd.models.quant import (
        FP8Linear, _supports_scaled_mm, quantize_fp8,
    )

    if not _supports_scaled_mm() or not ops.hip_available("gelu_fp8"):
        pytest.skip("no fp8 gelu path")
    torch.manual_seed(9)
    m = FusedMLP(256, 1024, 256).cuda().to(torch.bfloat16)
    x = torch.randn(2, 32, 256, device="cuda", dtype=torch.bfloat16)
    ref = m(x).float()  # bf16 path before quantization
    quantize_fp8(m, min_features=256)
    assert isinstance(m.down, FP8Linear)
    calls = {"n": 0}
    orig = FP8Linear.gelu_quant

    def spy(self, *a, **k):
        calls["n"] += 1
        return orig(self, *a, **k)

    monkeypatch.setattr(FP8Linear, "gelu_quant", spy)
    out = m(x).float()
    assert calls["n"] > 0, "fused gelu+quant path not taken"
    rel = (out - ref).norm() / ref.norm()
    assert rel < 0.08, f"fp8 FusedMLP error {rel:.4f}"


def test_fp8_whole_model_error_vs_bf16():
    """Documented error bound of the fp8 serving mode: a quantized
    flux-tiny forward stays within rel-l2 0.15 of the bf16 model and
    correlates >0.99 — the bound the opt-in mode is sold under."""
    from comfyui_parallelanything_amd.models.quant import (
        _supports_scaled_mm, quantize_fp8,
    )
    from comfyui_parallelanything_amd.models.registry import (
        flux_inputs, make_flux,
    )

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    m = make_flux(dev="cuda", dtype=torch.bfloat16, tiny=True)
    x, t, c, kw = flux_inputs(2, dev="cuda", dtype=torch.bfloat16, tiny=True)
    with torch.no_grad():
        ref = m(x, t, context=c, **kw).float().clone()
        quantize_fp8(m, min_features=32)
        for _ in range(3):  # settle the delayed scales
            out = m(x, t, context=c, **kw).float()
    rel = (out - ref).norm() / ref.norm()
    assert rel < 0.15, f"fp8 whole-model rel-l2 {rel:.4f}"
    corr = torch.corrcoef(
        torch.stack([out.flatten(), ref.flatten()])
    )[0, 1].item()
    assert corr > 0.99, f"fp8 correlation {corr:.4f}"


def test_fp8_quant_kernels_deterministic():
    """Run-to-run bitwise determinism of the delayed-scaling kernels:
    the conditional atomicMax keeps amax order-independent and the
    finalize kernel is stream-ordered — identical inputs and state must
    give identical outputs, scales, and amax."""
    from comfyui_parallelanything_amd import ops

    if not ops.hip_available("quant_fp8"):
        pytest.skip("no quant_fp8 in extension")
    torch.manual_seed(11)
    x = torch.randn(256, 3072, device="cuda", dtype=torch.bfloat16)
    sc = torch.randn(1, 3072, device="cuda", dtype=torch.bfloat16) * 0.1
    sh = torch.randn(1, 3072, device="cuda", dtype=torch.bfloat16) * 0.1

    def run_all():
        s1 = torch.tensor([0.01], device="cuda")
        a1 = torch.zeros(2, device="cuda")
        u1 = torch.zeros(1, device="cuda")
        q8 = ops.quant_fp8(x, s1, a1, scale_used=u1)
        s2 = torch.tensor([0.01], device="cuda")
        a2 = torch.zeros(2, device="cuda")
        u2 = torch.zeros(1, device="cuda")
        g8 = ops.gelu_fp8(x, s2, a2, u2)
        s3 = torch.tensor([0.01], device="cuda")
        a3 = torch.zeros(2, device="cuda")
        u3 = torch.zeros(1, device="cuda")
        l8 = ops.layer_norm_mod_fp8(x.unsqueeze(0), sc, sh, s3, a3, u3)
        return [t.view(torch.uint8).cpu() for t in (q8, g8, l8)] + [
            torch.cat([s1, a1, u1, s2, a2, u2, s3, a3, u3]).cpu()
        ]

    a = run_all()
    b = run_all()
    for ta, tb in zip(a, b):
        assert torch.equal(ta, tb), "fp8 kernel nondeterminism"


def test_fp8_linear_cpu_input_raises():
    """fp8 mode is GPU-only: a cpu input (e.g. a cpu chain device after
    quantize_fp8) must fail loudly, not corrupt through _scaled_mm."""
    from comfyui_parallelanything_amd.models.quant import (
        FP8Linear, _supports_scaled_mm,
    )

    if not _supports_scaled_mm():
        pytest.skip("no fp8 _scaled_mm")
    lin = torch.nn.Linear(64, 64).cuda().to(torch.bfloat16)
    q = FP8Linear.from_linear(lin)
    with pytest.raises(RuntimeError, match="GPU-only"):
        q(torch.randn(4, 64, dtype=torch.bfloat16))
