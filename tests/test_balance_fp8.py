import pytest
import torch

from comfyui_parallelanything_amd.parallel import balance
from comfyui_parallelanything_amd.parallel import fp8


def test_no_gpu_chain_uses_static_weights():
    sizes = balance.auto_split_batch(10, ["cpu", "cpu"], [0.7, 0.3])
    assert sizes == [7, 3]


def test_blend_policy(monkeypatch):
    # two "GPUs": free VRAM 3000 vs 1000 MB; user 50/50
    vram = {"cuda:0": 3000.0, "cuda:1": 1000.0}
    monkeypatch.setattr(balance, "get_free_vram_mb", lambda d: vram[d])
    ws = balance.vram_blended_weights(["cuda:0", "cuda:1"], [0.5, 0.5])
    # 0.7*0.5 + 0.3*0.75 = 0.575 ; 0.7*0.5 + 0.3*0.25 = 0.425 (already normalized)
    assert ws[0] == pytest.approx(0.575)
    assert ws[1] == pytest.approx(0.425)
    assert sum(ws) == pytest.approx(1.0)


def test_blend_skips_cpu(monkeypatch):
    vram = {"cuda:0": 2000.0}
    monkeypatch.setattr(
        balance, "get_free_vram_mb", lambda d: vram.get(d, 0.0)
    )
    ws = balance.vram_blended_weights(["cuda:0", "cpu"], [0.5, 0.5])
    # cuda gets 0.7*0.5+0.3*1.0 = 0.65, cpu stays 0.5 -> normalize
    assert ws[0] == pytest.approx(0.65 / 1.15)
    assert ws[1] == pytest.approx(0.5 / 1.15)


def test_free_vram_cpu_is_zero():
    assert balance.get_free_vram_mb("cpu") == 0.0


# --- fp8 policy ---

def test_is_float8_detects_all_variants():
    assert fp8.is_float8_dtype(torch.float8_e4m3fn)
    assert fp8.is_float8_dtype(torch.float8_e5m2)
    if hasattr(torch, "float8_e4m3fnuz"):
        assert fp8.is_float8_dtype(torch.float8_e4m3fnuz)
    assert not fp8.is_float8_dtype(torch.bfloat16)


def test_cpu_gets_fp16_upcast():
    t = torch.zeros(4, dtype=torch.float8_e4m3fn)
    out = fp8.sanitize_param_dtype(t, "cpu")
    assert out.dtype == torch.float16


def test_fnuz_reencode_values():
    if not hasattr(torch, "float8_e4m3fnuz"):
        pytest.skip("no fnuz dtype in this torch")
    vals = torch.tensor([0.5, -1.0, 2.0, 0.0])
    fnuz = vals.to(torch.float8_e4m3fnuz)
    ocp = fp8.to_ocp_fp8(fnuz)
    assert ocp.dtype == torch.float8_e4m3fn
    # representable values survive the round trip
    assert torch.equal(ocp.float(), vals)


def test_gpu_policy_keeps_fp8():
    if not torch.cuda.is_available():
        # device_supports_float8 is a pure string check; safe without a GPU
        assert fp8.device_supports_float8("cuda:0")
    t = torch.zeros(4, dtype=torch.float8_e4m3fn)
    out = fp8.sanitize_param_dtype(t, "cuda:0")
    assert out.dtype == torch.float8_e4m3fn


def test_adaptive_balancer_shifts_load():
    from comfyui_parallelanything_amd.parallel.balance import AdaptiveBalancer

    b = AdaptiveBalancer(["a", "b"], [0.5, 0.5], blend=0.5, ema=1.0)
    assert b.weights() == [0.5, 0.5]  # no data yet -> user weights
    # device a is 3x faster
    for _ in range(3):
        b.record("a", 6, 1.0)
        b.record("b", 2, 1.0)
    ws = b.weights()
    # 0.5*0.5 + 0.5*0.75 = 0.625 vs 0.375
    assert ws[0] == pytest.approx(0.625)
    sizes = b.split(8)
    assert sizes[0] == 5 and sizes[1] == 3


def test_adaptive_balancer_in_engine():
    import torch

    from comfyui_parallelanything_amd.models.registry import make_sd15, sd15_inputs
    from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
    from comfyui_parallelanything_amd.parallel.engine import ParallelEngine

    m = make_sd15(tiny=True)
    eng = ParallelEngine(
        DeviceChain.from_list([make_entry("cpu", 50), make_entry("cpu", 50)]),
        auto_vram_balance=True,
    )
    eng.setup(m)
    x, t, c, kw = sd15_inputs(4, tiny=True)
    ref = m(x, t, context=c, **kw)
    for _ in range(3):
        out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
    assert eng.balancer is not None
