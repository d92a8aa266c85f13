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
