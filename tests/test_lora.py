"""Headless LoRA merge: key parsing, kohya path resolution, exactness,
invertibility, engine integration (reference bakes patches before
replication, any_device_parallel.py:971-1004)."""
import pytest
import torch

from comfyui_parallelanything_amd.models.lora import (
    load_lora,
    lora_target_names,
    merge_lora,
    merge_lora_file,
)
from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux


def _lora_for(w_out, w_in, r=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    down = torch.randn(r, w_in, generator=g) * 0.05
    up = torch.randn(w_out, r, generator=g) * 0.05
    return down, up


def test_merge_peft_keys_exact():
    m = make_flux(tiny=True, dtype=torch.float32)
    lin = m.double_blocks[0].img_attn_qkv
    lin = lin if isinstance(lin, torch.nn.Linear) else lin.lin
    down, up = _lora_for(*lin.weight.shape)
    w0 = lin.weight.clone()
    sd = {
        "double_blocks.0.img_attn_qkv.lora_A.weight": down,
        "double_blocks.0.img_attn_qkv.lora_B.weight": up,
    }
    assert merge_lora(m, sd, scale=1.0) == 1
    expected = w0 + (up.float() @ down.float()) * (4.0 / 4.0)
    torch.testing.assert_close(lin.weight, expected)


def test_merge_kohya_underscore_and_alpha():
    m = make_flux(tiny=True, dtype=torch.float32)
    lin = m.double_blocks[1].txt_attn_proj
    down, up = _lora_for(*lin.weight.shape, r=2)
    w0 = lin.weight.clone()
    sd = {
        "lora_unet_double_blocks_1_txt_attn_proj.lora_down.weight": down,
        "lora_unet_double_blocks_1_txt_attn_proj.lora_up.weight": up,
        "lora_unet_double_blocks_1_txt_attn_proj.alpha": torch.tensor(1.0),
    }
    assert merge_lora(m, sd) == 1
    expected = w0 + (up.float() @ down.float()) * (1.0 / 2.0)
    torch.testing.assert_close(lin.weight, expected)


def test_merge_unmerge_roundtrip_changes_forward():
    m = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(1, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    lin = m.double_blocks[0].img_attn_qkv
    lin = lin if isinstance(lin, torch.nn.Linear) else lin.lin
    down, up = _lora_for(*lin.weight.shape)
    sd = {
        "double_blocks.0.img_attn_qkv.lora_A.weight": down,
        "double_blocks.0.img_attn_qkv.lora_B.weight": up,
    }
    merge_lora(m, sd, scale=1.0)
    out = m(x, t, context=c, **kw)
    assert not torch.allclose(out, ref)  # LoRA took effect
    merge_lora(m, sd, scale=-1.0)  # unmerge
    out2 = m(x, t, context=c, **kw)
    torch.testing.assert_close(out2, ref, rtol=1e-4, atol=1e-5)


def test_unknown_and_mismatched_keys_skipped():
    m = make_flux(tiny=True, dtype=torch.float32)
    sd = {
        "nonexistent.module.lora_A.weight": torch.zeros(2, 8),
        "nonexistent.module.lora_B.weight": torch.zeros(8, 2),
        # resolves but wrong shapes:
        "double_blocks.0.img_attn_proj.lora_A.weight": torch.zeros(2, 3),
        "double_blocks.0.img_attn_proj.lora_B.weight": torch.zeros(5, 2),
    }
    assert merge_lora(m, sd) == 0


def test_target_names_diagnostic():
    m = make_flux(tiny=True, dtype=torch.float32)
    lin = m.double_blocks[0].img_attn_proj
    down, up = _lora_for(*lin.weight.shape)
    sd = {
        "double_blocks.0.img_attn_proj.lora_A.weight": down,
        "double_blocks.0.img_attn_proj.lora_B.weight": up,
        "bogus.lora_A.weight": down,
        "bogus.lora_B.weight": up,
    }
    assert lora_target_names(m, sd) == ["double_blocks.0.img_attn_proj"]


def test_file_roundtrip(tmp_path):
    from safetensors.torch import save_file

    m = make_flux(tiny=True, dtype=torch.float32)
    lin = m.single_blocks[0].linear2_attn
    lin = lin if isinstance(lin, torch.nn.Linear) else lin.lin
    down, up = _lora_for(*lin.weight.shape)
    p = str(tmp_path / "l.safetensors")
    save_file({
        "single_blocks.0.linear2_attn.lora_A.weight": down.contiguous(),
        "single_blocks.0.linear2_attn.lora_B.weight": up.contiguous(),
    }, p)
    assert len(load_lora(p)) == 2
    assert merge_lora_file(m, p, scale=0.5) == 1


def test_merged_weights_replicate_through_engine():
    """Merge-then-setup: every replica carries the patched weights
    (reference clone-after-patch invariant)."""
    from comfyui_parallelanything_amd.parallel.chain import (
        DeviceChain,
        make_entry,
    )
    from comfyui_parallelanything_amd.parallel.engine import ParallelEngine

    m = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(2, tiny=True, dtype=torch.float32)
    lin = m.double_blocks[0].img_attn_qkv
    lin = lin if isinstance(lin, torch.nn.Linear) else lin.lin
    down, up = _lora_for(*lin.weight.shape)
    sd = {
        "double_blocks.0.img_attn_qkv.lora_A.weight": down,
        "double_blocks.0.img_attn_qkv.lora_B.weight": up,
    }
    merge_lora(m, sd)
    ref = m(x, t, context=c, **kw)
    eng = ParallelEngine(
        DeviceChain.from_list([make_entry("cpu", 50), make_entry("cpu", 50)]),
        auto_vram_balance=False,
    )
    eng.setup(m, force_copy_lead=True)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
