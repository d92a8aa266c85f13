import pytest
import torch

from comfyui_parallelanything_amd.models.checkpoint import (
    load_checkpoint,
    save_checkpoint,
)
from comfyui_parallelanything_amd.models.registry import (
    flux_inputs,
    make_flux,
    make_sd15,
    sd15_inputs,
)


def test_roundtrip_flux(tmp_path):
    m = make_flux(tiny=True, dtype=torch.float32)
    path = str(tmp_path / "flux_tiny")
    save_checkpoint(m, path)
    m2 = load_checkpoint(path)
    x, t, c, kw = flux_inputs(2, tiny=True, dtype=torch.float32)
    torch.testing.assert_close(
        m(x, t, context=c, **kw), m2(x, t, context=c, **kw)
    )


def test_roundtrip_unet_with_dtype_cast(tmp_path):
    m = make_sd15(tiny=True)
    path = str(tmp_path / "sd15_tiny")
    save_checkpoint(m, path)
    m2 = load_checkpoint(path, dtype=torch.float32)
    for (k1, p1), (k2, p2) in zip(
        m.state_dict().items(), m2.state_dict().items()
    ):
        assert k1 == k2
        torch.testing.assert_close(p1.float(), p2.float())


def test_unknown_class_raises(tmp_path):
    import json

    with open(tmp_path / "bad.json", "w") as f:
        json.dump({"class": "NotAModel", "config": {}}, f)
    with pytest.raises(ValueError, match="unknown checkpoint class"):
        load_checkpoint(str(tmp_path / "bad"))


def test_replica_cache_cleared():
    from comfyui_parallelanything_amd.parallel.replicate import replicate_module

    m = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(1, tiny=True, dtype=torch.float32)
    m(x, t, context=c, **kw)  # populate _pe_cache
    assert m._pe_cache
    r = replicate_module(m, "cpu", force_copy=True)
    assert not r._pe_cache  # replica starts with a clean per-device cache
    assert m._pe_cache  # source untouched
