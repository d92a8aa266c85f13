"""Layer-level equivalence tests for the fused-dispatch helpers.

The round-2 fp8 LN fusion rerouted block call sites through
forward_ln/ln_mod_into/forward_timestep. On CPU (and on GPU outside fp8
mode) these MUST be bitwise identical to the composed ops they replace —
the whole-model golden tests compare DP-vs-single through the SAME code
path, so only these direct comparisons catch a swapped scale/shift or a
wrong sub-layer."""
import torch

from comfyui_parallelanything_amd import ops
from comfyui_parallelanything_amd.models.layers import (
    FusedMLP,
    GELULinear,
    MLPEmbedder,
    ln_mod_into,
)


def _mods(b, d):
    torch.manual_seed(1)
    return (torch.randn(b, d) * 0.1, torch.randn(b, d) * 0.1)


def test_gelulinear_forward_ln_equals_composed():
    torch.manual_seed(0)
    gl = GELULinear(16, 32)
    x = torch.randn(2, 5, 16)
    sc, sh = _mods(2, 16)
    assert torch.equal(
        gl.forward_ln(x, sc, sh), gl(ops.layer_norm_mod(x, sc, sh))
    )


def test_fusedmlp_forward_ln_equals_composed():
    torch.manual_seed(2)
    m = FusedMLP(16, 48, 16)
    x = torch.randn(3, 4, 16)
    sc, sh = _mods(3, 16)
    assert torch.equal(
        m.forward_ln(x, sc, sh), m(ops.layer_norm_mod(x, sc, sh))
    )


def test_ln_mod_into_plain_linear_equals_composed():
    torch.manual_seed(3)
    lin = torch.nn.Linear(16, 8)
    x = torch.randn(2, 6, 16)
    sc, sh = _mods(2, 16)
    assert torch.equal(
        ln_mod_into(lin, x, sc, sh), lin(ops.layer_norm_mod(x, sc, sh))
    )


def test_mlpembedder_forward_timestep_equals_composed():
    torch.manual_seed(4)
    emb = MLPEmbedder(32, 64)
    t = torch.rand(4)
    ref = emb(ops.timestep_embedding(t, 32).to(torch.float32))
    assert torch.equal(emb.forward_timestep(t), ref)
