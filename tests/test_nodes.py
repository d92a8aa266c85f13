"""ComfyUI node-surface tests against a stub MODEL wrapper (no ComfyUI)."""
import gc

import pytest
import torch
from torch import nn

from comfyui_parallelanything_amd import (
    NODE_CLASS_MAPPINGS,
    NODE_DISPLAY_NAME_MAPPINGS,
    ParallelAnything,
    ParallelDevice,
    ParallelDeviceList,
)
from comfyui_parallelanything_amd.models.registry import make_sd15, sd15_inputs
from comfyui_parallelanything_amd.parallel.cleanup import cleanup_parallel_model


class FakeInnerModel:
    def __init__(self, diffusion_model):
        self.diffusion_model = diffusion_model


class FakeModelWrapper:
    """Mimics ComfyUI's ModelPatcher surface the node touches."""

    def __init__(self, diffusion_model):
        self.model = FakeInnerModel(diffusion_model)
        self.load_device = torch.device("cpu")
        self.patches = {}


def test_registry_names_match_reference():
    assert set(NODE_CLASS_MAPPINGS) == {
        "ParallelDevice", "ParallelDeviceList", "ParallelAnything",
    }
    assert NODE_DISPLAY_NAME_MAPPINGS["ParallelAnything"] == (
        "Parallel Anything (True Multi-GPU)"
    )


def test_device_node_schema():
    it = ParallelDevice.INPUT_TYPES()
    assert "device_id" in it["required"]
    assert "percentage" in it["required"]
    assert "previous_devices" in it["optional"]
    assert ParallelDevice.RETURN_TYPES == ("DEVICE_CHAIN",)
    assert ParallelDevice.FUNCTION == "add_device"


def test_device_list_schema():
    it = ParallelDeviceList.INPUT_TYPES()
    assert {"device_1", "pct_1", "device_2", "pct_2"} <= set(it["required"])
    assert {"device_3", "pct_3", "device_4", "pct_4"} <= set(it["optional"])


def test_anything_schema_defaults():
    it = ParallelAnything.INPUT_TYPES()
    opts = it["optional"]
    assert opts["workload_split"][1]["default"] is True
    assert opts["auto_vram_balance"][1]["default"] is True
    assert opts["purge_cache"][1]["default"] is True
    assert opts["purge_models"][1]["default"] is False
    # extensions default OFF: stock node behavior == reference behavior
    assert opts["pipeline_microbatches"][1]["default"] == 1
    assert opts["use_hip_graphs"][1]["default"] is False


def test_chain_building_via_nodes():
    n = ParallelDevice()
    (c1,) = n.add_device("cpu", 40)
    (c2,) = n.add_device("cpu", 60, previous_devices=c1)
    assert [e["weight"] for e in c2] == [0.4, 0.6]

    nl = ParallelDeviceList()
    (cl,) = nl.create_list("cpu", 50, "cpu", 50, "cpu", 0, "cpu", 0)
    assert len(cl) == 2


def _chain(*pcts):
    node = ParallelDevice()
    chain = None
    for p in pcts:
        (chain,) = node.add_device("cpu", p, previous_devices=chain)
    return chain


def test_setup_parallel_end_to_end():
    dm = make_sd15(tiny=True)
    wrapper = FakeModelWrapper(dm)
    node = ParallelAnything()
    (out,) = node.setup_parallel(wrapper, _chain(50, 50), auto_vram_balance=False)
    assert out is wrapper
    assert dm._true_parallel_active
    assert dm._parallel_devices == ("cpu", "cpu")

    x, t, c, kw = sd15_inputs(2, tiny=True)
    ref_engineless = make_sd15(tiny=True)(x, t, context=c, **kw)
    y = dm(x, t, context=c, **kw)  # monkeypatched forward
    torch.testing.assert_close(y, ref_engineless, rtol=1e-4, atol=1e-5)

    cleanup_parallel_model(dm)
    assert not getattr(dm, "_true_parallel_active", False)


def test_setup_none_model_passthrough():
    node = ParallelAnything()
    assert node.setup_parallel(None, _chain(100)) == (None,)
    w = FakeModelWrapper(make_sd15(tiny=True))
    assert node.setup_parallel(w, []) == (w,)


def test_rerun_setup_is_clean():
    dm = make_sd15(tiny=True)
    wrapper = FakeModelWrapper(dm)
    node = ParallelAnything()
    node.setup_parallel(wrapper, _chain(50, 50), auto_vram_balance=False)
    first_engine = dm._parallel_engine
    node.setup_parallel(wrapper, _chain(100), auto_vram_balance=False)
    assert dm._parallel_engine is not first_engine
    assert dm._parallel_devices == ("cpu",)
    cleanup_parallel_model(dm)


def test_unwrap_precedence():
    from comfyui_parallelanything_amd.utils.comfy_shim import unwrap_model

    dm = nn.Identity()
    w = FakeModelWrapper(dm)
    assert unwrap_model(w)[0] is dm

    class DirectHolder:
        def __init__(self, m):
            self.diffusion_model = m

    assert unwrap_model(DirectHolder(dm))[0] is dm
    assert unwrap_model(dm)[0] is dm  # raw module passthrough


def test_lora_detection_forces_lead_copy():
    dm = make_sd15(tiny=True)
    wrapper = FakeModelWrapper(dm)
    wrapper.patches = {"some.key": [("lora", None)]}
    patched = {}

    def patch_model(device_to=None):
        patched["dev"] = device_to

    wrapper.patch_model = patch_model
    node = ParallelAnything()
    node.setup_parallel(wrapper, _chain(50, 50), auto_vram_balance=False)
    assert "dev" in patched
    # lead replica must be a COPY (not the source model) when LoRA is live
    eng = dm._parallel_engine
    assert eng.replicas["cpu"] is not dm
    cleanup_parallel_model(dm)


def test_cli_tiny_runs(capsys):
    from comfyui_parallelanything_amd.cli import main

    main(["--model", "sd15", "--devices", "cpu,cpu", "--batch", "2",
          "--steps", "2", "--tiny", "--no-balance"])
    out = capsys.readouterr().out
    assert "images_per_s" in out


class ForeignDiT(nn.Module):
    """A hostile third-party diffusion model, reference-style: lazily
    caches device-bound tensors under the names clear_flux_caches scrubs
    (any_device_parallel.py:167-174) and carries a non-picklable attr —
    the model class the reference's clone ladder exists for."""

    def __init__(self):
        super().__init__()
        import threading

        self.inp = nn.Linear(4, 16)
        self.blk = nn.Linear(16, 16)
        self.out = nn.Linear(16, 4)
        self._lock = threading.Lock()   # kills deepcopy
        self.freqs_cis = None           # device-bound lazy cache
        self.img_ids = None

    def forward(self, x, timesteps, context=None, **kwargs):
        B, C, H, W = x.shape
        if self.freqs_cis is None or self.freqs_cis.device != x.device:
            self.freqs_cis = torch.linspace(0, 1, 16, device=x.device)
            self.img_ids = torch.zeros(H * W, device=x.device)
        h = x.permute(0, 2, 3, 1).reshape(B, H * W, C)
        h = self.inp(h) + self.freqs_cis + timesteps.reshape(B, 1, 1)
        h = torch.tanh(self.blk(h)) + 0 * self.img_ids.sum()
        return self.out(h).reshape(B, H, W, C).permute(0, 3, 1, 2)


def test_setup_parallel_foreign_hostile_model():
    """The node path over a FOREIGN model: structural-clone fallback +
    cache scrub + DP split must reproduce the single-model output —
    the reference's core drop-in promise (setup_parallel over arbitrary
    diffusion_model classes)."""
    dm = ForeignDiT()
    x = torch.randn(4, 4, 8, 8)
    t = torch.rand(4)
    ref = dm(x, t).clone()              # populates the source's caches
    assert dm.freqs_cis is not None

    wrapper = FakeModelWrapper(dm)
    node = ParallelAnything()
    (out,) = node.setup_parallel(
        wrapper, _chain(50, 50), auto_vram_balance=False
    )
    assert out is wrapper and dm._true_parallel_active
    y = dm(x, t)                        # DP split through the engine
    torch.testing.assert_close(y, ref, rtol=1e-5, atol=1e-6)
    cleanup_parallel_model(dm)
    assert not getattr(dm, "_true_parallel_active", False)
    # uninstalled model still runs standalone
    torch.testing.assert_close(dm(x, t), ref, rtol=1e-5, atol=1e-6)


class ForeignBlockDiT(nn.Module):
    """Foreign model exposing a reference-recognized block list
    (transformer_blocks, any_device_parallel.py:1156) so batch==1 routes
    through the block-sharded pipeline."""

    def __init__(self):
        super().__init__()
        self.inp = nn.Linear(4, 16)
        self.transformer_blocks = nn.ModuleList(
            nn.Linear(16, 16) for _ in range(4)
        )
        self.out = nn.Linear(16, 4)

    def forward(self, x, timesteps, context=None, **kwargs):
        B, C, H, W = x.shape
        h = self.inp(x.permute(0, 2, 3, 1).reshape(B, H * W, C))
        h = h + timesteps.reshape(B, 1, 1)
        for blk in self.transformer_blocks:
            h = torch.tanh(blk(h))
        return self.out(h).reshape(B, H, W, C).permute(0, 3, 1, 2)


def test_setup_parallel_foreign_pipeline_batch1():
    """batch==1 on a foreign model with transformer_blocks: the node
    wires block-sharded pipeline mode and matches the plain forward."""
    dm = ForeignBlockDiT()
    x = torch.randn(1, 4, 8, 8)
    t = torch.rand(1)
    ref = dm(x, t).clone()
    wrapper = FakeModelWrapper(dm)
    node = ParallelAnything()
    node.setup_parallel(wrapper, _chain(60, 40), auto_vram_balance=False)
    eng = dm._parallel_engine
    assert eng.pipeline is not None, "pipeline mode not configured"
    y = dm(x, t)
    torch.testing.assert_close(y, ref, rtol=1e-5, atol=1e-6)
    cleanup_parallel_model(dm)
    torch.testing.assert_close(dm(x, t), ref, rtol=1e-5, atol=1e-6)
