"""Coverage for utility modules: profiling, shim, cleanup edges."""
import time

import pytest
import torch

from comfyui_parallelanything_amd.utils import comfy_shim
from comfyui_parallelanything_amd.utils.profiling import StepTimer, trace_range


def test_step_timer_summary():
    t = StepTimer(devices=["cpu"])
    for _ in range(3):
        t.start()
        time.sleep(0.01)
        t.stop(images=4)
    s = t.summary()
    assert s["steps"] == 3
    assert s["sec_per_it"] >= 0.01
    assert s["images_per_s"] > 0
    assert s["min_step_s"] <= s["max_step_s"]


def test_step_timer_dump(tmp_path):
    t = StepTimer()
    t.start()
    t.stop(images=1)
    p = tmp_path / "m.json"
    t.dump(str(p))
    import json

    assert json.loads(p.read_text())["steps"] == 1


def test_trace_range_noop_on_cpu():
    with trace_range("pa::test"):
        pass  # must not raise without a GPU


def test_shim_headless_functions():
    comfy_shim.soft_empty_cache()
    comfy_shim.unload_all_models()
    dev = comfy_shim.get_torch_device()
    assert isinstance(dev, torch.device)


def test_shim_lora_detection_variants():
    class W:
        patches = {}
        object_patches = {}

    assert not comfy_shim.detect_lora_patches(W())
    w = W()
    w.object_patches = {"x": 1}
    assert comfy_shim.detect_lora_patches(w)
    assert not comfy_shim.detect_lora_patches(None)


def test_apply_lora_without_patch_model():
    assert not comfy_shim.apply_lora_patches(object(), "cpu")


def test_aggressive_cleanup_no_gpu():
    from comfyui_parallelanything_amd.parallel.cleanup import aggressive_cleanup

    aggressive_cleanup()  # must not raise


def test_finalizer_fires_on_gc():
    import gc

    from comfyui_parallelanything_amd.models.registry import make_sd15
    from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
    from comfyui_parallelanything_amd.parallel.cleanup import register_finalizer
    from comfyui_parallelanything_amd.parallel.engine import (
        ParallelEngine,
        install_parallel_forward,
    )

    m = make_sd15(tiny=True)
    eng = ParallelEngine(
        DeviceChain.from_list([make_entry("cpu", 100)]), auto_vram_balance=False
    )
    eng.setup(m)
    install_parallel_forward(m, eng)

    class Owner:
        pass

    owner = Owner()
    fin = register_finalizer(owner, m)
    assert m._true_parallel_active
    del owner
    gc.collect()
    assert not fin.alive
    assert not getattr(m, "_true_parallel_active", False)


def test_comm_stats_summary_shape():
    from comfyui_parallelanything_amd.parallel.dist import CommStats

    cs = CommStats()
    cs.sent(torch.zeros(4, dtype=torch.float32))
    cs.recvd(torch.zeros(2, dtype=torch.float32))
    s = cs.summary()
    assert s == {"sent_bytes": 16, "recv_bytes": 8, "p2p_ops": 2}
