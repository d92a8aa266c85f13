"""ComfyUI node surface: ParallelDevice / ParallelDeviceList / ParallelAnything.

Drop-in compatible with the reference's registry and link schema
(any_device_parallel.py:1473-1483): same node names, same DEVICE_CHAIN
list[{"device","percentage","weight"}] payload, same MODEL-in/MODEL-out
orchestrator contract, same option set (workload_split, auto_vram_balance,
purge_cache, purge_models; defaults :893-909). The runtime behind the nodes
is the MI355X engine (parallel.engine) instead of a thread pool.
"""
from __future__ import annotations

import logging

import torch

from .parallel.chain import DeviceChain, available_devices, chain_append, chain_from_slots
from .parallel.cleanup import cleanup_parallel_model, register_finalizer
from .parallel.engine import ParallelEngine, install_parallel_forward
from .parallel.pipeline import configure_pipeline
from .utils import comfy_shim

log = logging.getLogger("parallelanything")


class ParallelDevice:
    """Chainable device-config node (reference :768-832)."""

    @classmethod
    def get_available_devices(cls):
        return available_devices()

    @classmethod
    def INPUT_TYPES(cls):
        avail = cls.get_available_devices()
        default = "cuda:0" if "cuda:0" in avail else avail[0]
        return {
            "required": {
                "device_id": (avail, {
                    "default": default,
                    "tooltip": "Select available compute device (CPU/HIP GPU)",
                }),
                "percentage": ("FLOAT", {
                    "default": 50.0, "min": 1.0, "max": 100.0, "step": 1.0,
                    "tooltip": "Percentage of batch (or layers for batch=1) for this device",
                }),
            },
            "optional": {
                "previous_devices": ("DEVICE_CHAIN", {
                    "tooltip": "Connect from another ParallelDevice node to chain multiple GPUs",
                }),
            },
        }

    RETURN_TYPES = ("DEVICE_CHAIN",)
    RETURN_NAMES = ("device_chain",)
    FUNCTION = "add_device"
    CATEGORY = "utils/hardware"
    DESCRIPTION = "Add a GPU/CPU device to the parallel processing chain"

    def add_device(self, device_id, percentage, previous_devices=None):
        return (chain_append(previous_devices, device_id, percentage),)


class ParallelDeviceList:
    """1-4 device single-node variant (reference :834-882)."""

    @classmethod
    def get_available_devices(cls):
        return available_devices()

    @classmethod
    def INPUT_TYPES(cls):
        devs = cls.get_available_devices()
        def_dev = "cuda:0" if "cuda:0" in devs else devs[0]
        return {
            "required": {
                "device_1": (devs, {"default": def_dev}),
                "pct_1": ("FLOAT", {"default": 50.0, "min": 1.0, "max": 100.0, "step": 1.0}),
                "device_2": (devs, {"default": devs[1] if len(devs) > 1 else def_dev}),
                "pct_2": ("FLOAT", {"default": 50.0, "min": 0.0, "max": 100.0, "step": 1.0}),
            },
            "optional": {
                "device_3": (devs, {"default": devs[2] if len(devs) > 2 else "cpu"}),
                "pct_3": ("FLOAT", {"default": 0.0, "min": 0.0, "max": 100.0, "step": 1.0}),
                "device_4": (devs, {"default": devs[3] if len(devs) > 3 else "cpu"}),
                "pct_4": ("FLOAT", {"default": 0.0, "min": 0.0, "max": 100.0, "step": 1.0}),
            },
        }

    RETURN_TYPES = ("DEVICE_CHAIN",)
    RETURN_NAMES = ("device_chain",)
    FUNCTION = "create_list"
    CATEGORY = "utils/hardware"

    def create_list(self, device_1, pct_1, device_2, pct_2,
                    device_3="cpu", pct_3=0, device_4="cpu", pct_4=0):
        return (chain_from_slots([
            (device_1, pct_1), (device_2, pct_2), (device_3, pct_3), (device_4, pct_4),
        ]),)


class ParallelAnything:
    """Orchestrator node: replicate + install the parallel forward
    (reference :884-1471)."""

    @classmethod
    def INPUT_TYPES(cls):
        return {
            "required": {
                "model": ("MODEL",),
                "device_chain": ("DEVICE_CHAIN", {"tooltip": "Connect from ParallelDevice nodes"}),
            },
            "optional": {
                "workload_split": ("BOOLEAN", {
                    "default": True, "tooltip": "Enable multi-device processing"}),
                "auto_vram_balance": ("BOOLEAN", {
                    "default": True,
                    "tooltip": "Automatically adjust batch split based on available VRAM"}),
                "purge_cache": ("BOOLEAN", {
                    "default": True,
                    "tooltip": "Purge HIP cache when cleaning up parallel resources"}),
                "purge_models": ("BOOLEAN", {
                    "default": False,
                    "tooltip": "Unload all models when cleaning up (aggressive)"}),
                "pipeline_microbatches": ("INT", {
                    "default": 1, "min": 1, "max": 16,
                    "tooltip": "Micro-batches for block-sharded pipeline mode "
                               "(>1 overlaps stages for small batches; "
                               "extension over the reference)"}),
                "use_hip_graphs": ("BOOLEAN", {
                    "default": False,
                    "tooltip": "Capture repeated same-shape forwards into "
                               "hipGraphs (one launch per denoise step; "
                               "extension over the reference)"}),
            },
        }

    RETURN_TYPES = ("MODEL",)
    RETURN_NAMES = ("model",)
    FUNCTION = "setup_parallel"
    CATEGORY = "utils/hardware"

    def setup_parallel(self, model, device_chain, workload_split=True,
                       auto_vram_balance=True, purge_cache=True,
                       purge_models=False, pipeline_microbatches=1,
                       use_hip_graphs=False):
        if model is None or not device_chain:
            return (model,)

        target_model, wrapper = comfy_shim.unwrap_model(model)
        chain = DeviceChain.from_list(device_chain)
        lead = torch.device(chain.lead)

        # Stranded-on-CPU restore (reference :932-961): a model the manager
        # offloaded to host gets moved to the lead device before replication.
        try:
            p = next(target_model.parameters())
            if p.device.type == "cpu" and lead.type == "cuda":
                target_model.to(lead)
        except StopIteration:
            pass

        # LoRA: bake patches, then force-copy even the lead replica so the
        # patched weights are preserved per replica (reference :971-1004,
        # :1073-1081).
        has_lora = comfy_shim.detect_lora_patches(wrapper)
        if has_lora:
            comfy_shim.apply_lora_patches(wrapper, lead)

        # Re-run: release any prior parallel state first (reference :1006-1013).
        cleanup_parallel_model(target_model)
        if purge_models:
            comfy_shim.unload_all_models()
        comfy_shim.soft_empty_cache()

        engine = ParallelEngine(
            chain,
            workload_split=workload_split,
            auto_vram_balance=auto_vram_balance,
            use_hip_graphs=use_hip_graphs,
        )
        try:
            engine.setup(target_model, force_copy_lead=has_lora)
        except Exception:  # noqa: BLE001
            log.exception("replication failed; returning model unchanged")
            engine.release()
            return (model,)
        configure_pipeline(engine, microbatches=pipeline_microbatches)

        install_parallel_forward(target_model, engine)
        target_model._parallel_purge_cache = purge_cache
        target_model._parallel_purge_models = purge_models
        register_finalizer(model, target_model)

        # Retarget the wrapper's load_device to the lead GPU so ComfyUI's
        # model manager keeps inputs there (reference :1461-1465).
        if wrapper is not None and hasattr(wrapper, "load_device"):
            try:
                wrapper.load_device = lead
            except Exception:  # noqa: BLE001
                pass
        log.info(
            "parallel setup: devices=%s weights=%s",
            engine.chain.devices, tuple(round(w, 4) for w in engine.chain.weights),
        )
        return (model,)


NODE_CLASS_MAPPINGS = {
    "ParallelDevice": ParallelDevice,
    "ParallelDeviceList": ParallelDeviceList,
    "ParallelAnything": ParallelAnything,
}

NODE_DISPLAY_NAME_MAPPINGS = {
    "ParallelDevice": "Parallel Device Config",
    "ParallelDeviceList": "Parallel Device List (1-4x)",
    "ParallelAnything": "Parallel Anything (True Multi-GPU)",
}
