import torch

from comfyui_parallelanything_amd.parallel.dist import DistInfo
from comfyui_parallelanything_amd.parallel.elastic import (
    abort_to_local,
    step_with_fallback,
)


def _info(world=2):
    return DistInfo(rank=0, world_size=world, local_rank=0,
                    device=torch.device("cpu"), backend="gloo")


def test_single_rank_runs_local():
    out, info = step_with_fallback(
        dist_step=lambda i: (_ for _ in ()).throw(AssertionError("no dist")),
        local_step=lambda: "local",
        info=_info(world=1),
    )
    assert out == "local" and info.world_size == 1


def test_dist_step_success_keeps_group():
    out, info = step_with_fallback(
        dist_step=lambda i: "dist",
        local_step=lambda: "local",
        info=_info(world=2),
    )
    assert out == "dist" and info.world_size == 2


def test_failure_degrades_to_local():
    calls = []

    def failing(i):
        calls.append("dist")
        raise RuntimeError("NCCL communicator was aborted")

    out, info = step_with_fallback(failing, lambda: "local", _info(world=2))
    assert out == "local"
    assert info.world_size == 1  # degraded permanently
    # subsequent steps skip the dist path entirely
    out2, info = step_with_fallback(failing, lambda: "local2", info)
    assert out2 == "local2" and calls == ["dist"]


def test_abort_to_local_without_group():
    info = abort_to_local(_info(world=4))
    assert info.world_size == 1 and info.rank == 0


def test_debug_env(monkeypatch):
    from comfyui_parallelanything_amd.utils.debug import apply_debug_env

    monkeypatch.setenv("PA_DEBUG_SERIALIZE", "1")
    applied = apply_debug_env()
    assert applied.get("serialize")
    import os

    assert os.environ["AMD_SERIALIZE_KERNEL"] == "3"
