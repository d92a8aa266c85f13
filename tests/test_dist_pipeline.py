"""Process-group pipeline mode over p2p (gloo world=2, CPU)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from comfyui_parallelanything_amd.parallel.dist import DistInfo


def _init(rank, world, tmpdir):
    dist.init_process_group(
        "gloo", init_method=f"file://{os.path.join(tmpdir, 'store')}",
        rank=rank, world_size=world,
    )
    return DistInfo(rank=rank, world_size=world, local_rank=rank,
                    device=torch.device("cpu"), backend="gloo")


def _flux_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux
    from comfyui_parallelanything_amd.parallel.dist_pipeline import (
        install_dist_pipeline,
        uninstall_dist_pipeline,
    )

    info = _init(rank, world, tmpdir)
    model = make_flux(tiny=True, dtype=torch.float32)  # same seed everywhere
    x, t, c, kw = flux_inputs(1, tiny=True, dtype=torch.float32)
    ref = model(x, t, context=c, **kw)  # pre-install reference

    n = install_dist_pipeline(model, info)
    assert n == 4  # 2 double + 2 single blocks wrapped
    out = model(x, t, context=c, **kw)
    if rank == 0:
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    assert uninstall_dist_pipeline(model) == n
    out2 = model(x, t, context=c, **kw)  # restored local execution
    torch.testing.assert_close(out2, ref)
    dist.destroy_process_group()


def _zimage_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.models.registry import make_zimage, zimage_inputs
    from comfyui_parallelanything_amd.parallel.dist_pipeline import (
        install_dist_pipeline,
    )

    info = _init(rank, world, tmpdir)
    model = make_zimage(tiny=True, dtype=torch.float32)
    x, t, c, kw = zimage_inputs(1, tiny=True, dtype=torch.float32)
    ref = model(x, t, context=c, **kw)
    install_dist_pipeline(model, info, weights=[0.7, 0.3])
    out = model(x, t, context=c, **kw)
    if rank == 0:
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.parametrize("worker", [_flux_worker, _zimage_worker])
def test_dist_pipeline_world2(worker, tmp_path):
    mp.spawn(worker, args=(2, str(tmp_path)), nprocs=2, join=True)


def _flux_worker3(rank, world, tmpdir):
    from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux
    from comfyui_parallelanything_amd.parallel.dist_pipeline import (
        install_dist_pipeline,
    )

    info = _init(rank, world, tmpdir)
    model = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(1, tiny=True, dtype=torch.float32)
    ref = model(x, t, context=c, **kw)
    install_dist_pipeline(model, info, weights=[0.5, 0.3, 0.2])
    out = model(x, t, context=c, **kw)
    if rank == 0:
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    dist.destroy_process_group()


def test_dist_pipeline_world3(tmp_path):
    mp.spawn(_flux_worker3, args=(3, str(tmp_path)), nprocs=3, join=True)


def _wan_i2v_worker4(rank, world, tmpdir):
    from comfyui_parallelanything_amd.models.registry import (
        make_wan_i2v, wan_i2v_inputs,
    )
    from comfyui_parallelanything_amd.parallel.dist_pipeline import (
        install_dist_pipeline, uninstall_dist_pipeline,
    )

    info = _init(rank, world, tmpdir)
    torch.manual_seed(0)
    model = make_wan_i2v(tiny=True, dtype=torch.float32)
    x, t, c, kw = wan_i2v_inputs(1, tiny=True, dtype=torch.float32)
    ref = model(x, t, context=c, **kw)
    n = install_dist_pipeline(model, info)
    assert n > 0  # transformer_blocks sharded
    out = model(x, t, context=c, **kw)
    if rank == 0:
        torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    assert uninstall_dist_pipeline(model) == n
    out2 = model(x, t, context=c, **kw)
    torch.testing.assert_close(out2, ref, rtol=0, atol=0)
    dist.destroy_process_group()


def test_dist_pipeline_world4_wan_i2v(tmp_path):
    """4-rank layer sharding of the I2V video DiT (image_cond kwarg flows
    locally on every rank; hidden state hands off by p2p)."""
    mp.spawn(_wan_i2v_worker4, args=(4, str(tmp_path)), nprocs=4, join=True)
