"""Process-group transport tests over gloo (world_size 2, CPU).

Exercises the exact call pattern the RCCL path uses on MI355X: p2p
scatterv/gatherv and the flat bucketed weight broadcast."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from comfyui_parallelanything_amd.parallel.dist import DistInfo


def _init(rank, world, tmpdir):
    store_path = os.path.join(tmpdir, "store")
    dist.init_process_group(
        "gloo", init_method=f"file://{store_path}", rank=rank, world_size=world
    )
    return DistInfo(rank=rank, world_size=world, local_rank=rank,
                    device=torch.device("cpu"), backend="gloo")


def _scatter_gather_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.parallel.dist import gatherv, scatterv

    info = _init(rank, world, tmpdir)
    sizes = [3, 2]
    full = torch.arange(5 * 4, dtype=torch.float32).view(5, 4)
    template = torch.empty(0, 4)
    chunk = scatterv(full if rank == 0 else None, sizes, info, src=0,
                     template=template)
    assert chunk.shape[0] == sizes[rank]
    if rank == 0:
        assert torch.equal(chunk, full[:3])
    else:
        assert torch.equal(chunk, full[3:])

    out = gatherv(chunk * 2, sizes, info, dst=0)
    if rank == 0:
        assert torch.equal(out, full * 2)
    else:
        assert out is None
    dist.destroy_process_group()


def _broadcast_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.parallel.replicate import broadcast_module

    info = _init(rank, world, tmpdir)
    torch.manual_seed(rank)  # ranks start with DIFFERENT weights
    m = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.LayerNorm(16), torch.nn.Linear(16, 8)
    )
    m.to(torch.float32)
    broadcast_module(m, src_rank=0, bucket_bytes=256)  # force multiple buckets
    # all ranks must now hold rank 0's weights
    torch.manual_seed(0)
    ref = torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.LayerNorm(16), torch.nn.Linear(16, 8)
    )
    for p, q in zip(m.parameters(), ref.parameters()):
        assert torch.equal(p.data, q.data), "broadcast diverged from src weights"
    dist.destroy_process_group()


def _all_max_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.parallel.dist import all_max

    info = _init(rank, world, tmpdir)
    v = all_max(float(rank + 1), info)
    assert v == float(world)
    dist.destroy_process_group()


@pytest.mark.parametrize(
    "worker", [_scatter_gather_worker, _broadcast_worker, _all_max_worker]
)
def test_world2_gloo(worker, tmp_path):
    mp.spawn(worker, args=(2, str(tmp_path)), nprocs=2, join=True)


def _comm_stats_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.parallel.dist import (
        COMM_STATS, gatherv, scatterv,
    )

    info = _init(rank, world, tmpdir)
    full = torch.zeros(4, 8)
    chunk = scatterv(full if rank == 0 else None, [2, 2], info, template=full)
    gatherv(chunk, [2, 2], info, dst=0)
    s = COMM_STATS.summary()
    # each rank either sends or receives 2*8 floats per direction
    assert s["sent_bytes"] + s["recv_bytes"] == 2 * 2 * 8 * 4
    assert s["p2p_ops"] == 2
    dist.destroy_process_group()


def test_comm_stats(tmp_path):
    mp.spawn(_comm_stats_worker, args=(2, str(tmp_path)), nprocs=2, join=True)


def _world4_worker(rank, world, tmpdir):
    from comfyui_parallelanything_amd.parallel.dist import gatherv, scatterv

    info = _init(rank, world, tmpdir)
    sizes = [3, 2, 2, 1]  # weighted split, driver-style N=4
    full = torch.arange(8 * 4, dtype=torch.float32).view(8, 4)
    chunk = scatterv(full if rank == 0 else None, sizes, info, src=0,
                     template=full)
    assert chunk.shape[0] == sizes[rank]
    out = gatherv(chunk + rank, sizes, info, dst=0)
    if rank == 0:
        offs = [0, 3, 5, 7, 8]
        for r in range(4):
            assert torch.equal(out[offs[r]:offs[r + 1]],
                               full[offs[r]:offs[r + 1]] + r)
    dist.destroy_process_group()


def test_world4_gloo(tmp_path):
    mp.spawn(_world4_worker, args=(4, str(tmp_path)), nprocs=4, join=True)
