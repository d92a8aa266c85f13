"""End-to-end bench.py step-loop tests at world 4 and 8 over gloo.

VERDICT round-1 item 1: before the first real 8-GPU run, the EXACT loop the
driver benches (scatterv -> forward -> gatherv -> all_max, weighted sizes,
template paths, zero-size ranks) must pass multi-process on CPU. These
tests call ``bench.main`` itself — not a reimplementation — so every
collective call site, the routing of --weights, and the result contract are
exercised as a unit. The gloo backend runs the identical torch.distributed
call pattern RCCL runs on MI355X.
"""
import json
import os
import socket
import sys

import pytest
import torch
import torch.multiprocessing as mp

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

BASE_ARGS = [
    "--model", "sd15", "--tiny", "--steps", "2", "--warmup", "1",
    "--px", "64",
]


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _bench_worker(rank, world, port, json_path, extra_argv):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.manual_seed(0)  # identical random-init weights on every rank
    torch.set_num_threads(1)
    sys.path.insert(0, REPO_ROOT)
    import bench

    res = bench.main(BASE_ARGS + list(extra_argv))
    if rank == 0:
        with open(json_path, "w") as f:
            json.dump(res, f)
    import torch.distributed as dist

    if dist.is_initialized():
        dist.destroy_process_group()


def _run_world(world, tmp_path, extra_argv):
    port = _free_port()
    out = str(tmp_path / f"bench_w{world}.json")
    mp.spawn(
        _bench_worker,
        args=(world, port, out, extra_argv),
        nprocs=world,
        join=True,
    )
    with open(out) as f:
        return json.load(f)


def _single_proc_reference(extra_argv):
    """bench.main in-process, world 1, same seed -> checksum ground truth."""
    for var in ("RANK", "WORLD_SIZE", "LOCAL_RANK"):
        os.environ.pop(var, None)
    torch.manual_seed(0)
    sys.path.insert(0, REPO_ROOT)
    import bench

    return bench.main(BASE_ARGS + list(extra_argv))


@pytest.mark.parametrize("world", [4, 8])
def test_bench_step_loop_matches_single_process(world, tmp_path):
    """N-rank scatter/forward/gather must produce the same final latent as
    the single-process run (CPU math is bitwise chunk-stable)."""
    extra = ["--batch", "8"]
    ref = _single_proc_reference(extra)
    res = _run_world(world, tmp_path, extra)
    assert res["n_gpus"] == world
    assert res["ms_per_step"] > 0
    assert res["config"]["parallelism"] == f"dp{world}"
    assert res["x_checksum"] == pytest.approx(ref["x_checksum"], rel=1e-5)


def test_bench_weighted_split_world4(tmp_path):
    """The Z-Image-style weighted split path (--weights) at world 4."""
    extra = ["--batch", "8", "--weights", "40,30,20,10"]
    ref = _single_proc_reference(["--batch", "8"])
    res = _run_world(4, tmp_path, extra)
    assert "weighted" in res["config"]["parallelism"]
    assert res["x_checksum"] == pytest.approx(ref["x_checksum"], rel=1e-5)


def test_bench_zero_size_ranks_world4(tmp_path):
    """batch < world leaves some ranks with zero-size chunks — the loop
    (incl. the source rank holding a 0-row slice) must not deadlock."""
    extra = ["--batch", "2"]
    ref = _single_proc_reference(extra)
    res = _run_world(4, tmp_path, extra)
    assert res["x_checksum"] == pytest.approx(ref["x_checksum"], rel=1e-5)


def test_bench_wan_i2v_world2(tmp_path):
    """wan_i2v through the process-group loop at world 2: 5-D latents and
    the batch-shaped image_cond kwarg ride scatterv/gatherv."""
    extra = ["--batch", "4"]
    base = [a for a in BASE_ARGS]
    base[base.index("sd15")] = "wan_i2v"
    # remove --px (wan path takes no px inputs)
    i = base.index("--px")
    del base[i:i + 2]
    import tests.test_bench_loop as me

    old = me.BASE_ARGS
    me.BASE_ARGS = base
    try:
        ref = _single_proc_reference(extra)
        res = _run_world(2, tmp_path, extra)
        assert res["x_checksum"] == pytest.approx(ref["x_checksum"], rel=1e-5)
    finally:
        me.BASE_ARGS = old
