"""Property-based invariants for the batch-split math (hypothesis).

The reference's split rules (any_device_parallel.py:1321-1337) have sharp
edge cases (min-1 floors over-committing, zero weights, more devices than
samples); these properties pin the repaired semantics for ALL inputs, not
just the examples in test_split.py.
"""
import pytest

hyp = pytest.importorskip("hypothesis")

import torch  # noqa: E402
from hypothesis import given, settings, strategies as st  # noqa: E402

from comfyui_parallelanything_amd.parallel.split import (  # noqa: E402
    active_split,
    compute_split_sizes,
    concatenate_results,
    split_batch,
    split_kwargs,
)

weights_st = st.lists(
    st.floats(min_value=0.001, max_value=1.0), min_size=1, max_size=9
).map(lambda ws: [w / sum(ws) for w in ws])


@settings(max_examples=200, deadline=None)
@given(batch=st.integers(min_value=0, max_value=512), weights=weights_st)
def test_sizes_sum_and_nonnegative(batch, weights):
    sizes = compute_split_sizes(batch, weights)
    assert sum(sizes) == batch
    assert all(s >= 0 for s in sizes)
    assert len(sizes) == len(weights)


@settings(max_examples=100, deadline=None)
@given(batch=st.integers(min_value=1, max_value=256), weights=weights_st)
def test_active_split_covers_batch(batch, weights):
    sizes = compute_split_sizes(batch, weights)
    devs = [f"cpu{i}" for i in range(len(weights))]
    a_devs, a_w, a_sizes = active_split(devs, weights, sizes)
    assert sum(a_sizes) == batch
    assert all(s > 0 for s in a_sizes)
    assert len(a_devs) == len(a_sizes) <= len(devs)


@settings(max_examples=60, deadline=None)
@given(batch=st.integers(min_value=1, max_value=64), weights=weights_st,
       feat=st.integers(min_value=1, max_value=8))
def test_split_concat_roundtrip(batch, weights, feat):
    x = torch.arange(batch * feat, dtype=torch.float32).reshape(batch, feat)
    sizes = compute_split_sizes(batch, weights)
    chunks = split_batch(x, sizes)
    assert torch.equal(concatenate_results(chunks, dim=0), x)


@settings(max_examples=60, deadline=None)
@given(batch=st.integers(min_value=1, max_value=32), weights=weights_st)
def test_kwargs_split_preserves_batch_rows(batch, weights):
    sizes = compute_split_sizes(batch, weights)
    kw = {
        "batched": torch.randn(batch, 3),
        "broadcast": torch.randn(batch + 1, 3),  # wrong B: broadcast whole
        "scalar": 7,
    }
    outs = split_kwargs(kw, sizes, batch)
    assert len(outs) == len(sizes)
    recon = torch.cat([o["batched"] for o in outs if o["batched"].numel()], 0)
    assert torch.equal(recon, kw["batched"])
    for o in outs:
        assert o["scalar"] == 7
        assert o["broadcast"] is kw["broadcast"]


@settings(max_examples=100, deadline=None)
@given(n=st.integers(min_value=1, max_value=64), weights=weights_st)
def test_block_ranges_partition(n, weights):
    from comfyui_parallelanything_amd.parallel.pipeline import (
        assign_block_ranges,
    )

    owners = assign_block_ranges(n, weights)
    assert len(owners) == n
    assert owners == sorted(owners)          # contiguous stage ranges
    assert all(0 <= o < len(weights) for o in owners)
