import pytest
import torch

from comfyui_parallelanything_amd.models.registry import flux_inputs, make_flux
from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
from comfyui_parallelanything_amd.parallel.cleanup import unwrap_pipeline_blocks
from comfyui_parallelanything_amd.parallel.engine import ParallelEngine
from comfyui_parallelanything_amd.parallel.pipeline import (
    ParallelBlock,
    assign_block_ranges,
    configure_pipeline,
    pipeline_mode_active,
)


def cpu_chain(*pcts):
    return DeviceChain.from_list([make_entry("cpu", p) for p in pcts])


def test_assign_block_ranges_even():
    owners = assign_block_ranges(10, [0.5, 0.5])
    assert owners == [0] * 5 + [1] * 5


def test_assign_block_ranges_weighted_remainder_last():
    # round(0.6*10)=6 to dev0; remainder 4 to dev1 (reference :1168-1178)
    assert assign_block_ranges(10, [0.6, 0.4]) == [0] * 6 + [1] * 4


def test_assign_block_ranges_covers_all():
    for n in (1, 3, 7, 19, 38):
        for ws in ([0.5, 0.5], [0.9, 0.1], [0.34, 0.33, 0.33]):
            owners = assign_block_ranges(n, ws)
            assert len(owners) == n
            assert owners == sorted(owners)  # contiguous ranges


@pytest.fixture(scope="module")
def engine():
    m = make_flux(tiny=True, dtype=torch.float32)
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(m, force_copy_lead=True)  # force distinct replicas
    configure_pipeline(eng)
    return m, eng


def test_blocks_wrapped(engine):
    m, eng = engine
    lead = eng.lead_replica
    assert all(isinstance(b, ParallelBlock) for b in lead.double_blocks)
    assert all(isinstance(b, ParallelBlock) for b in lead.single_blocks)


def test_batch1_routes_to_pipeline_and_matches(engine):
    m, eng = engine
    x, t, c, kw = flux_inputs(1, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)
    assert not pipeline_mode_active()  # flag restored


def test_dp_still_works_with_wrappers(engine):
    m, eng = engine
    x, t, c, kw = flux_inputs(4, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_unwrap_restores_blocks():
    m = make_flux(tiny=True, dtype=torch.float32)
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(m, force_copy_lead=True)
    configure_pipeline(eng)
    lead = eng.lead_replica
    n = unwrap_pipeline_blocks(lead)
    assert n > 0
    assert not any(isinstance(b, ParallelBlock) for b in lead.double_blocks)


def test_microbatched_pipeline_matches_reference():
    """GPipe-style micro-batching: batch 3 through a 3-device pipeline
    in 3 concurrent micro-batches equals the single-device forward."""
    m = make_flux(tiny=True, dtype=torch.float32)
    eng = ParallelEngine(cpu_chain(34, 33, 33), auto_vram_balance=False)
    eng.setup(m, force_copy_lead=True)
    configure_pipeline(eng, microbatches=4)
    x, t, c, kw = flux_inputs(3, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    out = eng.forward(x, t, context=c, **kw)  # 3 < 3 devices? no: 3 == 3 → DP
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
    # batch 2 < 3 devices → pipeline with 2 micro-batches (DP impossible)
    x, t, c, kw = flux_inputs(2, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    out = eng.pipeline.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)
    assert not pipeline_mode_active()


def test_microbatch_routing_small_batch():
    """1 < batch < n_devices routes to the pipeline when micro-batching is
    on, and to the lead when it is off (reference behavior)."""
    m = make_flux(tiny=True, dtype=torch.float32)
    eng = ParallelEngine(cpu_chain(34, 33, 33), auto_vram_balance=False)
    eng.setup(m, force_copy_lead=True)
    configure_pipeline(eng, microbatches=2)

    calls = {"pipe": 0}
    orig = eng.pipeline.forward

    def spy(*a, **kw):
        calls["pipe"] += 1
        return orig(*a, **kw)

    eng.pipeline.forward = spy
    x, t, c, kw = flux_inputs(2, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    out = eng.forward(x, t, context=c, **kw)
    assert calls["pipe"] == 1
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)

    eng.pipeline.microbatches = 1  # reference routing: lead-only
    eng.forward(x, t, context=c, **kw)
    assert calls["pipe"] == 1


def test_microbatch_error_propagates_and_clears_flag():
    class Bad(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = torch.nn.ModuleList([torch.nn.Identity()])

        def forward(self, x, t, context=None):
            raise ValueError("mb failure")

    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(Bad(), force_copy_lead=True)
    configure_pipeline(eng, microbatches=2)
    if eng.pipeline is None:
        pytest.skip("no block list wired")
    with pytest.raises(ValueError):
        eng.pipeline.forward(torch.zeros(4, 3), torch.zeros(4))
    assert not pipeline_mode_active()


def test_flag_cleared_on_error():
    class Bad(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.layers = torch.nn.ModuleList([torch.nn.Identity()])

        def forward(self, x, t, context=None):
            raise RuntimeError("inner failure")

    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(Bad(), force_copy_lead=True)
    configure_pipeline(eng)
    if eng.pipeline is None:
        pytest.skip("no block list wired")
    with pytest.raises(RuntimeError):
        eng.forward(torch.zeros(1, 3), torch.zeros(1))
    assert not pipeline_mode_active()
