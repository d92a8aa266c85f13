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
