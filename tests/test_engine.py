"""Engine routing + golden-equality tests on the [cpu,cpu] chain —
BASELINE config 1 (SD1.5-class, plumbing, no GPU needed)."""
import pytest
import torch
from torch import nn

from comfyui_parallelanything_amd.models.registry import (
    flux_inputs,
    make_flux,
    make_sd15,
    sd15_inputs,
)
from comfyui_parallelanything_amd.parallel.chain import DeviceChain, make_entry
from comfyui_parallelanything_amd.parallel.engine import (
    ParallelEngine,
    WorkerError,
    install_parallel_forward,
    uninstall_parallel_forward,
)


def cpu_chain(*pcts):
    return DeviceChain.from_list([make_entry("cpu", p) for p in pcts])


@pytest.fixture(scope="module")
def sd15():
    return make_sd15(tiny=True)


def test_golden_dp_equals_single_sd15(sd15):
    """BASELINE config 1: SD1.5-class 50/50 [cpu,cpu] == single device."""
    x, t, c, kw = sd15_inputs(2, tiny=True)
    ref = sd15(x, t, context=c, **kw)
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    assert out.shape == ref.shape
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_golden_chunkwise_bitwise(sd15):
    """Each parallel chunk is bit-identical to running that chunk alone."""
    x, t, c, kw = sd15_inputs(4, tiny=True)
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    chunk0 = sd15(x[:2], t[:2], context=c[:2], **{k: v[:2] for k, v in kw.items()})
    assert torch.equal(out[:2], chunk0)


def test_golden_weighted_split_flux():
    m = make_flux(tiny=True, dtype=torch.float32)
    x, t, c, kw = flux_inputs(7, tiny=True, dtype=torch.float32)
    ref = m(x, t, context=c, **kw)
    eng = ParallelEngine(cpu_chain(60, 40), auto_vram_balance=False)
    eng.setup(m)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_routing_lead_only_when_batch_below_devices(sd15):
    x, t, c, kw = sd15_inputs(2, tiny=True)
    eng = ParallelEngine(cpu_chain(34, 33, 33), auto_vram_balance=False)
    eng.setup(sd15)
    ref = sd15(x, t, context=c, **kw)
    out = eng.forward(x, t, context=c, **kw)  # batch 2 < 3 devices
    assert torch.equal(out, ref)


def test_routing_split_disabled(sd15):
    x, t, c, kw = sd15_inputs(4, tiny=True)
    eng = ParallelEngine(cpu_chain(50, 50), workload_split=False,
                         auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    assert torch.equal(out, sd15(x, t, context=c, **kw))


def test_worker_error_attribution():
    class Boom(nn.Module):
        def forward(self, x, t, context=None):
            if x.shape[0] == 1:  # second chunk
                raise ValueError("boom")
            return x

    eng = ParallelEngine(cpu_chain(75, 25), auto_vram_balance=False)
    eng.setup(Boom())
    with pytest.raises(WorkerError) as ei:
        eng.forward(torch.zeros(4, 3), torch.zeros(4))
    assert "cpu" in str(ei.value)


def test_setup_oom_drops_device(monkeypatch):
    from comfyui_parallelanything_amd.parallel import engine as eng_mod

    calls = []

    real = eng_mod.replicate_module

    def fake_replicate(src, dev, force_copy=False, non_blocking=True):
        calls.append(dev)
        if dev == "meta":  # second device "OOMs"
            raise torch.cuda.OutOfMemoryError("fake OOM")
        return real(src, dev, force_copy, non_blocking)

    monkeypatch.setattr(eng_mod, "replicate_module", fake_replicate)
    m = nn.Linear(3, 3)
    chain = DeviceChain(devices=("cpu", "meta", "cpu"), weights=(0.4, 0.4, 0.2))
    eng = eng_mod.ParallelEngine(chain, auto_vram_balance=False)
    eng.setup(m)
    assert eng.chain.devices == ("cpu", "cpu")
    assert eng.chain.weights[0] == pytest.approx(40 / 60)


def test_install_uninstall_roundtrip(sd15):
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(sd15)
    orig_forward = sd15.forward
    install_parallel_forward(sd15, eng)
    assert sd15._true_parallel_active
    assert sd15._parallel_devices == ("cpu", "cpu")
    x, t, c, kw = sd15_inputs(2, tiny=True)
    out = sd15(x, t, context=c, **kw)  # goes through the engine
    assert out.shape == x.shape
    uninstall_parallel_forward(sd15)
    assert not hasattr(sd15, "_true_parallel_active")
    assert sd15.forward == orig_forward


def test_forward_without_context(sd15):
    x, t, _, _ = sd15_inputs(2, tiny=True)
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t)  # context omitted entirely
    torch.testing.assert_close(out, sd15(x, t), rtol=1e-4, atol=1e-5)


def test_release_clears_state(sd15):
    eng = ParallelEngine(cpu_chain(50, 50), auto_vram_balance=False)
    eng.setup(sd15)
    assert eng.replicas
    eng.release()
    assert not eng.replicas and not eng.streams and eng.pipeline is None


def test_golden_three_device_asymmetric(sd15):
    """70/20/10 [cpu,cpu,cpu]: min-1 floors + remainder; == single device."""
    x, t, c, kw = sd15_inputs(10, tiny=True)
    ref = sd15(x, t, context=c, **kw)
    eng = ParallelEngine(cpu_chain(70, 20, 10), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_golden_more_devices_than_batch_dp_threshold(sd15):
    """batch == n_devices: every device gets exactly one sample."""
    x, t, c, kw = sd15_inputs(3, tiny=True)
    ref = sd15(x, t, context=c, **kw)
    eng = ParallelEngine(cpu_chain(34, 33, 33), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-5)


def test_hip_graphs_cpu_falls_back_to_eager(sd15):
    """use_hip_graphs on a CPU chain must be a transparent no-op."""
    x, t, c, kw = sd15_inputs(2, tiny=True)
    ref = sd15(x, t, context=c, **kw)
    eng = ParallelEngine(
        cpu_chain(100), auto_vram_balance=False, use_hip_graphs=True
    )
    eng.setup(sd15)
    for _ in range(4):  # past WARMUP_CALLS: still must not try to capture
        out = eng.forward(x, t, context=c, **kw)
    assert torch.equal(out, ref)
    assert not eng.graphs._graphs
    eng.release()


def test_graph_runner_key_rules():
    from comfyui_parallelanything_amd.parallel.hipgraphs import GraphRunner

    r = GraphRunner()
    x, t = torch.zeros(2, 3), torch.zeros(2)
    k1 = r.key_for("cuda:0", x, t, None, {"y": torch.ones(2, 5)})
    k2 = r.key_for("cuda:0", x, t, None, {"y": torch.ones(2, 5)})
    assert k1 == k2 and k1 is not None
    # different shape, device, or dtype -> different signature
    assert r.key_for("cuda:0", torch.zeros(3, 3), t, None, {}) != k1
    assert r.key_for("cuda:1", x, t, None, {"y": torch.ones(2, 5)}) != k1
    assert (
        r.key_for("cuda:0", x.double(), t, None, {"y": torch.ones(2, 5)}) != k1
    )
    # non-tensor kwarg -> not graphable
    assert r.key_for("cuda:0", x, t, None, {"flag": True}) is None
    # same shape but different STRIDES (transposed view) -> different key:
    # a graph captured on contiguous buffers must not replay on a view
    # with different layout semantics
    sq = torch.zeros(3, 3)
    assert (
        r.key_for("cuda:0", sq.t(), t, None, {})
        != r.key_for("cuda:0", sq, t, None, {})
    )


def test_launch_order_gpu_before_cpu():
    """Hybrid chains: stream-backed devices enqueue before cpu workers
    (the cpu chunk computes while GPU streams are already busy)."""
    from comfyui_parallelanything_amd.parallel.chain import DeviceChain
    from comfyui_parallelanything_amd.parallel.engine import ParallelEngine

    eng = ParallelEngine(
        DeviceChain(
            devices=("cpu", "cuda:0", "cuda:1"), weights=(0.34, 0.33, 0.33)
        ),
        auto_vram_balance=False,
    )
    # simulate stream-backed GPUs without hardware
    eng.streams = {"cpu": None, "cuda:0": object(), "cuda:1": object()}
    order = eng._launch_order(["cpu", "cuda:0", "cuda:1"])
    assert order == [1, 2, 0]
    # all-cpu chains keep natural order
    eng.streams = {"cpu": None}
    assert eng._launch_order(["cpu", "cpu"]) == [0, 1]


def test_single_device_chain_routes_lead_only(sd15):
    x, t, c, kw = sd15_inputs(4, tiny=True)
    eng = ParallelEngine(cpu_chain(100), auto_vram_balance=False)
    eng.setup(sd15)
    out = eng.forward(x, t, context=c, **kw)
    assert torch.equal(out, sd15(x, t, context=c, **kw))
